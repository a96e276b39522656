"""Distributed D4PG over torch.distributed (RCCL on ROCm, gloo on CPU).

MI355X-native re-expression of the reference's shared-memory topology
(/root/reference/main.py:382-405): instead of HogWild
workers aliasing gradients into shared host memory, one process per GPU
with an explicit role split —

  rank 0            learner: owns params, Adam state, targets and the
                    ENTIRE prioritized replay (on-HBM via the fused HIP
                    engine when a GPU is present; eager-torch otherwise)
  ranks 1..W-1      actor ranks: envs + exploration noise + n-step fold
                    + HER relabel (reference main.py:137-184 semantics),
                    pure inference clients
  last rank         optional evaluator: greedy rollouts, EWMA return
                    (reference main.py:103-134 semantics)

Round-based schedule with a bounded staleness window (one round):

  1. broadcast(actor params ++ global step) learner -> all.  The blob is
     ~0.5 MB fp32 — latency-bound on xGMI, so one flat broadcast per
     round, not per train step (reference pulled params EVERY step via
     load_state_dict, ddpg.py:118-120 — pointless on a GPU fabric).
  2. each actor rank collects `episodes_per_round` episodes.
  3. fixed-size SoA all_gather (counts, then transition blocks)
     actor -> learner.  all_gather (not gather) so the same code runs on
     both RCCL and gloo.
  4. learner ingests + runs `train_steps_per_cycle` grad steps while the
     actors immediately start the next collection window (compute/env
     overlap happens across the round boundary).

The global step counter rides in the broadcast blob — replacing the
reference's non-atomic shared `global_count += 1` (main.py:307) with a
learner-owned monotonic counter.
"""

from __future__ import annotations

import os
import time

import numpy as np
import torch
import torch.distributed as dist

from ..algo.d4pg import DDPG
from ..config import critic_dist_info, noise_kwargs, run_dir_name
from ..envs import make, obs_act_dims
from ..her import add_experience, rollout_episode
from ..ops import pack_net, unpack_net
from ..utils.logging import Meter, SummaryWriter


class _ListBuffer:
    """Replay-interface shim that records transitions (scalar adds or
    whole [M,...] batches), so the actor side can reuse add_experience
    (n-step fold + HER relabel) or the vectorized collector and ship the
    result over the wire instead of into a local buffer."""

    def __init__(self):
        self.items = []
        self.batches = []

    def add(self, state, action, reward, next_state, done):
        self.items.append((np.asarray(state, np.float32).ravel(),
                           np.asarray(action, np.float32).ravel(),
                           np.float32(reward),
                           np.asarray(next_state, np.float32).ravel(),
                           np.float32(done)))

    def add_batch(self, S, A, R, S2, D):
        self.batches.append((np.asarray(S, np.float32),
                             np.asarray(A, np.float32),
                             np.asarray(R, np.float32).ravel(),
                             np.asarray(S2, np.float32),
                             np.asarray(D, np.float32).ravel()))

    def to_arrays(self, obs_dim, act_dim):
        parts = list(self.batches)
        if self.items:
            parts.append((
                np.stack([p[0] for p in self.items]),
                np.stack([p[1] for p in self.items]),
                np.asarray([p[2] for p in self.items], np.float32),
                np.stack([p[3] for p in self.items]),
                np.asarray([p[4] for p in self.items], np.float32)))
        if not parts:
            z = np.zeros((0, obs_dim), np.float32)
            za = np.zeros((0, act_dim), np.float32)
            zr = np.zeros(0, np.float32)
            return z, za, zr, z.copy(), zr.copy()
        return tuple(np.concatenate([p[i] for p in parts])
                     for i in range(5))

    def __len__(self):
        return len(self.items) + sum(len(b[2]) for b in self.batches)


def _row_width(obs_dim: int, act_dim: int) -> int:
    return 2 * obs_dim + act_dim + 2


def _encode(lb, cap, obs_dim, act_dim, device):
    buf = torch.zeros(cap, _row_width(obs_dim, act_dim), device=device)
    S, A, R, S2, D = (lb.to_arrays(obs_dim, act_dim)
                      if hasattr(lb, "to_arrays") else lb)
    n = min(len(R), cap)
    if n:
        o, a = obs_dim, act_dim
        arr = np.empty((n, _row_width(o, a)), np.float32)
        arr[:, :o] = S[:n]
        arr[:, o:o + a] = A[:n]
        arr[:, o + a] = R[:n]
        arr[:, o + a + 1:2 * o + a + 1] = S2[:n]
        arr[:, 2 * o + a + 1] = D[:n]
        buf[:n] = torch.from_numpy(arr).to(device)
    return buf, n


def _decode(buf, n, obs_dim, act_dim):
    o, a = obs_dim, act_dim
    arr = buf[:n].cpu().numpy()
    return (arr[:, :o], arr[:, o:o + a], arr[:, o + a],
            arr[:, o + a + 1:2 * o + a + 1], arr[:, 2 * o + a + 1])


class DistributedD4PG:
    """One rank of the distributed actor/learner topology.  Construct on
    every rank with the same args, then call run(rounds)."""

    def __init__(self, args, rank=None, world=None, device=None,
                 with_evaluator=True, writer=None):
        self.args = args
        self.rank = int(os.environ.get("RANK", "0")) if rank is None else rank
        self.world = (int(os.environ.get("WORLD_SIZE", "1"))
                      if world is None else world)
        if device is None:
            if torch.cuda.is_available():
                local = int(os.environ.get("LOCAL_RANK", str(self.rank)))
                torch.cuda.set_device(local % torch.cuda.device_count())
                device = "cuda"
            else:
                device = "cpu"
        # collective-buffer device must match the process-group backend
        # (gloo = host memory; ranks can share one GPU for compute while
        # exchanging over gloo — the single-GPU-box topology)
        if dist.is_initialized() and dist.get_backend() != "nccl":
            self.comm_device = torch.device("cpu")
        else:
            self.comm_device = torch.device(device)
        self.device = torch.device(device)
        self.is_learner = self.rank == 0
        self.eval_rank = (self.world - 1
                          if (with_evaluator and self.world >= 3) else -1)
        self.is_evaluator = self.rank == self.eval_rank

        seed = (args.seed or 0) + 7919 * self.rank
        self.env = make(args.env, seed=seed)
        self.env._max_episode_steps = args.max_steps
        self.obs_dim, self.act_dim = obs_act_dims(self.env,
                                                  her=bool(args.her))
        backend = "eager"
        if self.is_learner and self.device.type == "cuda":
            backend = "hip"
        self.agent = DDPG(
            self.obs_dim, self.act_dim, env=self.env,
            memory_size=args.rmsize if self.is_learner else 1,
            batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
            lr_actor=getattr(args, "lr_actor", 1e-4),
            lr_critic=getattr(args, "lr_critic", 1e-3),
            prioritized_replay=bool(args.p_replay) and self.is_learner,
            critic_dist_info=critic_dist_info(args), n_steps=args.n_steps,
            device=str(self.device) if self.is_learner else "cpu",
            backend=backend, seed=seed, **noise_kwargs(args))
        self.rng = np.random.default_rng(seed)

        # wire geometry: per-round per-rank transition cap.  HER can add up
        # to 2x (real + hindsight) transitions per step.
        self.episodes_per_round = max(
            1, args.episodes_per_cycle // max(1, self.world - 1))
        mult = 2 if args.her else 1
        self.push_cap = self.episodes_per_round * args.max_steps * mult
        # vectorized actor mode: M batched envs per actor rank, one [M,obs]
        # policy forward per tick (Pendulum only; HER needs dict obs)
        self.vector = None
        if (int(getattr(args, "vector_envs", 0)) > 0 and not args.her
                and args.env.startswith("Pendulum")
                and not self.is_learner and not self.is_evaluator):
            from ..envs.vector import VecNStep, VectorPendulum
            m = int(args.vector_envs)
            self.vector = VectorPendulum(m, seed=seed + 17,
                                         horizon=args.max_steps)
            self.vfold = VecNStep(m, self.obs_dim, self.act_dim,
                                  args.n_steps, args.gamma)
        # GPU-resident actor serving (SURVEY K11 + §2c): when this actor
        # rank has a GPU, the whole collection loop — vectorized Pendulum
        # dynamics, fused actor forward, exploration noise, n-step fold —
        # runs as device kernels (engine.hip rollout section), one hipGraph
        # per episode; the host only ships the matured transitions to the
        # learner.  Re-expresses /root/reference/main.py:142-152.
        self.roll_engine = None
        ga = int(getattr(args, "gpu_actors", -1))
        if (self.vector is not None and ga != 0
                and torch.cuda.is_available()):
            from ..ops import FusedEngine
            m = int(args.vector_envs)
            cap = m * max(1, args.max_steps - args.n_steps + 1)
            self.roll_engine = FusedEngine(
                obs_dim=self.obs_dim, act_dim=self.act_dim, hidden=256,
                n_atoms=args.n_atoms, batch=64, capacity=cap,
                v_min=args.v_min, v_max=args.v_max,
                gamma_n=args.gamma ** args.n_steps, tau=args.tau,
                lr_actor=1e-4, lr_critic=1e-4, seed=seed + 31)
            self.roll_engine.rollout_alloc(
                m, args.n_steps, horizon=args.max_steps, gamma=args.gamma,
                noise=getattr(args, "noise", "gaussian"),
                eps=getattr(args, "noise_eps", 0.3),
                ou_theta=args.ou_theta, ou_sigma=args.ou_sigma,
                ou_mu=args.ou_mu, seed=seed + 37)
        # vector-actor exploration noise honors --noise/--noise_eps/--ou_*
        # (this path used to hardcode eps=0.3).  OU
        # runs batched over [M, act]; Gaussian stays a one-liner below.
        self.noise_eps = float(getattr(args, "noise_eps", 0.3))
        self.vec_noise = None
        if (self.vector is not None
                and getattr(args, "noise", "gaussian") == "ou"):
            from ..noise import OrnsteinUhlenbeckProcess
            self.vec_noise = OrnsteinUhlenbeckProcess(
                (self.vector.n, self.act_dim), mu=args.ou_mu,
                theta=args.ou_theta, sigma=args.ou_sigma,
                rng=np.random.default_rng(seed + 29))
        if int(getattr(args, "vector_envs", 0)) > 0 and not args.her \
                and args.env.startswith("Pendulum"):
            # every rank must agree on the wire size
            self.push_cap = int(args.vector_envs) * args.max_steps
        self.blob_len = pack_net(self.agent.actor).numel() + 1

        self.grad_meter = Meter()
        self.env_meter = Meter()
        self.global_step = 0
        self.ewma = None
        self.writer = writer
        self.run_dir = run_dir_name(args)
        # failure visibility: the reference silently loses
        # throughput when a worker dies; here the learner tracks per-rank
        # heartbeats (a rank that gathers 0 transitions for many
        # consecutive rounds is flagged).
        self.last_seen = {r: 0 for r in range(1, self.world)}
        self.heartbeat_warn_rounds = 10
        # wire-cap overflow accounting (see _exchange)
        self.dropped_transitions = 0

    # -- round phases -----------------------------------------------------

    def _broadcast_params(self):
        if self.is_learner:
            if self.agent.backend == "hip" and self.agent.engine is not None:
                self.agent.engine.sync_params_if_dirty()
            blob = torch.cat([pack_net(self.agent.actor),
                              torch.tensor([float(self.global_step)])])
            blob = blob.to(self.comm_device)
        else:
            blob = torch.zeros(self.blob_len, device=self.comm_device)
        dist.broadcast(blob, src=0)
        if not self.is_learner:
            blob = blob.cpu()
            unpack_net(self.agent.actor, blob[:-1])
            self.global_step = int(blob[-1].item())

    def _collect(self):
        lb = _ListBuffer()
        if self.is_learner:
            return lb
        if self.is_evaluator:
            _, R, _ = rollout_episode(self.agent, self.env, noise=False)
            self.ewma = R if self.ewma is None else \
                0.95 * self.ewma + 0.05 * R
            print(f"[eval] step {self.global_step} return {R:.2f} "
                  f"ewma {self.ewma:.2f}", flush=True)
            return lb
        if self.roll_engine is not None:
            # device rollout: load the freshly broadcast policy into the
            # rollout engine's actor slab, run one on-device episode, ship
            # the already-n-step-folded transitions
            from ..ops import pack_net as _pack
            self.roll_engine.load_slab("actor", _pack(self.agent.actor))
            env_steps, _ = self.roll_engine.rollout_run(1)
            s, a, r, s2, d = self.roll_engine.replay_rows()
            lb.add_batch(s, a, r, s2, d)
            self.env_meter.add(env_steps)
            return lb
        if self.vector is not None:
            import torch as _t
            obs = self.vector.reset()
            self.vfold.reset()
            actor = self.agent.actor
            if self.vec_noise is not None:
                self.vec_noise.reset()
            with _t.no_grad():
                for t in range(self.vector.horizon):
                    a = actor(_t.from_numpy(obs)).numpy()
                    if self.vec_noise is not None:   # batched OU [M, act]
                        a = a + self.vec_noise.sample()
                    else:
                        a = a + (self.noise_eps
                                 * self.rng.standard_normal(a.shape))
                    a = np.clip(a, -1.0, 1.0).astype(np.float32)
                    obs2, r, done = self.vector.step(a)
                    out = self.vfold.push(obs, a, r, obs2, done)
                    if out is not None:
                        lb.add_batch(*out)
                    obs = obs2
            self.env_meter.add(self.vector.horizon * self.vector.n)
            return lb
        for _ in range(self.episodes_per_round):
            ep, _, _ = rollout_episode(self.agent, self.env, noise=True)
            self.env_meter.add(len(ep))
            add_experience(lb, self.env, ep, her=bool(self.args.her),
                           n_steps=self.args.n_steps, gamma=self.args.gamma,
                           rng=self.rng)
        return lb

    def _exchange(self, lb):
        dev = self.comm_device
        buf, n = _encode(lb, self.push_cap, self.obs_dim,
                         self.act_dim, dev)
        # the advertised count must match what _encode actually shipped:
        # len(lb) > push_cap means the wire buffer truncated (e.g. an env /
        # HER change emitting more transitions per round than the cap was
        # sized for) — count the drop and say so instead of letting decode
        # and counts silently disagree.
        if len(lb) > n:
            self.dropped_transitions += len(lb) - n
            print(f"[rank {self.rank}] WARNING: push_cap {self.push_cap} "
                  f"truncated {len(lb) - n} of {len(lb)} transitions this "
                  f"round ({self.dropped_transitions} dropped total) — "
                  f"raise --episodes_per_cycle sizing or vector_envs cap",
                  flush=True)
        cnt = torch.tensor([float(n)], device=dev)
        counts = [torch.zeros_like(cnt) for _ in range(self.world)]
        dist.all_gather(counts, cnt)
        self._last_counts = counts
        blocks = [torch.zeros_like(buf) for _ in range(self.world)]
        dist.all_gather(blocks, buf)
        if not self.is_learner:
            return 0
        total = 0
        for r in range(1, self.world):
            n_r = int(counts[r].item())
            if n_r == 0:
                continue
            s, a, rw, s2, d = _decode(blocks[r], n_r, self.obs_dim,
                                      self.act_dim)
            buf = self.agent.replayBuffer
            if hasattr(buf, "add_batch"):
                buf.add_batch(s, a, rw, s2, d)
            else:
                for i in range(n_r):
                    buf.add(s[i], a[i], rw[i], s2[i], d[i])
            total += n_r
        return total

    def _train(self):
        if not self.is_learner:
            return
        floor = max(self.args.bsize,
                    getattr(self.args, "warmup", 0) * self.args.max_steps)
        if len(self.agent.replayBuffer) < floor:
            return
        n = self.args.train_steps_per_cycle
        if self.agent.backend == "hip":
            if self.agent.engine is None:
                from ..ops import build_fused_engine
                self.agent._fused = build_fused_engine(self.agent)
            # go through the bridge (flush + multi-step + params-dirty flag;
            # bypassing it once silently broadcast STALE initial actor
            # params every round — the dirty flag is what makes
            # sync_params_if_dirty pull the trained weights back)
            self.agent.engine.step(n=n)
        else:
            for _ in range(n):
                self.agent.train()
        self.grad_meter.add(n)
        self.global_step += n

    # -- main loop --------------------------------------------------------

    def _heartbeat(self, counts, rnd):
        for r in range(1, self.world):
            if r == self.eval_rank:
                continue
            if int(counts[r].item()) > 0:
                self.last_seen[r] = rnd
            elif rnd - self.last_seen[r] >= self.heartbeat_warn_rounds:
                print(f"[learner] WARNING: actor rank {r} silent for "
                      f"{rnd - self.last_seen[r]} rounds (dead actor? the "
                      f"learner keeps training on remaining actors)",
                      flush=True)

    def run(self, rounds: int, save: bool = False):
        interval = max(1, int(getattr(self.args, "broadcast_interval", 1)))
        for rnd in range(rounds):
            # --broadcast_interval K: refresh actor params every K rounds
            # (bounded staleness knob; K=1 = every round)
            if rnd % interval == 0:
                self._broadcast_params()
            lb = self._collect()
            ingested = self._exchange(lb)
            if self.is_learner:
                self._heartbeat(self._last_counts, rnd)
            self._train()
            if self.is_learner:
                if self.writer is not None:
                    self.writer.add_scalar("grad_steps_per_sec",
                                           self.grad_meter.rate(),
                                           self.global_step)
                    self.writer.add_scalar("replay_occupancy",
                                           len(self.agent.replayBuffer),
                                           self.global_step)
                if self.args.debug:
                    print(f"[learner] round {rnd} step {self.global_step} "
                          f"ingested {ingested} "
                          f"replay {len(self.agent.replayBuffer)}",
                          flush=True)
                if save and self.run_dir:
                    os.makedirs(self.run_dir, exist_ok=True)
                    self.agent.save(self.run_dir)
        # final param sync so every rank ends with the trained policy
        self._broadcast_params()
        return self.global_step


def init_process_group(backend: str | None = None):
    if dist.is_initialized():
        return
    if backend is None:
        backend = os.environ.get("D4PG_DIST_BACKEND")
    if backend is None:
        # nccl needs one DISTINCT GPU per rank; on a single-GPU box with
        # several ranks (learner + CPU actor ranks) fall back to gloo
        n_gpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
        world = int(os.environ.get("WORLD_SIZE", "1"))
        backend = "nccl" if n_gpu >= world else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29521")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group(backend=backend)


def run_distributed(args, rounds: int | None = None, with_evaluator=True):
    """Entry point for `torch.distributed.run ... -m d4pg_amd.parallel.learner`
    or direct invocation from tests (with env RANK/WORLD_SIZE preset)."""
    init_process_group()
    rank = dist.get_rank()
    writer = None
    if rank == 0:
        writer = SummaryWriter(run_dir_name(args))
    node = DistributedD4PG(args, rank=rank, world=dist.get_world_size(),
                           with_evaluator=with_evaluator, writer=writer)
    t0 = time.perf_counter()
    n_rounds = rounds if rounds is not None else args.n_eps * args.cycles_per_epoch
    step = node.run(n_rounds, save=(rank == 0))
    dt = time.perf_counter() - t0
    if rank == 0:
        print(f"[learner] done: {step} grad steps in {dt:.1f}s "
              f"({step / max(dt, 1e-9):.1f} steps/s)", flush=True)
    elif not node.is_evaluator:
        print(f"[actor {rank}] done: {node.env_meter.count} env steps in "
              f"{dt:.1f}s ({node.env_meter.count / max(dt, 1e-9):.1f} "
              f"env-steps/s)", flush=True)
    dist.barrier()
    dist.destroy_process_group()
    return step


def main(argv=None):
    from ..config import configure_env_params, make_parser
    p = make_parser()
    p.add_argument("--rounds", type=int, default=None)
    args = p.parse_args(argv)
    configure_env_params(args)
    run_distributed(args, rounds=args.rounds)


if __name__ == "__main__":
    main()
