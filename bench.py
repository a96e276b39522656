#!/usr/bin/env python3
"""Driver benchmark contract.

``python bench.py --gpus N --steps K --warmup W`` runs the flagship
training step — the Pendulum-v1 D4PG learner (51-atom C51 critic,
n-step=5, prioritized replay, batch 64) of BASELINE.json — on N GPUs of
one node, one rank per GPU (launched via torch.distributed.run for N>1),
on synthetic transitions and random-init weights.

N>1 is COMMUNICATION-BEARING weak scaling (not independent engines): the
default mode is the HogWild re-expression over RCCL — every rank owns a
full fused HIP learner engine and its own replay shard (the reference's
replay is per-worker too, main.py:188-197) and all ranks start from the
same broadcast parameters; every --sync_every local steps the parameter
slabs (actor, critic, both targets) are all-reduce-AVERAGED over xGMI and
a block of synthetic transitions is all_gathered and ingested by every
rank (the DistributedD4PG wire path, parallel/learner.py:_exchange).
Both collectives sit INSIDE the timed region.  This replaces the
reference's unbounded HogWild staleness (shared-memory params,
ddpg.py:104-120) with a bounded sync_every-step window; per-step gradient
sync (--mode dp, parallel/dp.py) is also available but serializes the
0.3 ms flagship step behind two collective latencies, so the bounded-
staleness mode is the default — see SURVEY.md §2b's xGMI latency note.

The value reported is the WHOLE-JOB aggregate learner grad-steps/sec:
each rank performs K real optimizer steps (HogWild accounting, same as
the reference's shared global_count, main.py:307), timed over the MAX
elapsed across ranks for EXACTLY K steps bracketed by barrier +
torch.cuda.synchronize on both sides.

Reference parity: the train step is the full D4PG update of
/root/reference/ddpg.py:200-255 (PER sample + IS weights, target
forwards, C51 categorical projection, cross-entropy critic loss +
backward + Adam, policy loss + backward + Adam, target soft-update, PER
priority writeback) — nothing is skipped inside the timed region.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np
import torch

FLAGSHIP = dict(obs_dim=3, act_dim=1, hidden=256, n_atoms=51, batch=64,
                capacity=1_000_000, v_min=-300.0, v_max=0.0,
                gamma=0.99, n_steps=5, tau=0.001,
                lr_actor=1e-4, lr_critic=1e-3)


def _dist_init(args):
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        import torch.distributed as dist
        backend = os.environ.get("BENCH_DIST_BACKEND")
        if backend is None:
            # nccl(=RCCL) needs one DISTINCT GPU per rank; a single-GPU
            # box rehearsing world>1 exchanges over gloo instead (engines
            # still run on the GPU)
            n_gpu = (torch.cuda.device_count()
                     if torch.cuda.is_available() else 0)
            backend = "nccl" if n_gpu >= world else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        try:
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world)
        except Exception:
            if backend != "gloo":
                dist.init_process_group(backend="gloo", rank=rank,
                                        world_size=world)
            else:
                raise
        return dist, world, rank, local
    return None, 1, 0, 0


def _barrier(dist):
    if dist is not None:
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def _max_over_ranks(dist, x: float) -> float:
    if dist is None:
        return x
    use_cuda = torch.cuda.is_available() and dist.get_backend() == "nccl"
    t = torch.tensor([x], dtype=torch.float64,
                     device="cuda" if use_cuda else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def make_gpu_engine(seed: int):
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    cfg = FLAGSHIP
    eng = FusedEngine(obs_dim=cfg["obs_dim"], act_dim=cfg["act_dim"],
                      hidden=cfg["hidden"], n_atoms=cfg["n_atoms"],
                      batch=cfg["batch"], capacity=cfg["capacity"],
                      v_min=cfg["v_min"], v_max=cfg["v_max"],
                      gamma_n=cfg["gamma"] ** cfg["n_steps"],
                      tau=cfg["tau"], lr_actor=cfg["lr_actor"],
                      lr_critic=cfg["lr_critic"], seed=seed)
    torch.manual_seed(seed)
    a = actor(cfg["obs_dim"], cfg["act_dim"], hidden=cfg["hidden"])
    c = critic(cfg["obs_dim"], cfg["act_dim"],
               {"type": "categorical", "v_min": cfg["v_min"],
                "v_max": cfg["v_max"], "n_atoms": cfg["n_atoms"]},
               hidden=cfg["hidden"])
    eng.load_from_modules(a, a, c, c)
    eng.synth_fill(cfg["capacity"], seed=seed + 7)
    return eng


def make_cpu_agent(seed: int):
    """Eager CPU fallback so `python bench.py` runs in a GPU-less
    container (numbers are then CPU numbers, not the MI355X headline)."""
    from d4pg_amd.algo.d4pg import DDPG
    cfg = FLAGSHIP
    agent = DDPG(cfg["obs_dim"], cfg["act_dim"],
                 memory_size=100000, batch_size=cfg["batch"],
                 gamma=cfg["gamma"], tau=cfg["tau"], prioritized_replay=True,
                 critic_dist_info={"type": "categorical",
                                   "v_min": cfg["v_min"],
                                   "v_max": cfg["v_max"],
                                   "n_atoms": cfg["n_atoms"]},
                 n_steps=cfg["n_steps"], device="cpu", backend="eager",
                 seed=seed)
    rng = np.random.default_rng(seed)
    for _ in range(5000):
        agent.replayBuffer.add(
            rng.standard_normal(cfg["obs_dim"]).astype("f"),
            rng.uniform(-1, 1, cfg["act_dim"]).astype("f"),
            -rng.random(), rng.standard_normal(cfg["obs_dim"]).astype("f"),
            0.0)
    return agent


def _synth_block(rng, n, obs_dim, act_dim):
    """One rank's synthetic transition block for the wire exchange
    (flat [n, 2*obs+act+2] rows, the _encode layout of parallel/learner)."""
    w = 2 * obs_dim + act_dim + 2
    arr = rng.standard_normal((n, w)).astype(np.float32)
    arr[:, obs_dim + act_dim] = -rng.random(n)          # rewards
    arr[:, -1] = 0.0                                    # dones
    return arr


class _WireExchange:
    """The actor->replay wire path inside the timed region: all_gather a
    fixed-size synthetic transition block from every rank, ingest all of
    them (mirrors DistributedD4PG._exchange over the same collectives)."""

    def __init__(self, dist_mod, world, rank, eng, block=512):
        self.dist = dist_mod
        self.world = world
        self.eng = eng
        cfg = FLAGSHIP
        self.o, self.a = cfg["obs_dim"], cfg["act_dim"]
        self.block = block
        self.rng = np.random.default_rng(10_000 + rank)
        use_cuda = (torch.cuda.is_available()
                    and dist_mod.get_backend() == "nccl")
        self.dev = torch.device("cuda" if use_cuda else "cpu")

    def round(self):
        arr = _synth_block(self.rng, self.block, self.o, self.a)
        buf = torch.from_numpy(arr).to(self.dev)
        blocks = [torch.empty_like(buf) for _ in range(self.world)]
        self.dist.all_gather(blocks, buf)
        o, a = self.o, self.a
        for b in blocks:
            rows = b.cpu().numpy()
            self.eng.ingest(torch.from_numpy(rows[:, :o]),
                            torch.from_numpy(rows[:, o:o + a]),
                            torch.from_numpy(rows[:, o + a]),
                            torch.from_numpy(rows[:, o + a + 1:2 * o + a + 1]),
                            torch.from_numpy(rows[:, 2 * o + a + 1]))


def run_gpu_multirank(dist_mod, world, rank, args):
    """Communication-bearing weak scaling: local-SGD parameter averaging +
    transition all_gather every --sync_every steps (module docstring)."""
    from d4pg_amd.parallel.dp import DPEngine, LocalSGDSync
    eng = make_gpu_engine(seed=1000 + rank)
    K, W, S = args.steps, args.warmup, max(1, args.sync_every)
    mode = args.mode
    if mode == "auto":
        mode = "localsgd"

    if mode == "dp":
        dp = DPEngine(eng)
        dp.train_steps(max(1, W))
        _barrier(dist_mod)
        t0 = time.perf_counter()
        dp.train_steps(K)
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        _barrier(dist_mod)
        return elapsed, "dp%d (sync grad all-reduce per step)" % world

    if mode == "indep":
        spg = args.steps_per_graph
        eng.train_steps(max(W, spg), steps_per_graph=spg)
        _barrier(dist_mod)
        t0 = time.perf_counter()
        eng.train_steps(K, steps_per_graph=spg)
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        _barrier(dist_mod)
        return elapsed, "indep%d (no communication; A/B baseline)" % world

    # localsgd (default): bounded-staleness HogWild over RCCL
    sync = LocalSGDSync(eng)
    wire = _WireExchange(dist_mod, world, rank, eng) \
        if args.xfer_transitions else None
    sync.broadcast_initial(src=0)

    def run_rounds(nsteps):
        done = 0
        while done < nsteps:
            n = min(S, nsteps - done)
            eng.step(n)                      # n local steps, one launch
            sync.average()                   # param all-reduce over xGMI
            if wire is not None:
                wire.round()                 # transition all_gather+ingest
            done += n

    run_rounds(max(W, S))
    _barrier(dist_mod)
    t0 = time.perf_counter()
    run_rounds(K)
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    _barrier(dist_mod)
    return elapsed, ("hogwild-rccl x%d (param avg + transition all_gather "
                     "every %d steps)" % (world, S))


def measure_env_steps_per_sec(seed: int, n: int = 3000) -> float:
    """Auxiliary (untimed-region) metric: actor-side env-steps/sec — native
    Pendulum dynamics + B=1 actor inference + exploration noise, the per-env-
    step path of /root/reference/main.py:142-152."""
    from d4pg_amd.envs import make
    from d4pg_amd.models import actor
    from d4pg_amd.noise import GaussianNoise
    env = make("Pendulum-v1", seed=seed)
    net = actor(3, 1)
    net.eval()
    noise = GaussianNoise(1, rng=np.random.default_rng(seed))
    obs = env.reset()
    with torch.no_grad():
        t0 = time.perf_counter()
        for _ in range(n):
            a = net(torch.as_tensor(obs, dtype=torch.float32)[None])[0]
            act = np.clip(a.numpy() + noise.sample(), -1.0, 1.0)
            obs, r, done, _ = env.step(act)
            if done:
                obs = env.reset()
        dt = time.perf_counter() - t0
    return n / dt


def measure_env_steps_vector(seed: int, m: int = 64, ticks: int = 1500):
    """Aux metric: vectorized-actor env-steps/sec — M batched native
    Pendulums + one [M,obs] policy forward per tick (the MI355X-native
    actor mode, parallel/learner.py --vector_envs)."""
    from d4pg_amd.envs.vector import VectorPendulum
    from d4pg_amd.models import actor
    env = VectorPendulum(m, seed=seed)
    net = actor(3, 1)
    net.eval()
    rng = np.random.default_rng(seed)
    obs = env.reset()
    with torch.no_grad():
        t0 = time.perf_counter()
        for t in range(ticks):
            a = net(torch.from_numpy(obs)).numpy()
            a = np.clip(a + 0.3 * rng.standard_normal(a.shape),
                        -1, 1).astype(np.float32)
            obs, r, done = env.step(a)
            if done:
                obs = env.reset()
        dt = time.perf_counter() - t0
    return ticks * m / dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4000)
    ap.add_argument("--warmup", type=int, default=400)
    ap.add_argument("--steps-per-graph", type=int, default=32)
    ap.add_argument("--mode", default="auto",
                    choices=["auto", "localsgd", "dp", "indep"],
                    help="N>1 coupling: localsgd = bounded-staleness "
                         "HogWild over RCCL (default), dp = per-step grad "
                         "all-reduce, indep = no communication (A/B only)")
    ap.add_argument("--sync_every", type=int, default=16,
                    help="localsgd: local steps between param averages")
    ap.add_argument("--xfer_transitions", type=int, default=1,
                    help="localsgd: all_gather+ingest a synthetic "
                         "transition block every sync round")
    args = ap.parse_args()

    dist, world, rank, local = _dist_init(args)
    n_gpus = max(world, args.gpus) if world > 1 else args.gpus
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        # modulo is the identity on a full node (one rank per GPU) and
        # lets single-GPU rehearsals run several ranks on device 0
        torch.cuda.set_device(local % torch.cuda.device_count())

    K, W = args.steps, args.warmup
    parallelism = "dp1 (one learner per GPU)"
    if use_gpu and world > 1:
        elapsed, parallelism = run_gpu_multirank(dist, world, rank, args)
    elif use_gpu:
        eng = make_gpu_engine(seed=1000 + rank)
        spg = args.steps_per_graph
        eng.train_steps(max(W, spg), steps_per_graph=spg)  # warmup+capture
        _barrier(dist)
        t0 = time.perf_counter()
        eng.train_steps(K, steps_per_graph=spg)
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        _barrier(dist)
    else:
        K = min(K, 400)
        W = min(W, 20)
        agent = make_cpu_agent(seed=1000 + rank)
        if world > 1:
            # CPU rehearsal of the multi-rank topology: eager agents with
            # per-step gradient averaging over gloo (parallel/dp.py hook)
            from d4pg_amd.parallel.dp import eager_grad_sync
            agent.grad_sync = eager_grad_sync()
            parallelism = "dp%d (eager, gloo rehearsal)" % world
        for _ in range(W):
            agent.train()
        _barrier(dist)
        t0 = time.perf_counter()
        for _ in range(K):
            agent.train()
        elapsed = time.perf_counter() - t0
        _barrier(dist)

    elapsed = _max_over_ranks(dist, elapsed)
    env_sps = measure_env_steps_per_sec(seed=1234 + rank) if rank == 0 else 0.0
    vec_sps = measure_env_steps_vector(seed=4321 + rank) if rank == 0 else 0.0
    gpu_sps = None
    if rank == 0 and use_gpu:
        # device-resident actor serving (env + policy + noise + fold all
        # on-GPU, engine.hip rollout section): M=8192 envs, 200-tick
        # episodes, one hipGraph per episode.  Throughput keeps scaling
        # with M (211M env-steps/s at M=65536, profiles/README.md); 8192
        # is a practical per-actor-rank size.
        eng2 = make_gpu_engine(seed=999)
        eng2.rollout_alloc(8192, FLAGSHIP["n_steps"], horizon=200,
                           gamma=FLAGSHIP["gamma"], eps=0.3, seed=555)
        eng2.rollout_run(1)                       # capture + warm
        t0 = time.perf_counter()
        steps_done, _ = eng2.rollout_run(6)
        gpu_sps = steps_done / (time.perf_counter() - t0)

    if rank == 0:
        # In every mode each rank performs K real optimizer steps, so the
        # whole-job aggregate is N*K grad steps (HogWild accounting — the
        # reference's shared global_count sums worker steps, main.py:307).
        # EXCEPT dp mode: there the N per-step gradients merge into ONE
        # global step at global batch N*B, so the job made K steps.
        is_dp = parallelism.startswith("dp") and world > 1
        total_steps = K if is_dp else n_gpus * K
        out = {
            "metric": "learner grad-steps/sec (Pendulum-v1 D4PG)",
            "value": total_steps / elapsed,
            "unit": "grad_steps/s",
            "n_gpus": n_gpus,
            "steps": K,
            "warmup": W,
            "ms_per_step": elapsed / K * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (random-init weights, synthetic replay "
                    "transitions; reference publishes no numbers)",
            "config": {"model": "D4PG Pendulum-v1 (obs3/act1, 4x256 MLP "
                                "actor+critic, 51 atoms)",
                       "global_batch": FLAGSHIP["batch"] *
                                       (n_gpus if is_dp else 1),
                       "seq_len": None,
                       "parallelism": parallelism,
                       "n_step": FLAGSHIP["n_steps"],
                       "prioritized_replay": True,
                       "replay_capacity": FLAGSHIP["capacity"],
                       "device": "cuda" if use_gpu else "cpu-fallback"},
            "env_steps_per_sec_1actor": env_sps,
            "env_steps_per_sec_vector64": vec_sps,
            "env_steps_per_sec_gpu_rollout8192": gpu_sps,
        }
        print(json.dumps(out), flush=True)
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
