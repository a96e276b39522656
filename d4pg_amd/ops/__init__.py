"""HIP extension loader + the fused-engine wrapper.

The compiled extension (`_d4pg_hip.so`, built in-tree by setup.py /
__graft_entry__.build()) owns all GPU state: parameter slabs (weights stored
transposed [in][out]), Adam moments, target slabs, the on-HBM PER sum/min
trees and SoA replay store, device RNG and schedule counters.  Python only
packs/unpacks torch state_dicts into flat slabs and drives
step/capture/replay.

Policy: on a GPU machine the HIP path is mandatory — a missing or broken
extension raises ImportError loudly (no silent eager fallback; eager exists
for CPU tests only).
"""

from __future__ import annotations

import glob
import importlib.util
import os

import numpy as np
import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _find_so():
    here = os.path.dirname(os.path.abspath(__file__))
    cands = sorted(glob.glob(os.path.join(here, "_d4pg_hip*.so")))
    return cands[0] if cands else None


def load_extension():
    """Import the in-tree compiled extension (no JIT — the .so must have
    been built by setup.py / __graft_entry__.build())."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    so = _find_so()
    if so is None:
        _EXT_ERR = ImportError(
            "_d4pg_hip.so not found — build it with "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build())")
        raise _EXT_ERR
    spec = importlib.util.spec_from_file_location("_d4pg_hip", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _EXT = mod
    return mod


def extension_available() -> bool:
    try:
        load_extension()
        return True
    except Exception:
        return False


# ---------------------------------------------------------------------------
# slab packing: torch module (fc1/fc2/fc2_2/fc3) <-> flat engine slab
# ---------------------------------------------------------------------------

_LAYERS = ["fc1", "fc2", "fc2_2", "fc3"]


def pack_net(module: torch.nn.Module) -> torch.Tensor:
    """Flatten a 4-layer MLP into the engine slab layout:
    per layer, transposed weight [in][out] then bias [out]."""
    parts = []
    for name in _LAYERS:
        lin = getattr(module, name)
        parts.append(lin.weight.detach().t().contiguous().reshape(-1))
        parts.append(lin.bias.detach().reshape(-1))
    return torch.cat(parts).to(torch.float32).cpu()


def unpack_net(module: torch.nn.Module, flat: torch.Tensor) -> None:
    off = 0
    with torch.no_grad():
        for name in _LAYERS:
            lin = getattr(module, name)
            o, i = lin.weight.shape
            w = flat[off:off + i * o].reshape(i, o).t()
            off += i * o
            b = flat[off:off + o]
            off += o
            lin.weight.copy_(w)
            lin.bias.copy_(b)
    assert off == flat.numel(), "slab size mismatch on unpack"


# ---------------------------------------------------------------------------
# GPU replay adapter (host-side buffering -> batched device ingestion)
# ---------------------------------------------------------------------------

class GPUReplayAdapter:
    """Drop-in for the learner's replayBuffer when the replay lives on-HBM.
    add() buffers host-side; flush() ships one batched ingest (H2D + a
    single level-synced tree-insert kernel)."""

    def __init__(self, engine: "FusedEngine"):
        self.engine = engine
        self._pending = []
        self._batches = []

    def add(self, state, action, reward, next_state, done):
        self._pending.append((
            np.asarray(state, np.float32).ravel(),
            np.asarray(action, np.float32).ravel(),
            np.float32(reward),
            np.asarray(next_state, np.float32).ravel(),
            np.float32(done)))
        if len(self._pending) >= 32768:
            self.flush()

    def add_batch(self, states, actions, rewards, next_states, dones):
        """Array-form ingestion (the distributed learner's exchange path
        decodes whole SoA blocks — a per-transition python loop was the
        learner's bottleneck at high actor counts)."""
        self._batches.append((
            np.asarray(states, np.float32),
            np.asarray(actions, np.float32),
            np.asarray(rewards, np.float32).ravel(),
            np.asarray(next_states, np.float32),
            np.asarray(dones, np.float32).ravel()))

    def flush(self):
        parts = list(self._batches)
        self._batches.clear()
        if self._pending:
            parts.append((
                np.stack([p[0] for p in self._pending]),
                np.stack([p[1] for p in self._pending]),
                np.asarray([p[2] for p in self._pending], np.float32),
                np.stack([p[3] for p in self._pending]),
                np.asarray([p[4] for p in self._pending], np.float32)))
            self._pending.clear()
        for s, a, r, s2, d in parts:
            self.engine.ingest(torch.from_numpy(s), torch.from_numpy(a),
                               torch.from_numpy(r), torch.from_numpy(s2),
                               torch.from_numpy(d))

    def __len__(self):
        return (int(self.engine.counters()["size"]) + len(self._pending)
                + sum(len(b[2]) for b in self._batches))

    # -- exact resume (SURVEY §5 checkpoint row): snapshot/restore the
    # on-HBM SoA store + sum/min trees --
    def state_dict(self):
        self.flush()
        return dict(self.engine.ext.replay_state(self.engine.h))

    def load_state_dict(self, st):
        self._pending.clear()
        self.engine.ext.load_replay_state(
            self.engine.h, st["s"], st["a"], st["r"], st["s2"], st["d"],
            st["sum_tree"], st["min_tree"], int(st["size"]), int(st["pos"]),
            float(st["max_priority"]))


# ---------------------------------------------------------------------------
# the fused engine wrapper
# ---------------------------------------------------------------------------

class FusedEngine:
    """Owns one device-side D4PG learner (see ops/hip/engine.hip)."""

    def __init__(self, obs_dim, act_dim, hidden, n_atoms, batch, capacity,
                 v_min, v_max, gamma_n, tau, lr_actor, lr_critic,
                 per_alpha=0.6, per_beta0=0.4, per_beta_iters=100000,
                 per_eps=1e-6, seed=0, is_weighting=False):
        self.ext = load_extension()
        self.h = self.ext.create(
            obs_dim, act_dim, hidden, n_atoms, batch, capacity,
            float(v_min), float(v_max), float(gamma_n), float(tau),
            float(lr_actor), float(lr_critic), float(per_alpha),
            float(per_beta0), int(per_beta_iters), float(per_eps),
            int(seed), bool(is_weighting))
        self.batch = batch
        self._captured = 0
        self._persistent = bool(self.ext.info(self.h).get("persistent", False))

    def __del__(self):
        try:
            self.ext.destroy(self.h)
        except Exception:
            pass

    # -- params --
    SLABS = {"actor": 0, "actor_target": 1, "critic": 2, "critic_target": 3,
             "g_actor": 4, "g_critic": 5, "m_actor": 6, "v_actor": 7,
             "m_critic": 8, "v_critic": 9}

    # split-step phase masks (engine.hip enum): the learner-DP path runs
    # GRADS, all-reduces the grad slab over RCCL, then runs APPLY.
    PH_CRITIC_GRADS = 1
    PH_CRITIC_APPLY = 2
    PH_ACTOR_GRADS = 4
    PH_ACTOR_APPLY = 8
    PH_ALL = 15

    def load_from_modules(self, actor, actor_target, critic, critic_target):
        self.ext.load_slab(self.h, 0, pack_net(actor))
        self.ext.load_slab(self.h, 1, pack_net(actor_target))
        self.ext.load_slab(self.h, 2, pack_net(critic))
        self.ext.load_slab(self.h, 3, pack_net(critic_target))

    def store_to_modules(self, actor, actor_target, critic, critic_target):
        unpack_net(actor, self.ext.store_slab(self.h, 0))
        unpack_net(actor_target, self.ext.store_slab(self.h, 1))
        unpack_net(critic, self.ext.store_slab(self.h, 2))
        unpack_net(critic_target, self.ext.store_slab(self.h, 3))

    def load_slab(self, name, flat):
        self.ext.load_slab(self.h, self.SLABS[name], flat)

    def store_slab(self, name):
        return self.ext.store_slab(self.h, self.SLABS[name])

    # -- replay --
    def synth_fill(self, n, seed=1234):
        self.ext.synth_fill(self.h, int(n), int(seed))

    # device staging buffer rows per ingest call (engine.hip ing_cap)
    INGEST_CHUNK = 65536

    def ingest(self, s, a, r, s2, d):
        """Append transitions; chunks transparently at the engine's
        staging-buffer capacity (large actor pushes — e.g. GPU-rollout
        ranks at hundreds of envs — exceed one staging buffer)."""
        n = int(r.numel() if hasattr(r, "numel") else len(r))
        if n <= self.INGEST_CHUNK:
            self.ext.ingest(self.h, s, a, r, s2, d)
            return
        for lo in range(0, n, self.INGEST_CHUNK):
            hi = min(n, lo + self.INGEST_CHUNK)
            self.ext.ingest(self.h, s[lo:hi], a[lo:hi], r[lo:hi],
                            s2[lo:hi], d[lo:hi])

    def device_slab(self, name) -> torch.Tensor:
        """Zero-copy torch CUDA view of an engine slab — feed directly to
        torch.distributed collectives (RCCL).  Aliases engine memory."""
        return self.ext.device_tensor(self.h, self.SLABS[name])

    # -- stepping --
    def step(self, n=1):
        """Uncaptured (eager-launch) steps — used by parity tests."""
        self.ext.step(self.h, int(n))

    def step_part(self, mask):
        """Run only the PH_* phases of one train step (synchronizing), so
        the caller can all-reduce gradient slabs between GRADS and APPLY
        (learner data parallelism, SURVEY §2c collectives list)."""
        self.ext.step_part(self.h, int(mask))

    def train_steps(self, n=1, steps_per_graph=8):
        """Graph-replayed steps: captures once (steps_per_graph per replay),
        then replays; the remainder runs uncaptured.  On the persistent-
        megakernel path this is a single multi-step launch instead (no graph
        needed — the step loop lives inside the kernel)."""
        if self._persistent:
            self.ext.step(self.h, int(n))
            return
        if self._captured != steps_per_graph:
            self.ext.capture(self.h, steps_per_graph)
            self._captured = steps_per_graph
        full, rem = divmod(int(n), steps_per_graph)
        if full:
            self.ext.replay(self.h, full)
        if rem:
            self.ext.step(self.h, rem)

    def train_steps_async(self, n, steps_per_graph=8):
        if self._captured != steps_per_graph:
            self.ext.capture(self.h, steps_per_graph)
            self._captured = steps_per_graph
        assert n % steps_per_graph == 0
        self.ext.replay_async(self.h, n // steps_per_graph)

    def sync(self):
        self.ext.sync(self.h)

    # -- introspection / resume --
    def counters(self):
        return self.ext.counters(self.h)

    def set_schedule(self, steps_done, max_priority=1.0):
        """Restore schedule counters to 'steps_done completed' (resume).
        Counters live in device memory (read by the kernels each step), so
        no graph invalidation is needed here."""
        self.ext.set_schedule(self.h, int(steps_done), float(max_priority))

    def set_seed(self, seed):
        """Reseed the device philox stream.  The seed rides in captured
        kernel arguments, so this invalidates any captured hipGraph and
        forces a recapture on the next train_steps()."""
        self.ext.set_seed(self.h, int(seed))
        self._captured = 0

    def read(self, name):
        return self.ext.read_buffer(self.h, name)

    def actor_forward(self, x):
        return self.ext.actor_forward(self.h, x)

    # -- GPU-resident rollout (device Pendulum env + K11 noise + K15 fold,
    # transitions written straight into this engine's on-HBM replay;
    # see engine.hip "GPU-resident actor rollout") --
    def rollout_alloc(self, n_envs, n_steps, horizon=200, gamma=0.99,
                      noise="gaussian", eps=0.3, ou_theta=0.15,
                      ou_sigma=0.2, ou_mu=0.0, seed=0):
        kind = {"gaussian": 0, "ou": 1}[noise]
        self.ext.rollout_alloc(self.h, int(n_envs), int(n_steps),
                               int(horizon), float(gamma), kind, float(eps),
                               float(ou_theta), float(ou_sigma),
                               float(ou_mu), int(seed))

    def rollout_run(self, episodes=1, reset=True, use_graph=True):
        """Run whole episodes on-device (one hipGraph per episode when
        reset=True); returns (env_steps, transitions_emitted)."""
        self.ext.rollout_run(self.h, int(episodes), bool(reset),
                             bool(use_graph))
        info = self.ext.rollout_info(self.h)
        return (episodes * info["env_steps_per_episode"],
                episodes * info["emits_per_episode"])

    def rollout_set_state(self, th, thdot):
        self.ext.rollout_set_state(self.h, torch.as_tensor(th),
                                   torch.as_tensor(thdot))

    def rollout_info(self):
        return self.ext.rollout_info(self.h)

    def replay_rows(self):
        """Host copy of the occupied replay rows (s, a, r, s2, d) — used
        by GPU actor ranks to ship device-generated transitions to the
        learner, and by tests."""
        st = self.ext.replay_state(self.h)
        return (st["s"].numpy(), st["a"].numpy(),
                st["r"].numpy().ravel(), st["s2"].numpy(),
                st["d"].numpy().ravel())

    def info(self):
        return self.ext.info(self.h)


class FusedDDPGBridge:
    """Glue between the Python DDPG object and the device engine for the
    'hip' backend: replaces the replay with a GPUReplayAdapter, runs train
    steps on-device, and lazily syncs parameters back to the torch modules
    for acting/eval/checkpointing."""

    def __init__(self, ddpg):
        self.ddpg = ddpg
        self.engine = FusedEngine(
            obs_dim=ddpg.obs_dim, act_dim=ddpg.act_dim,
            hidden=ddpg.actor.hidden, n_atoms=ddpg.n_atoms,
            batch=ddpg.batch_size, capacity=ddpg.memory_size,
            v_min=ddpg.v_min, v_max=ddpg.v_max, gamma_n=ddpg.n_step_gamma,
            tau=ddpg.tau,
            lr_actor=ddpg.optimizer_actor.param_groups[0]["lr"],
            lr_critic=ddpg.optimizer_critic.param_groups[0]["lr"],
            is_weighting=ddpg.is_weighting,
            seed=0 if ddpg.rng is None else int(ddpg.rng.integers(1 << 31)))
        self.engine.load_from_modules(ddpg.actor, ddpg.actor_target,
                                      ddpg.critic, ddpg.critic_target)
        # migrate any CPU replay contents, then swap in the GPU adapter
        adapter = GPUReplayAdapter(self.engine)
        old = getattr(ddpg, "replayBuffer", None)
        if old is not None and len(old) > 0:
            st = getattr(old, "_store", None) or getattr(old, "store", None)
            if st is not None and st.size > 0:
                n = st.size
                self.engine.ingest(
                    torch.from_numpy(st.states[:n]),
                    torch.from_numpy(st.actions[:n]),
                    torch.from_numpy(st.rewards[:n]),
                    torch.from_numpy(st.next_states[:n]),
                    torch.from_numpy(st.dones[:n]))
        ddpg.replayBuffer = adapter
        self._params_dirty = False

    def step(self, batch=None, n=1):
        self.ddpg.replayBuffer.flush()
        self.engine.train_steps(n)
        self._params_dirty = True
        self.ddpg.train_steps_done += n
        return float("nan"), float("nan")

    def sync_params_if_dirty(self):
        if self._params_dirty:
            self.engine.store_to_modules(
                self.ddpg.actor, self.ddpg.actor_target,
                self.ddpg.critic, self.ddpg.critic_target)
            self._params_dirty = False


def build_fused_engine(ddpg):
    if not torch.cuda.is_available():
        raise RuntimeError(
            "backend='hip' requires a GPU (use backend='eager' on CPU)")
    return FusedDDPGBridge(ddpg)
