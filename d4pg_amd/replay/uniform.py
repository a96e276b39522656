"""Uniform experience replay.

Capability parity with /root/reference/replay_memory.py:4-80 (ring buffer,
random minibatch, random-policy n-step prefill via ``initialize``), but a new
design: structure-of-arrays storage with preallocated float32 numpy arrays
and fully vectorized gather — the same SoA layout the on-HBM GPU replay
(the rs/ra/rr/rs2/rd slabs in ops/hip/engine.hip, fronted by
ops.GPUReplayAdapter) uses, so transitions move host<->device as flat
contiguous slabs.

API surface kept: ``Replay(max_size, env=None, n_steps=1, gamma=0.99)`` with
``add(state, action, reward, next_state, done)``, ``initialize(init_length)``
and ``sample(batch_size) -> (s, a, r, s2, done)`` stacked arrays of shape
[B, *] (reference returned float64; this build standardizes on float32, the
GPU compute dtype).
"""

from __future__ import annotations

import numpy as np


class SoAStore:
    """Preallocated SoA transition storage shared by uniform and PER buffers."""

    def __init__(self, capacity: int):
        self.capacity = int(capacity)
        self.size = 0
        self.pos = 0
        self._alloc_done = False

    def _alloc(self, state, action):
        s = np.asarray(state, dtype=np.float32).reshape(-1)
        a = np.asarray(action, dtype=np.float32).reshape(-1)
        c = self.capacity
        self.states = np.zeros((c, s.size), dtype=np.float32)
        self.actions = np.zeros((c, a.size), dtype=np.float32)
        self.rewards = np.zeros((c,), dtype=np.float32)
        self.next_states = np.zeros((c, s.size), dtype=np.float32)
        self.dones = np.zeros((c,), dtype=np.float32)
        self._alloc_done = True

    def add(self, state, action, reward, next_state, done) -> int:
        if not self._alloc_done:
            self._alloc(state, action)
        i = self.pos
        self.states[i] = np.asarray(state, dtype=np.float32).reshape(-1)
        self.actions[i] = np.asarray(action, dtype=np.float32).reshape(-1)
        self.rewards[i] = float(reward)
        self.next_states[i] = np.asarray(next_state, dtype=np.float32).reshape(-1)
        self.dones[i] = float(done)
        self.pos = (i + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)
        return i

    def gather(self, idx: np.ndarray):
        idx = np.asarray(idx)
        return (self.states[idx], self.actions[idx],
                self.rewards[idx].reshape(-1, 1), self.next_states[idx],
                self.dones[idx].reshape(-1, 1))

    def __len__(self) -> int:
        return self.size


class Replay:
    def __init__(self, max_size: int, env=None, n_steps: int = 1,
                 gamma: float = 0.99, rng: np.random.Generator | None = None):
        self.store = SoAStore(max_size)
        self.capacity = int(max_size)
        self.env = env
        self.n_steps = n_steps
        self.gamma = gamma
        self.rng = rng or np.random.default_rng()

    # reference-compat alias
    @property
    def buffer(self):
        return self.store

    def __len__(self) -> int:
        return len(self.store)

    def add(self, state, action, reward, next_state, done) -> None:
        self.store.add(state, action, reward, next_state, done)

    def initialize(self, init_length: int) -> None:
        """Prefill with random-policy n-step transitions
        (reference replay_memory.py:21-58 semantics): each stored tuple is
        (s_t, a_t, sum_{k<n} gamma^k r_{t+k}, s_{t+n}, done)."""
        from .nstep import NStepFolder
        env = self.env
        folder = NStepFolder(self.n_steps, self.gamma)
        state = env.reset()
        while len(self.store) < init_length:
            action = self.rng.uniform(-1.0, 1.0, size=env.action_space.shape)
            next_state, reward, done, _ = env.step(action)
            for tr in folder.push(state, action, reward, next_state, done):
                self.add(*tr)
            if done:
                state = env.reset()
                folder.reset()
            else:
                state = next_state

    def sample(self, batch_size: int):
        idx = self.rng.integers(0, len(self.store), size=batch_size)
        return self.store.gather(idx)
