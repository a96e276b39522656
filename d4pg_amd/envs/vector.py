"""Vectorized environments + batched n-step folding.

The reference steps one env per worker process with a B=1 policy forward
per step (main.py:142-152) — ~25-30k env-steps/s per actor on this host.
The MI355X-native actor instead simulates M envs per process in numpy
batch form and runs ONE [M, obs] policy forward per tick, so each actor
rank feeds the learner 10-30x more transitions for the same process count.

VecNStep reproduces NStepFolder's semantics (replay/nstep.py — emit
(s_{t-n+1}, a_{t-n+1}, sum gamma^k r, s_{t+1}, done) once the window is
full; reset on episode end without flushing) in array form.  All M
pendulums share the fixed 200-step horizon, so episodes stay synchronized
and the window reset is a single buffer clear.
"""

from __future__ import annotations

import numpy as np

from .pendulum import angle_normalize


class VectorPendulum:
    """M independent Pendulum-v1 instances in numpy arrays.  Actions are
    NORMALIZED to (-1, 1) (the NormalizeAction convention); the affine map
    to [-2, 2] torque happens inside."""

    obs_dim = 3
    act_dim = 1
    horizon = 200
    max_speed = 8.0
    max_torque = 2.0
    dt = 0.05
    g = 10.0
    m = 1.0
    l = 1.0

    def __init__(self, n_envs: int, seed: int | None = None,
                 horizon: int | None = None):
        self.n = int(n_envs)
        self.rng = np.random.default_rng(seed)
        if horizon is not None:
            self.horizon = int(horizon)
        self.th = np.zeros(self.n)
        self.thdot = np.zeros(self.n)
        self.t = 0

    def reset(self) -> np.ndarray:
        self.th = self.rng.uniform(-np.pi, np.pi, self.n)
        self.thdot = self.rng.uniform(-1.0, 1.0, self.n)
        self.t = 0
        return self._obs()

    def _obs(self) -> np.ndarray:
        return np.stack([np.cos(self.th), np.sin(self.th), self.thdot],
                        axis=1).astype(np.float32)

    def step(self, actions: np.ndarray):
        """actions [M, 1] in (-1, 1).  Returns (obs, rewards, done) where
        done is a scalar — the synchronized horizon flag.  Auto-resets
        AFTER the caller has read the terminal obs via the return value."""
        u = np.clip(actions[:, 0], -1.0, 1.0) * self.max_torque
        cost = (angle_normalize(self.th) ** 2 + 0.1 * self.thdot ** 2
                + 0.001 * u ** 2)
        newthdot = self.thdot + (3.0 * self.g / (2.0 * self.l)
                                 * np.sin(self.th)
                                 + 3.0 / (self.m * self.l ** 2) * u) * self.dt
        self.thdot = np.clip(newthdot, -self.max_speed, self.max_speed)
        self.th = self.th + self.thdot * self.dt
        self.t += 1
        return self._obs(), (-cost).astype(np.float32), self.t >= self.horizon


class VecNStep:
    """Array-form n-step folder for M synchronized envs."""

    def __init__(self, n_envs: int, obs_dim: int, act_dim: int,
                 n_steps: int, gamma: float):
        self.m = n_envs
        self.n = max(1, int(n_steps))
        self.gamma = float(gamma)
        self.obs_dim, self.act_dim = obs_dim, act_dim
        self.reset()

    def reset(self):
        self._s = np.zeros((self.n, self.m, self.obs_dim), np.float32)
        self._a = np.zeros((self.n, self.m, self.act_dim), np.float32)
        self._r = np.zeros((self.n, self.m), np.float32)
        self._len = 0
        self._head = 0

    def push(self, s, a, r, s2, done_flag: bool):
        """Feed one synchronized step for all M envs; returns a matured
        (S, A, R, S2, D) tuple of [M, ...] arrays or None."""
        i = (self._head + self._len) % self.n      # tail slot
        self._s[i] = s
        self._a[i] = a
        self._r[i] = r
        if self._len < self.n:
            self._len += 1
        else:
            self._head = (self._head + 1) % self.n
        if self._len < self.n:
            return None
        start = self._head                          # oldest entry
        R = np.zeros(self.m, np.float32)
        for k in range(self.n):
            R += (self.gamma ** k) * self._r[(start + k) % self.n]
        out = (self._s[start].copy(), self._a[start].copy(), R,
               np.asarray(s2, np.float32).copy(),
               np.full(self.m, float(done_flag), np.float32))
        if done_flag:
            self.reset()
        return out
