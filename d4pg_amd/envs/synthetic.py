"""Spec-matched synthetic environments.

mujoco/pybullet are not importable here, so the BASELINE.json configs that
name HalfCheetah-v4 (obs 17, act 6) and Humanoid-v4 (obs 376, act 17) run
against synthetic envs with the same observation/action spaces and episode
horizons: a smooth random recurrent dynamics (fixed random orthogonal-ish
mixing matrix + tanh squash + action injection) with a dense bounded reward.
Deterministic given a seed, so the distributed plumbing and throughput
benches exercise real data paths with reproducible streams.

``SyntheticEnv`` is also the fake-env used by the multi-process tests
("distributed without a cluster" requires a deterministic
synthetic transition generator).
"""

from __future__ import annotations

import numpy as np

from .core import Box, Env


class SyntheticEnv(Env):
    def __init__(self, obs_dim: int, act_dim: int, horizon: int = 1000,
                 act_high: float = 1.0, seed: int | None = None):
        super().__init__(seed)
        self._max_episode_steps = horizon
        self.obs_dim = obs_dim
        self.act_dim = act_dim
        self.action_space = Box(-act_high, act_high, (act_dim,), rng=self.rng)
        self.observation_space = Box(-np.inf, np.inf, (obs_dim,),
                                     rng=self.rng)
        mix_rng = np.random.default_rng(0 if seed is None else seed)
        a = mix_rng.standard_normal((obs_dim, obs_dim)) / np.sqrt(obs_dim)
        self.W = 0.9 * a
        self.U = mix_rng.standard_normal((act_dim, obs_dim)) / np.sqrt(act_dim)
        self.state = np.zeros(obs_dim)

    def _reset(self):
        self.state = self.rng.standard_normal(self.obs_dim) * 0.1
        return self.state.astype(np.float32).copy()

    def _step(self, action):
        a = np.clip(action, self.action_space.low, self.action_space.high)
        self.state = np.tanh(self.state @ self.W + a @ self.U
                             + 0.01 * self.rng.standard_normal(self.obs_dim))
        reward = float(-np.mean(self.state ** 2) + 0.1 * np.mean(a ** 2))
        return self.state.astype(np.float32).copy(), reward, False, {}


# (obs_dim, act_dim, act_high, horizon) for the mujoco env ids BASELINE.json
# names; used when the real simulator is unavailable.
MUJOCO_SPECS = {
    "HalfCheetah-v4": (17, 6, 1.0, 1000),
    "Walker2d-v4": (17, 6, 1.0, 1000),
    "Ant-v4": (27, 8, 1.0, 1000),
    "Hopper-v4": (11, 3, 1.0, 1000),
    "Humanoid-v4": (376, 17, 0.4, 1000),
}
