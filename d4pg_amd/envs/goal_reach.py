"""Goal-conditioned env for HER (dict observations).

The reference trains HER on gym robotics-style envs with dict observations
``{observation, desired_goal, achieved_goal}``, ``info['is_success']`` and an
``env.compute_reward(achieved, desired)`` hook (/root/reference/main.py:73-79,
144-184).  None of those envs exist in this image, so this module provides a
native point-mass reaching task with exactly that interface: sparse reward
(0 on success, -1 otherwise), which is the regime HER was designed for.

Dynamics: a point in [-1, 1]^dim moves by a velocity action (scaled by
``speed``); success when within ``tol`` of the goal.
"""

from __future__ import annotations

import numpy as np

from .core import Box, Env


class GoalReachEnv(Env):
    _max_episode_steps = 50

    def __init__(self, dim: int = 2, speed: float = 0.1, tol: float = 0.05,
                 seed: int | None = None):
        super().__init__(seed)
        self.dim = dim
        self.speed = speed
        self.tol = tol
        self.action_space = Box(-1.0, 1.0, (dim,), rng=self.rng)
        self.observation_space = Box(-1.0, 1.0, (dim,), rng=self.rng)
        self.pos = np.zeros(dim)
        self.goal = np.zeros(dim)

    def compute_reward(self, achieved_goal, desired_goal, info=None):
        achieved_goal = np.asarray(achieved_goal, np.float64)
        desired_goal = np.asarray(desired_goal, np.float64)
        d = np.linalg.norm(achieved_goal - desired_goal, axis=-1)
        return np.where(d <= self.tol, 0.0, -1.0)

    def _dict_obs(self):
        return {
            "observation": self.pos.astype(np.float32).copy(),
            "achieved_goal": self.pos.astype(np.float32).copy(),
            "desired_goal": self.goal.astype(np.float32).copy(),
        }

    def _reset(self):
        self.pos = self.rng.uniform(-1.0, 1.0, self.dim)
        self.goal = self.rng.uniform(-1.0, 1.0, self.dim)
        return self._dict_obs()

    def _step(self, action):
        a = np.clip(action, -1.0, 1.0)
        self.pos = np.clip(self.pos + self.speed * a, -1.0, 1.0)
        r = float(self.compute_reward(self.pos, self.goal))
        success = r == 0.0
        return self._dict_obs(), r, False, {"is_success": success}
