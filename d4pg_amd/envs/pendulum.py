"""Native Pendulum-v1 (the reference's default env, main.py:40).

Implements the standard classic-control inverted-pendulum swing-up dynamics
(the public OpenAI-gym task definition) directly in numpy so the framework
trains without gym installed:

  state (theta, theta_dot); obs = [cos(theta), sin(theta), theta_dot]
  torque u clipped to [-2, 2]
  reward = -(wrap(theta)^2 + 0.1*theta_dot^2 + 0.001*u^2)
  theta_dot' = theta_dot + (3g/(2l) sin(theta) + 3/(m l^2) u) dt, clip [-8, 8]
  theta'     = theta + theta_dot' dt
  dt = 0.05, g = 10, m = 1, l = 1
  init: theta ~ U[-pi, pi], theta_dot ~ U[-1, 1]; horizon 200 steps

Return range per 200-step episode is about [-1700, 0], which is why the
reference overrides the C51 support to [v_min, v_max] = [-300, 0] for this
env (main.py:84-88) — with n-step backups most of the mass sits well above
the worst case.
"""

from __future__ import annotations

import numpy as np

from .core import Box, Env


def angle_normalize(x):
    return ((x + np.pi) % (2 * np.pi)) - np.pi


class PendulumEnv(Env):
    _max_episode_steps = 200

    max_speed = 8.0
    max_torque = 2.0
    dt = 0.05
    g = 10.0
    m = 1.0
    l = 1.0

    def __init__(self, seed: int | None = None):
        super().__init__(seed)
        self.action_space = Box(-self.max_torque, self.max_torque, (1,),
                                rng=self.rng)
        high = np.array([1.0, 1.0, self.max_speed], dtype=np.float32)
        self.observation_space = Box(-high, high, rng=self.rng)
        self.th = 0.0
        self.thdot = 0.0

    def _obs(self):
        return np.array([np.cos(self.th), np.sin(self.th), self.thdot],
                        dtype=np.float32)

    def _reset(self):
        self.th = self.rng.uniform(-np.pi, np.pi)
        self.thdot = self.rng.uniform(-1.0, 1.0)
        return self._obs()

    def _step(self, action):
        u = float(np.clip(action, -self.max_torque, self.max_torque)[0])
        th, thdot = self.th, self.thdot
        cost = angle_normalize(th) ** 2 + 0.1 * thdot ** 2 + 0.001 * u ** 2
        newthdot = thdot + (3.0 * self.g / (2.0 * self.l) * np.sin(th)
                            + 3.0 / (self.m * self.l ** 2) * u) * self.dt
        newthdot = float(np.clip(newthdot, -self.max_speed, self.max_speed))
        self.th = th + newthdot * self.dt
        self.thdot = newthdot
        return self._obs(), -cost, False, {}
