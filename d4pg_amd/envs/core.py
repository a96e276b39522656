"""Environment layer.

gym is not importable in this image, so the framework ships its own minimal
gym-compatible env API (old-gym conventions matching the reference's usage:
``reset() -> obs``, ``step(a) -> (obs, reward, done, info)``, 4-tuple, and a
writable ``_max_episode_steps`` — /root/reference/main.py:68-69) plus an
adapter so real gym/gymnasium envs plug in when present.

``NormalizeAction`` reproduces /root/reference/normalize_env.py:3-14: the
agent acts in tanh-range (-1, 1); the wrapper affine-maps to
[space.low, space.high] on the way in and back on the way out.
"""

from __future__ import annotations

import numpy as np


class Box:
    """Minimal continuous space (gym.spaces.Box-compatible surface)."""

    def __init__(self, low, high, shape=None, dtype=np.float32,
                 rng: np.random.Generator | None = None):
        if shape is None:
            shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        self.shape = tuple(shape)
        self.low = np.broadcast_to(np.asarray(low, dtype), self.shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype), self.shape).copy()
        self.dtype = dtype
        self._rng = rng or np.random.default_rng()

    def sample(self) -> np.ndarray:
        return self._rng.uniform(self.low, self.high).astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and np.all(x >= self.low - 1e-6) \
            and np.all(x <= self.high + 1e-6)

    def __repr__(self):
        return f"Box{self.shape}"


class Env:
    """Base env: subclasses set observation_space/action_space and implement
    _reset/_step; episode-length truncation is handled here via
    ``_max_episode_steps`` (writable, reference main.py:69 contract)."""

    _max_episode_steps = 1000

    def __init__(self, seed: int | None = None):
        self.rng = np.random.default_rng(seed)
        self._elapsed = 0

    def seed(self, seed: int | None = None):
        self.rng = np.random.default_rng(seed)
        return [seed]

    def reset(self):
        self._elapsed = 0
        return self._reset()

    def step(self, action):
        obs, reward, done, info = self._step(np.asarray(action, np.float64))
        self._elapsed += 1
        if self._elapsed >= self._max_episode_steps:
            done = True
        return obs, reward, done, info

    # old-gym `.env` unwrap attribute used by the reference (main.py:68)
    @property
    def env(self):
        return self

    def compute_reward(self, achieved_goal, desired_goal, info=None):
        raise NotImplementedError


class NormalizeAction:
    """Affine action rescale wrapper (reference normalize_env.py:3-14).

    ``_action``:  agent's (-1, 1) -> [low, high]  (applied in step)
    ``_reverse_action``: [low, high] -> (-1, 1)
    """

    def __init__(self, env):
        self.wrapped = env

    def __getattr__(self, name):
        if name == "wrapped":      # guard: unpickling calls __getattr__
            raise AttributeError(name)   # before __dict__ is populated
        return getattr(self.wrapped, name)

    def _action(self, action):
        sp = self.wrapped.action_space
        act_k = (sp.high - sp.low) / 2.0
        act_b = (sp.high + sp.low) / 2.0
        return act_k * np.asarray(action, np.float64) + act_b

    def _reverse_action(self, action):
        sp = self.wrapped.action_space
        act_k_inv = 2.0 / (sp.high - sp.low)
        act_b = (sp.high + sp.low) / 2.0
        return act_k_inv * (np.asarray(action, np.float64) - act_b)

    def step(self, action):
        return self.wrapped.step(self._action(action))

    def reset(self):
        return self.wrapped.reset()

    @property
    def _max_episode_steps(self):
        return self.wrapped._max_episode_steps

    @_max_episode_steps.setter
    def _max_episode_steps(self, v):
        self.wrapped._max_episode_steps = v


class GymAdapter(Env):
    """Wraps a real gym/gymnasium env (when the library exists) into this
    module's old-gym surface; handles the 5-tuple step and tuple reset of
    new-style gym APIs."""

    def __init__(self, gym_env):
        super().__init__()
        self.gym_env = gym_env
        self.observation_space = gym_env.observation_space
        self.action_space = gym_env.action_space
        self._max_episode_steps = getattr(gym_env, "_max_episode_steps", 1000)

    def reset(self):
        out = self.gym_env.reset()
        return out[0] if isinstance(out, tuple) else out

    def step(self, action):
        out = self.gym_env.step(action)
        if len(out) == 5:
            obs, reward, terminated, truncated, info = out
            return obs, reward, terminated or truncated, info
        return out

    def compute_reward(self, achieved_goal, desired_goal, info=None):
        return self.gym_env.compute_reward(achieved_goal, desired_goal, info)
