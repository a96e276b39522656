"""Env registry: ``make(env_id)`` mirrors the reference's
``NormalizeAction(gym.make(args.env).env)`` construction site
(/root/reference/main.py:68) with native fallbacks when gym is absent."""

from __future__ import annotations

import numpy as np

from .core import Box, Env, NormalizeAction, GymAdapter  # noqa: F401
from .pendulum import PendulumEnv
from .goal_reach import GoalReachEnv
from .synthetic import SyntheticEnv, MUJOCO_SPECS

_NATIVE = {
    "Pendulum-v0": lambda seed: PendulumEnv(seed=seed),
    "Pendulum-v1": lambda seed: PendulumEnv(seed=seed),
    "GoalReach-v0": lambda seed: GoalReachEnv(seed=seed),
}


def make(env_id: str, seed: int | None = None, normalize: bool = True,
         prefer_native: bool = True):
    """Build an env by id.  Resolution order: native implementations,
    real gym (if importable), synthetic spec-matched fallback."""
    env = None
    if prefer_native and env_id in _NATIVE:
        env = _NATIVE[env_id](seed)
    if env is None:
        try:
            import gymnasium as gym
            env = GymAdapter(gym.make(env_id))
        except Exception:
            try:
                import gym
                env = GymAdapter(gym.make(env_id).env)
            except Exception:
                env = None
    if env is None:
        if env_id.startswith("Synthetic"):
            # "Synthetic-<obs>x<act>" ad-hoc spec
            try:
                dims = env_id.split("-", 1)[1]
                o, a = (int(x) for x in dims.split("x"))
                env = SyntheticEnv(o, a, seed=seed)
            except Exception:
                env = SyntheticEnv(16, 4, seed=seed)
        elif env_id in MUJOCO_SPECS:
            o, a, hi, horizon = MUJOCO_SPECS[env_id]
            env = SyntheticEnv(o, a, horizon=horizon, act_high=hi, seed=seed)
        else:
            raise ValueError(
                f"unknown env id {env_id!r}: no native implementation, gym "
                f"is not importable, and no synthetic spec is registered")
    return NormalizeAction(env) if normalize else env


def obs_act_dims(env, her: bool = False):
    """Observation/action dims the way the reference derives them
    (main.py:70-80): flat obs dim (observation+goal concat under HER), and
    discrete-vs-continuous action detection (``Discrete.n`` vs
    ``Box.shape``)."""
    if her:
        o = env.reset()
        obs_dim = int(np.asarray(o["observation"]).size
                      + np.asarray(o["desired_goal"]).size)
    else:
        obs_dim = int(np.prod(env.observation_space.shape))
    space = env.action_space
    if hasattr(space, "n") and not getattr(space, "shape", None):
        act_dim = int(space.n)          # gym.spaces.Discrete
    else:
        act_dim = int(np.prod(space.shape))
    return obs_dim, act_dim
