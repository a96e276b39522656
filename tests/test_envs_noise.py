"""Env layer + noise process tests."""

import numpy as np
import pytest

from d4pg_amd.envs import make, obs_act_dims
from d4pg_amd.envs.core import Box, NormalizeAction
from d4pg_amd.envs.pendulum import PendulumEnv, angle_normalize
from d4pg_amd.envs.synthetic import SyntheticEnv
from d4pg_amd.noise import GaussianNoise, OrnsteinUhlenbeckProcess


def test_pendulum_contract():
    env = PendulumEnv(seed=0)
    obs = env.reset()
    assert obs.shape == (3,)
    assert abs(obs[0] ** 2 + obs[1] ** 2 - 1.0) < 1e-5
    total = 0.0
    for _ in range(200):
        obs, r, done, info = env.step(np.array([0.0]))
        assert r <= 0.0
        total += r
        if done:
            break
    assert done                              # 200-step horizon
    assert -2000 < total < 0


def test_pendulum_physics_step():
    env = PendulumEnv(seed=0)
    env.reset()
    env.th, env.thdot = 0.1, 0.0
    obs, r, done, _ = env.step(np.array([0.0]))
    # hand-computed: thdot' = 0 + (15*sin(0.1))*0.05, th' = 0.1 + thdot'*0.05
    thdot = 15.0 * np.sin(0.1) * 0.05
    th = 0.1 + thdot * 0.05
    assert obs[2] == pytest.approx(thdot, abs=1e-6)
    assert obs[1] == pytest.approx(np.sin(th), abs=1e-6)
    assert r == pytest.approx(-(0.1 ** 2), abs=1e-6)


def test_angle_normalize():
    assert angle_normalize(np.pi + 0.1) == pytest.approx(-np.pi + 0.1)
    assert angle_normalize(-np.pi - 0.1) == pytest.approx(np.pi - 0.1)
    assert angle_normalize(0.3) == pytest.approx(0.3)


def test_normalize_action_affine():
    env = PendulumEnv(seed=0)            # action space [-2, 2]
    w = NormalizeAction(env)
    assert np.allclose(w._action(np.array([1.0])), [2.0])
    assert np.allclose(w._action(np.array([-1.0])), [-2.0])
    assert np.allclose(w._action(np.array([0.0])), [0.0])
    assert np.allclose(w._reverse_action(np.array([2.0])), [1.0])
    w._max_episode_steps = 50
    assert env._max_episode_steps == 50


def test_make_registry_and_dims():
    env = make("Pendulum-v1", seed=0)
    assert obs_act_dims(env) == (3, 1)
    env = make("HalfCheetah-v4", seed=0)     # synthetic fallback spec
    assert obs_act_dims(env) == (17, 6)
    env = make("Humanoid-v4", seed=0)
    assert obs_act_dims(env) == (376, 17)
    with pytest.raises(ValueError):
        make("NoSuchEnv-v99")


def test_her_env_dims():
    env = make("GoalReach-v0", seed=0)
    obs_dim, act_dim = obs_act_dims(env, her=True)
    assert (obs_dim, act_dim) == (4, 2)      # 2 obs + 2 goal
    o = env.reset()
    assert set(o.keys()) == {"observation", "achieved_goal", "desired_goal"}
    o2, r, done, info = env.step(np.zeros(2))
    assert r in (-1.0, 0.0)
    assert "is_success" in info


def test_goal_env_compute_reward_batch():
    env = make("GoalReach-v0", seed=0)
    a = np.zeros((5, 2))
    b = np.zeros((5, 2))
    b[2] = 1.0
    r = env.compute_reward(a, b)
    assert r.shape == (5,)
    assert r[0] == 0.0 and r[2] == -1.0


def test_synthetic_env_deterministic():
    e1 = SyntheticEnv(8, 2, seed=7)
    e2 = SyntheticEnv(8, 2, seed=7)
    o1, o2 = e1.reset(), e2.reset()
    np.testing.assert_allclose(o1, o2)
    a = np.ones(2) * 0.3
    np.testing.assert_allclose(e1.step(a)[0], e2.step(a)[0])


def test_box_sample_bounds():
    b = Box(-2.0, 3.0, (4,))
    for _ in range(10):
        x = b.sample()
        assert b.contains(x)


def test_gaussian_noise_moments():
    g = GaussianNoise(1000, eps=0.3, sigma=1.0,
                      rng=np.random.default_rng(0))
    x = g.sample()
    assert x.shape == (1000,)
    assert abs(x.std() - 0.3) < 0.03
    g.reset()                                 # decay off by default
    assert g.eps == 0.3


def test_ou_noise_mean_reversion():
    ou = OrnsteinUhlenbeckProcess(1, mu=0.0, theta=0.5, sigma=0.0, dt=0.1,
                                  rng=np.random.default_rng(0))
    ou.x = np.array([10.0])
    v = [ou.sample()[0] for _ in range(50)]
    assert v[-1] < v[0]                       # decays toward mu with sigma=0
    ou.reset()
    assert np.allclose(ou.x, 0.0)
    assert ou.eps < 1.0                       # reset-decay active (parity)
