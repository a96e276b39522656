"""Env resolution fallbacks: synthetic spec-matched envs for the BASELINE
mujoco configs (no gym/mujoco in this image), and the Synthetic-<o>x<a>
ad-hoc ids."""

import numpy as np

from d4pg_amd.envs import make, obs_act_dims


def test_mujoco_spec_fallback_dims():
    for env_id, (o, a) in {"HalfCheetah-v4": (17, 6),
                           "Humanoid-v4": (376, 17),
                           "Hopper-v4": (11, 3)}.items():
        env = make(env_id, seed=0)
        od, ad = obs_act_dims(env)
        assert (od, ad) == (o, a), env_id
        obs = env.reset()
        assert obs.shape == (o,)
        obs2, r, done, info = env.step(np.zeros(ad, np.float32))
        assert obs2.shape == (o,) and np.isfinite(r)


def test_adhoc_synthetic_id():
    env = make("Synthetic-5x2", seed=1)
    od, ad = obs_act_dims(env)
    assert (od, ad) == (5, 2)


def test_synthetic_env_is_deterministic_per_seed():
    e1 = make("HalfCheetah-v4", seed=7)
    e2 = make("HalfCheetah-v4", seed=7)
    o1, o2 = e1.reset(), e2.reset()
    np.testing.assert_allclose(o1, o2)
    a = np.full(6, 0.3, np.float32)
    s1 = e1.step(a)[0]
    s2 = e2.step(a)[0]
    np.testing.assert_allclose(s1, s2)
