"""Uniform replay + n-step folding tests."""

import numpy as np
import pytest

from d4pg_amd.envs import make
from d4pg_amd.replay.nstep import NStepFolder
from d4pg_amd.replay.uniform import Replay


def test_ring_buffer_overwrite():
    r = Replay(5)
    for i in range(12):
        r.add([i], [0.0], i, [i + 1], False)
    assert len(r) == 5
    s, a, rew, s2, d = r.sample(5)
    assert s.shape == (5, 1) and rew.shape == (5, 1)
    assert set(s.astype(int).ravel().tolist()) <= set(range(7, 12))


def test_sample_shapes_float32():
    r = Replay(100)
    for i in range(50):
        r.add(np.random.randn(3), np.random.randn(2), 0.5,
              np.random.randn(3), i % 10 == 0)
    s, a, rew, s2, d = r.sample(16)
    assert s.shape == (16, 3) and a.shape == (16, 2)
    assert s.dtype == np.float32


def test_nstep_folder_semantics():
    """Window of n=3 with gamma=0.5: emitted tuple must be
    (s_t, a_t, r_t + 0.5 r_{t+1} + 0.25 r_{t+2}, s_{t+3}, done)."""
    f = NStepFolder(3, 0.5)
    out = []
    rewards = [1.0, 2.0, 4.0, 8.0]
    for t in range(4):
        out += f.push(f"s{t}", f"a{t}", rewards[t], f"s{t+1}", t == 3)
    assert len(out) == 2
    s, a, r, s2, d = out[0]
    assert (s, a, s2, d) == ("s0", "a0", "s3", False)
    assert r == pytest.approx(1 + 0.5 * 2 + 0.25 * 4)
    s, a, r, s2, d = out[1]
    assert (s, a, s2, d) == ("s1", "a1", "s4", True)
    assert r == pytest.approx(2 + 0.5 * 4 + 0.25 * 8)


def test_nstep_incremental_matches_resum():
    rng = np.random.default_rng(0)
    f = NStepFolder(5, 0.99, resync=10 ** 9)   # never resync: pure increments
    g = NStepFolder(5, 0.99, resync=1)          # resum every step
    for t in range(500):
        r = float(rng.standard_normal())
        o1 = f.push(t, t, r, t + 1, False)
        o2 = g.push(t, t, r, t + 1, False)
        if o1:
            assert o1[0][2] == pytest.approx(o2[0][2], abs=1e-8)


def test_nstep_reset_no_partial_flush():
    f = NStepFolder(3, 0.9)
    out = f.push("s0", "a0", 1.0, "s1", False)
    assert out == []
    f.reset()
    out = f.push("s0", "a0", 1.0, "s1", False)
    assert out == []


def test_initialize_prefill():
    env = make("Pendulum-v1", seed=0)
    r = Replay(1000, env=env, n_steps=3, gamma=0.99,
               rng=np.random.default_rng(0))
    r.initialize(200)
    assert len(r) >= 200
    s, a, rew, s2, d = r.sample(32)
    assert s.shape == (32, 3)
    assert np.isfinite(rew).all()
