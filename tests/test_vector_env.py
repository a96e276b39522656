"""Vectorized env + batched n-step folding vs the scalar references."""

import numpy as np

from d4pg_amd.envs.pendulum import PendulumEnv
from d4pg_amd.envs.vector import VecNStep, VectorPendulum
from d4pg_amd.replay.nstep import NStepFolder


def test_vector_pendulum_matches_scalar_dynamics():
    sc = PendulumEnv(seed=0)
    sc.reset()
    vec = VectorPendulum(4, seed=1)
    vec.reset()
    # force env 2 of the vector into the scalar env's state
    vec.th[2] = sc.th
    vec.thdot[2] = sc.thdot
    rng = np.random.default_rng(3)
    for _ in range(50):
        a = rng.uniform(-1, 1, (4, 1))
        # scalar env takes raw torque in [-2, 2] (NormalizeAction maps);
        # the vector env takes normalized (-1,1) and scales internally
        o_s, r_s, _, _ = sc.step(np.array([a[2, 0] * 2.0]))
        o_v, r_v, _ = vec.step(a)
        np.testing.assert_allclose(o_v[2], o_s, rtol=1e-5, atol=1e-6)
        np.testing.assert_allclose(r_v[2], r_s, rtol=1e-5, atol=1e-6)


def test_vecnstep_matches_scalar_folder():
    m, n_steps, gamma, T = 3, 5, 0.97, 23
    rng = np.random.default_rng(0)
    S = rng.standard_normal((T + 1, m, 2)).astype(np.float32)
    A = rng.standard_normal((T, m, 1)).astype(np.float32)
    R = rng.standard_normal((T, m)).astype(np.float32)

    vec = VecNStep(m, 2, 1, n_steps, gamma)
    got = [[] for _ in range(m)]
    for t in range(T):
        out = vec.push(S[t], A[t], R[t], S[t + 1], t == T - 1)
        if out is not None:
            s, a, r, s2, d = out
            for e in range(m):
                got[e].append((s[e], a[e], r[e], s2[e], d[e]))

    for e in range(m):
        ref = NStepFolder(n_steps, gamma)
        want = []
        for t in range(T):
            for tr in ref.push(S[t, e], A[t, e], float(R[t, e]),
                               S[t + 1, e], t == T - 1):
                want.append(tr)
        assert len(got[e]) == len(want)
        for (gs, ga, gr, gs2, gd), (ws, wa, wr, ws2, wd) in zip(got[e], want):
            np.testing.assert_allclose(gs, ws, atol=1e-6)
            np.testing.assert_allclose(ga, wa, atol=1e-6)
            np.testing.assert_allclose(gr, wr, rtol=1e-4)
            np.testing.assert_allclose(gs2, ws2, atol=1e-6)
            assert bool(gd) == bool(wd)


def test_vector_pendulum_episode_shapes():
    vec = VectorPendulum(8, seed=0, horizon=10)
    obs = vec.reset()
    assert obs.shape == (8, 3)
    for t in range(10):
        obs, r, done = vec.step(np.zeros((8, 1)))
        assert obs.shape == (8, 3) and r.shape == (8,)
        assert done == (t == 9)
