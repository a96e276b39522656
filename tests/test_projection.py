"""C51 projection vs a per-element numpy oracle reimplementing the
reference's reproject2 semantics (ddpg.py:142-185) with the gamma**n
discount (SURVEY.md §4 unit-test spec)."""

import numpy as np
import pytest
import torch

from d4pg_amd.algo.projection import categorical_projection


def oracle_project(next_dist, rewards, dones, v_min, v_max, gamma_n):
    """Straightforward per-atom loop oracle (reproject2 semantics with
    gamma_n; terminal rows = delta at clip(r))."""
    B, K = next_dist.shape
    delta = (v_max - v_min) / (K - 1)
    m = np.zeros((B, K), dtype=np.float64)
    for i in range(B):
        if dones[i] > 0.5:
            tz = min(v_max, max(v_min, rewards[i]))
            b = (tz - v_min) / delta
            l, u = int(np.floor(b)), int(np.ceil(b))
            if l == u:
                m[i, l] = 1.0
            else:
                m[i, l] = u - b
                m[i, u] = b - l
            continue
        for j in range(K):
            z = v_min + j * delta
            tz = min(v_max, max(v_min, rewards[i] + gamma_n * z))
            b = (tz - v_min) / delta
            l, u = int(np.floor(b)), int(np.ceil(b))
            if l == u:
                m[i, l] += next_dist[i, j]
            else:
                m[i, l] += next_dist[i, j] * (u - b)
                m[i, u] += next_dist[i, j] * (b - l)
    return m


def _rand_dist(B, K, rng):
    p = rng.random((B, K))
    return p / p.sum(axis=1, keepdims=True)


@pytest.mark.parametrize("seed", [0, 1, 2])
@pytest.mark.parametrize("gamma_n", [0.99, 0.99 ** 5])
def test_projection_matches_oracle(seed, gamma_n):
    rng = np.random.default_rng(seed)
    B, K = 64, 51
    v_min, v_max = -300.0, 0.0
    p = _rand_dist(B, K, rng)
    r = rng.uniform(-350, 20, B)       # includes out-of-support rewards
    d = (rng.random(B) < 0.3).astype(np.float64)
    m = categorical_projection(
        torch.tensor(p, dtype=torch.float64), torch.tensor(r),
        torch.tensor(d), v_min, v_max, gamma_n).numpy()
    expect = oracle_project(p, r, d, v_min, v_max, gamma_n)
    np.testing.assert_allclose(m, expect, atol=1e-10)


def test_projection_mass_conserved():
    rng = np.random.default_rng(3)
    B, K = 128, 51
    p = _rand_dist(B, K, rng)
    m = categorical_projection(
        torch.tensor(p, dtype=torch.float64),
        torch.tensor(rng.uniform(-500, 100, B)),
        torch.tensor((rng.random(B) < 0.5).astype(np.float64)),
        -300.0, 0.0, 0.95).numpy()
    np.testing.assert_allclose(m.sum(axis=1), np.ones(B), atol=1e-9)
    assert np.all(m >= 0)


def test_projection_exact_atom_hit():
    """Tz landing exactly on an atom: full mass to that atom (the l==u
    adjustment path, ddpg.py:133-134 trick)."""
    K = 51
    p = np.zeros((1, K))
    p[0, 10] = 1.0
    # gamma=1, r=0: Tz = z exactly
    m = categorical_projection(torch.tensor(p), torch.zeros(1),
                               torch.zeros(1), -300.0, 0.0, 1.0).numpy()
    np.testing.assert_allclose(m, p, atol=1e-12)


def test_projection_edge_bins():
    K = 51
    p = np.full((2, K), 1.0 / K)
    r = np.array([-1e9, 1e9])      # clamps to v_min / v_max
    d = np.array([1.0, 1.0])
    m = categorical_projection(torch.tensor(p), torch.tensor(r),
                               torch.tensor(d), -300.0, 0.0, 0.99).numpy()
    assert m[0, 0] == pytest.approx(1.0)
    assert m[1, -1] == pytest.approx(1.0)


def test_terminal_row_is_delta():
    K = 51
    rng = np.random.default_rng(4)
    p = _rand_dist(1, K, rng)
    m = categorical_projection(torch.tensor(p), torch.tensor([-150.0]),
                               torch.tensor([1.0]), -300.0, 0.0, 0.99).numpy()
    # -150 sits exactly on atom 25 (delta=6)
    assert m[0, 25] == pytest.approx(1.0)
    assert m.sum() == pytest.approx(1.0)
