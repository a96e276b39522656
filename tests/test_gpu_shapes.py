"""Engine parity at awkward shapes: dims that don't divide the tile sizes
(64-col chunks, 4-row tiles, wave-wide softmax) and batches that select
each execution path — persistent (B<=256), row-block (256<B<=512), and
MFMA (B>=512) — all against the eager fp32 oracle."""

import copy

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

GAMMA_N = 0.99 ** 3


def run_one_step(O, A, H, K, B, cap=4096, seed=3):
    from d4pg_amd.algo.projection import categorical_projection
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    dist = {"type": "categorical", "v_min": -100.0, "v_max": 50.0,
            "n_atoms": K}
    eng = FusedEngine(obs_dim=O, act_dim=A, hidden=H, n_atoms=K, batch=B,
                      capacity=cap, v_min=-100.0, v_max=50.0,
                      gamma_n=GAMMA_N, tau=0.01, lr_actor=1e-4,
                      lr_critic=1e-4, seed=seed)
    torch.manual_seed(seed)
    a = actor(O, A, hidden=H)
    c = critic(O, A, dist, hidden=H)
    at = copy.deepcopy(a)
    ct = copy.deepcopy(c)
    eng.load_from_modules(a, at, c, ct)
    rng = np.random.default_rng(seed)
    n = max(2 * B, 512)
    eng.ingest(torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy(rng.uniform(-1, 1, (n, A)).astype("f")),
               torch.from_numpy(rng.uniform(-20, 5, n).astype("f")),
               torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy((rng.random(n) < 0.1).astype("f")))
    eng.step(1)

    s = torch.from_numpy(eng.read("bs").numpy())
    act = torch.from_numpy(eng.read("ba").numpy())
    r = torch.from_numpy(eng.read("br").numpy())
    s2 = torch.from_numpy(eng.read("bs2").numpy())
    d = torch.from_numpy(eng.read("bd").numpy())
    with torch.no_grad():
        a2 = at(s2)
        p_t = ct(s2, a2)
        m = categorical_projection(p_t, r, d, -100.0, 50.0, GAMMA_N)
        q = c(s, act)
    np.testing.assert_allclose(eng.read("a2").numpy(), a2.numpy(), atol=5e-5)
    np.testing.assert_allclose(eng.read("p_t").numpy(), p_t.numpy(),
                               atol=5e-5)
    np.testing.assert_allclose(eng.read("m_proj").numpy(), m.numpy(),
                               atol=1e-4)
    np.testing.assert_allclose(eng.read("q").numpy(), q.numpy(), atol=5e-5)
    cnt = eng.counters()
    assert cnt["adam_t_actor"] == 1
    assert np.isfinite(cnt["loss_critic"]) and np.isfinite(cnt["loss_actor"])


@pytest.mark.parametrize("O,A,H,K,B", [
    (5, 2, 192, 21, 48),       # persistent path, nothing divides nicely
    (7, 3, 128, 51, 33),       # persistent, odd batch
    (11, 1, 320, 64, 256),     # persistent upper edge, K == wave
    (5, 2, 192, 31, 300),      # row-block path (256 < B <= 512)
    (9, 4, 192, 41, 640),      # MFMA path, dims off the 64/128 tiles
    (9, 4, 256, 41, 2048),     # big-GEMM path (128x256 tiles + split-K)
])
def test_engine_shape_parity(O, A, H, K, B):
    run_one_step(O, A, H, K, B)


def test_large_capacity_engine():
    """Regression guard for the on-HBM replay sizing claim: a 2^24-slot
    engine allocates, fills, and trains (the 1e8 version is exercised in
    profiles/ evidence runs)."""
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    cap = 1 << 24
    eng = FusedEngine(obs_dim=3, act_dim=1, hidden=256, n_atoms=51,
                      batch=64, capacity=cap, v_min=-300.0, v_max=0.0,
                      gamma_n=0.99 ** 5, tau=0.001, lr_actor=1e-4,
                      lr_critic=1e-3, seed=0)
    torch.manual_seed(0)
    a = actor(3, 1)
    c = critic(3, 1, {"type": "categorical", "v_min": -300.0,
                      "v_max": 0.0, "n_atoms": 51})
    eng.load_from_modules(a, a, c, c)
    eng.synth_fill(cap, seed=1)
    assert eng.counters()["size"] == cap
    root = float(eng.read("sum_tree")[1])
    assert root == pytest.approx(cap, rel=1e-12)
    eng.train_steps(50)
    assert eng.counters()["adam_t_actor"] == 50
