"""GPU parity tests: the fused HIP engine vs the eager torch oracle.

Strategy (SURVEY.md §4 'kernel parity'): ingest a known transition set, run
ONE engine step, read back the batch the device sampled (bidx etc.), then
replay the identical train step in eager fp32 torch on that exact batch and
compare every intermediate (target dist, projection, critic dist, CE
gradient, priorities) and the post-step parameter slabs.
"""

import copy

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DIST = {"type": "categorical", "v_min": -300.0, "v_max": 0.0, "n_atoms": 51}
O, A, H, K, B = 3, 1, 256, 51, 64
GAMMA_N = 0.99 ** 5
TAU = 0.001
LR = 1e-4


def make_engine(seed=0, capacity=4096):
    from d4pg_amd.ops import FusedEngine
    return FusedEngine(obs_dim=O, act_dim=A, hidden=H, n_atoms=K, batch=B,
                       capacity=capacity, v_min=-300.0, v_max=0.0,
                       gamma_n=GAMMA_N, tau=TAU, lr_actor=LR, lr_critic=LR,
                       seed=seed)


def make_modules(seed=0):
    from d4pg_amd.models import actor, critic
    torch.manual_seed(seed)
    a = actor(O, A)
    c = critic(O, A, DIST)
    at = copy.deepcopy(a)
    ct = copy.deepcopy(c)
    return a, at, c, ct


def random_transitions(n, seed=0):
    rng = np.random.default_rng(seed)
    s = rng.standard_normal((n, O)).astype(np.float32)
    a = rng.uniform(-1, 1, (n, A)).astype(np.float32)
    r = rng.uniform(-30, 0, n).astype(np.float32)
    s2 = rng.standard_normal((n, O)).astype(np.float32)
    d = (rng.random(n) < 0.05).astype(np.float32)
    return s, a, r, s2, d


@pytest.fixture(scope="module")
def stepped():
    """One fused engine step on known data + everything needed to check it."""
    eng = make_engine()
    a, at, c, ct = make_modules()
    eng.load_from_modules(a, at, c, ct)
    tr = random_transitions(1024, seed=3)
    eng.ingest(*[torch.from_numpy(x) for x in tr])
    eng.step(1)
    out = {name: eng.read(name) for name in
           ["bs", "ba", "br", "bs2", "bd", "bw", "bidx", "a2", "p_t",
            "m_proj", "q", "dlog", "pri", "sum_tree", "min_tree"]}
    out["counters"] = eng.counters()
    out["slab_actor"] = eng.store_slab("actor")
    out["slab_critic"] = eng.store_slab("critic")
    out["slab_actor_t"] = eng.store_slab("actor_target")
    out["slab_critic_t"] = eng.store_slab("critic_target")
    out["g_actor"] = eng.store_slab("g_actor")
    out["g_critic"] = eng.store_slab("g_critic")
    out["tree_cap"] = eng.info()["tree_cap"]
    return {"eng_out": out, "modules": (a, at, c, ct), "transitions": tr}


def test_sampled_batch_consistent(stepped):
    """The gathered batch rows must equal the stored transitions at bidx."""
    o = stepped["eng_out"]
    s, a, r, s2, d = stepped["transitions"]
    idx = o["bidx"].numpy()
    assert ((0 <= idx) & (idx < 1024)).all()
    np.testing.assert_allclose(o["bs"].numpy(), s[idx], atol=0)
    np.testing.assert_allclose(o["ba"].numpy(), a[idx], atol=0)
    np.testing.assert_allclose(o["br"].numpy(), r[idx], atol=0)
    np.testing.assert_allclose(o["bs2"].numpy(), s2[idx], atol=0)
    np.testing.assert_allclose(o["bd"].numpy(), d[idx], atol=0)
    # uniform priorities at ingest => IS weights all 1
    np.testing.assert_allclose(o["bw"].numpy(), np.ones(B), atol=1e-5)


def test_forward_and_projection_parity(stepped):
    from d4pg_amd.algo.projection import categorical_projection
    o = stepped["eng_out"]
    a, at, c, ct = stepped["modules"]
    s = torch.from_numpy(o["bs"].numpy())
    act = torch.from_numpy(o["ba"].numpy())
    r = torch.from_numpy(o["br"].numpy())
    s2 = torch.from_numpy(o["bs2"].numpy())
    d = torch.from_numpy(o["bd"].numpy())
    with torch.no_grad():
        a2 = at(s2)
        p_t = ct(s2, a2)
        m = categorical_projection(p_t, r, d, -300.0, 0.0, GAMMA_N)
        q = c(s, act)
    np.testing.assert_allclose(o["a2"].numpy(), a2.numpy(), atol=2e-5)
    np.testing.assert_allclose(o["p_t"].numpy(), p_t.numpy(), atol=2e-5)
    np.testing.assert_allclose(o["m_proj"].numpy(), m.numpy(), atol=5e-5)
    np.testing.assert_allclose(o["q"].numpy(), q.numpy(), atol=2e-5)
    # CE gradient dlogits = (q - m)/B
    np.testing.assert_allclose(o["dlog"].numpy(),
                               ((q - m) / B).numpy(), atol=5e-6)
    # priorities = sum(m*q) + eps
    np.testing.assert_allclose(o["pri"].numpy(),
                               (m * q).sum(1).numpy() + 1e-6, atol=1e-5)


def test_post_step_params_parity(stepped):
    """Full step: compare post-Adam/soft-update slabs against an eager
    replication of the step on the identical batch."""
    from d4pg_amd.algo.projection import categorical_projection
    from d4pg_amd.ops import pack_net
    o = stepped["eng_out"]
    a, at, c, ct = [copy.deepcopy(m) for m in stepped["modules"]]
    s = torch.from_numpy(o["bs"].numpy())
    act = torch.from_numpy(o["ba"].numpy())
    r = torch.from_numpy(o["br"].numpy())
    s2 = torch.from_numpy(o["bs2"].numpy())
    d = torch.from_numpy(o["bd"].numpy())

    opt_c = torch.optim.Adam(c.parameters(), lr=LR)
    opt_a = torch.optim.Adam(a.parameters(), lr=LR)
    with torch.no_grad():
        m = categorical_projection(ct(s2, at(s2)), r, d, -300.0, 0.0, GAMMA_N)
    q = c(s, act)
    loss_c = -(m * torch.log(q + 1e-10)).sum(1).mean()
    c.zero_grad()
    loss_c.backward()
    # gradient parity first (tight)
    gc = torch.cat([torch.cat([getattr(c, n).weight.grad.t().reshape(-1),
                               getattr(c, n).bias.grad])
                    for n in ["fc1", "fc2", "fc2_2", "fc3"]])
    np.testing.assert_allclose(o["g_critic"].numpy(), gc.numpy(), atol=3e-5)
    opt_c.step()

    z = torch.linspace(-300.0, 0.0, K).reshape(-1, 1)
    pl = -(c(s, a(s)).matmul(z)).mean()
    a.zero_grad()
    pl.backward()
    ga = torch.cat([torch.cat([getattr(a, n).weight.grad.t().reshape(-1),
                               getattr(a, n).bias.grad])
                    for n in ["fc1", "fc2", "fc2_2", "fc3"]])
    np.testing.assert_allclose(o["g_actor"].numpy(), ga.numpy(), atol=3e-5)
    opt_a.step()

    with torch.no_grad():
        for tp, sp in zip(at.parameters(), a.parameters()):
            tp.lerp_(sp, TAU)
        for tp, sp in zip(ct.parameters(), c.parameters()):
            tp.lerp_(sp, TAU)

    np.testing.assert_allclose(o["slab_critic"].numpy(),
                               pack_net(c).numpy(), atol=2e-5)
    np.testing.assert_allclose(o["slab_actor"].numpy(),
                               pack_net(a).numpy(), atol=2e-5)
    np.testing.assert_allclose(o["slab_critic_t"].numpy(),
                               pack_net(ct).numpy(), atol=2e-5)
    np.testing.assert_allclose(o["slab_actor_t"].numpy(),
                               pack_net(at).numpy(), atol=2e-5)


def test_tree_writeback(stepped):
    """Sampled leaves carry priority^alpha after the step; root = sum."""
    o = stepped["eng_out"]
    cap = o["tree_cap"]
    tree = o["sum_tree"].numpy()
    idx = o["bidx"].numpy()
    pri = o["pri"].numpy()
    for i in range(B):
        assert tree[cap + idx[i]] == pytest.approx(pri[i] ** 0.6, rel=1e-5)
    leaves = tree[cap:cap + 1024]
    assert tree[1] == pytest.approx(leaves.sum(), rel=1e-9)
    mt = o["min_tree"].numpy()
    assert mt[1] == pytest.approx(leaves[leaves > 0].min(), rel=1e-9)


def test_counters_advance(stepped):
    cnt = stepped["eng_out"]["counters"]
    assert cnt["adam_t_actor"] == 1
    assert cnt["adam_t_critic"] == 1
    assert cnt["beta_t"] == 1
    assert cnt["size"] == 1024
    assert np.isfinite(cnt["loss_critic"])


def test_graph_replay_matches_eager_steps():
    """N captured-graph steps == N uncaptured steps (same RNG stream)."""
    eng1 = make_engine(seed=5)
    eng2 = make_engine(seed=5)
    a, at, c, ct = make_modules(seed=2)
    for e in (eng1, eng2):
        e.load_from_modules(a, at, c, ct)
        e.ingest(*[torch.from_numpy(x)
                   for x in random_transitions(512, seed=9)])
    eng1.step(6)
    eng2.train_steps(6, steps_per_graph=3)
    s1 = eng1.store_slab("critic").numpy()
    s2 = eng2.store_slab("critic").numpy()
    np.testing.assert_allclose(s1, s2, atol=0)      # bitwise identical
    c1, c2 = eng1.counters(), eng2.counters()
    assert c1["adam_t_critic"] == c2["adam_t_critic"] == 6


def test_per_sampling_tracks_priorities_gpu():
    """After many steps the engine's priority distribution drives sampling:
    leaves with larger priority are sampled more often."""
    eng = make_engine(seed=1)
    a, at, c, ct = make_modules(seed=1)
    eng.load_from_modules(a, at, c, ct)
    eng.ingest(*[torch.from_numpy(x) for x in random_transitions(256, seed=4)])
    counts = np.zeros(256)
    for _ in range(50):
        eng.step(1)
        idx = eng.read("bidx").numpy()
        np.add.at(counts, idx, 1)
    cap = eng.info()["tree_cap"]
    leaves = eng.read("sum_tree").numpy()[cap:cap + 256]
    # top-priority quartile should be sampled more than bottom quartile
    order = np.argsort(leaves)
    lo = counts[order[:64]].mean()
    hi = counts[order[-64:]].mean()
    assert hi > lo


def test_actor_forward_matches_module():
    eng = make_engine(seed=2)
    a, at, c, ct = make_modules(seed=3)
    eng.load_from_modules(a, at, c, ct)
    x = torch.randn(32, O)
    y = eng.actor_forward(x)
    with torch.no_grad():
        ref = a(x)
    np.testing.assert_allclose(y.numpy(), ref.numpy(), atol=2e-5)


def test_synth_fill_and_throughput_sanity():
    """synth_fill populates a consistent tree and the engine sustains
    hundreds of steps without numerical blowup."""
    eng = make_engine(seed=3, capacity=100000)
    a, at, c, ct = make_modules(seed=4)
    eng.load_from_modules(a, at, c, ct)
    eng.synth_fill(50000, seed=11)
    cnt = eng.counters()
    assert cnt["size"] == 50000
    tree = eng.read("sum_tree").numpy()
    cap = eng.info()["tree_cap"]
    assert tree[1] == pytest.approx(50000.0, rel=1e-9)
    eng.train_steps(200, steps_per_graph=10)
    q = eng.read("q")
    assert torch.isfinite(q).all()
    assert float((q.sum(1) - 1).abs().max()) < 1e-4
    sa = eng.store_slab("actor")
    assert torch.isfinite(sa).all()


def test_ingest_chunks_past_staging_cap():
    """Regression: one ingest() call larger than the 65536-row device
    staging buffer must chunk transparently (GPU-rollout actor ranks at
    hundreds of envs push >100k transitions per exchange)."""
    eng = make_engine(capacity=262144)
    n = 70000
    rng = np.random.default_rng(0)
    eng.ingest(torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy(rng.uniform(-1, 1, (n, A)).astype("f")),
               torch.from_numpy(rng.uniform(-30, 0, n).astype("f")),
               torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy(np.zeros(n, np.float32)))
    c = eng.counters()
    assert c["size"] == n and c["pos"] == n
    tree = eng.read("sum_tree").numpy()
    assert tree[1] == pytest.approx(float(n), rel=1e-9)
