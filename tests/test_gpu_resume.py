"""Exact-resume on the GPU engine: the full state_dict (params, targets,
Adam moments, on-HBM replay + sum/min trees, schedule counters, RNG)
restores training to a BITWISE identical trajectory — a capability the
reference lacks entirely (SURVEY.md §5 checkpoint row: save-only, no
loader)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _mk_agent(seed=0, batch_size=64):
    from d4pg_amd.algo.d4pg import DDPG
    return DDPG(3, 1, memory_size=8192, batch_size=batch_size,
                critic_dist_info={"type": "categorical", "v_min": -300.0,
                                  "v_max": 0.0, "n_atoms": 51},
                n_steps=5, gamma=0.99, prioritized_replay=True,
                device="cuda", backend="hip", seed=seed)


def _fill(agent, n=2000, seed=7):
    rng = np.random.default_rng(seed)
    for _ in range(n):
        agent.replayBuffer.add(rng.standard_normal(3).astype("f"),
                               rng.uniform(-1, 1, 1).astype("f"),
                               -rng.random(),
                               rng.standard_normal(3).astype("f"), 0.0)


def _flat_params(agent):
    from d4pg_amd.ops import pack_net
    agent.engine.sync_params_if_dirty()
    return torch.cat([pack_net(agent.actor), pack_net(agent.critic),
                      pack_net(agent.actor_target),
                      pack_net(agent.critic_target)])


def test_engine_resume_bitwise():
    a1 = _mk_agent(seed=0)
    _fill(a1)
    for _ in range(5):
        a1.train()
    st = a1.state_dict()
    assert st["engine"]["counters"]["adam_t_actor"] == 5
    assert "replay" in st and int(st["replay"]["size"]) == 2000

    # original continues 5 more steps
    for _ in range(5):
        a1.train()
    ref = _flat_params(a1)

    # fresh agent (different init), restore everything, continue 5 steps
    a2 = _mk_agent(seed=99)
    _fill(a2, n=50, seed=1)        # different replay — must be overwritten
    a2.train()                     # build the engine bridge
    a2.load_state_dict(st)
    cnt = a2.engine.engine.counters()
    assert cnt["adam_t_actor"] == 5
    assert cnt["size"] == 2000
    for _ in range(5):
        a2.train()
    got = _flat_params(a2)
    np.testing.assert_array_equal(got.numpy(), ref.numpy())
    c1 = a1.engine.engine.counters()
    c2 = a2.engine.engine.counters()
    assert c1["adam_t_actor"] == c2["adam_t_actor"] == 10
    # the loss scalar accumulates via cross-workgroup fp32 atomics, whose
    # order is nondeterministic — approx, not bitwise
    assert c1["loss_critic"] == pytest.approx(c2["loss_critic"], rel=1e-5)


def test_engine_resume_wide_batch_graph_path():
    """the wide/graph path (B >= 512, hipGraph
    replay) bakes the philox seed into captured kernel ARGUMENTS, so a
    restored seed must invalidate the graph — otherwise a resumed run
    silently keeps the stale seed.  Verifies the post-restore trajectory
    matches the original bitwise THROUGH graph-replayed steps."""
    a1 = _mk_agent(seed=3, batch_size=512)
    _fill(a1, n=3000, seed=11)
    a1.train()                       # builds the bridge
    eng1 = a1.engine.engine
    eng1.train_steps(8, steps_per_graph=4)   # capture + replay
    st = a1.state_dict()
    eng1.train_steps(8, steps_per_graph=4)
    ref = _flat_params(a1)

    a2 = _mk_agent(seed=77, batch_size=512)
    a2.load_state_dict(st)           # builds bridge + restores (incl seed)
    eng2 = a2.engine.engine
    assert eng2._captured == 0       # set_seed invalidated any capture
    eng2.train_steps(8, steps_per_graph=4)   # recapture with restored seed
    a2.engine.ddpg.train_steps_done += 8
    got = _flat_params(a2)
    np.testing.assert_array_equal(got.numpy(), ref.numpy())
