"""Learner data-parallelism tests (parallel/dp.py) — CPU/gloo.

The core parity claim: N ranks at batch B with
gradient all-reduce-averaging between backward and Adam are numerically ONE
learner at batch N*B.  Verified here with the eager backend over gloo at
world_size 2 (the same grad_sync hook bench.py uses for its CPU rehearsal);
the HIP-engine twin of this test runs on the GPU (test_gpu_dist.py) via the
engine split-step API.
"""

import os

import numpy as np
import torch
import torch.multiprocessing as mp


def _mk_agent(batch_size):
    from d4pg_amd.algo.d4pg import DDPG
    return DDPG(3, 1, memory_size=1000, batch_size=batch_size,
                gamma=0.99, tau=0.001, prioritized_replay=False,
                critic_dist_info={"type": "categorical", "v_min": -300.0,
                                  "v_max": 0.0, "n_atoms": 51},
                n_steps=5, device="cpu", backend="eager", seed=42)


def _full_batch(n=128):
    rng = np.random.default_rng(7)
    s = rng.standard_normal((n, 3)).astype(np.float32)
    a = rng.uniform(-1, 1, (n, 1)).astype(np.float32)
    r = (-rng.random(n)).astype(np.float32)
    s2 = rng.standard_normal((n, 3)).astype(np.float32)
    d = np.zeros(n, np.float32)
    return s, a, r, s2, d


def _params_vec(agent):
    return torch.cat([p.detach().reshape(-1)
                      for p in list(agent.actor.parameters())
                      + list(agent.critic.parameters())])


def _rank_main(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    from d4pg_amd.parallel.dp import eager_grad_sync
    dist.init_process_group("gloo", rank=rank, world_size=world)
    agent = _mk_agent(batch_size=64)
    agent.grad_sync = eager_grad_sync()
    s, a, r, s2, d = _full_batch(128)
    sl = slice(rank * 64, (rank + 1) * 64)
    for _ in range(3):
        agent._train_step_eager(
            (s[sl], a[sl], r[sl], s2[sl], d[sl], None, None))
    if rank == 0:
        q.put(_params_vec(agent).numpy())
    dist.barrier()
    dist.destroy_process_group()


def test_two_rank_dp_equals_big_batch():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, 29721, q))
             for r in range(2)]
    for p in procs:
        p.start()
    dp_params = q.get(timeout=300)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    ref = _mk_agent(batch_size=128)
    s, a, r, s2, d = _full_batch(128)
    for _ in range(3):
        ref._train_step_eager((s, a, r, s2, d, None, None))
    np.testing.assert_allclose(dp_params, _params_vec(ref).numpy(),
                               rtol=2e-5, atol=2e-6)


def test_grad_sync_hook_called_on_both_nets():
    """The hook must fire after EACH backward (critic then actor) so the
    critic Adam lands before the policy forward (reference order,
    ddpg.py:229-244)."""
    agent = _mk_agent(batch_size=32)
    seen = []
    agent.grad_sync = lambda m: seen.append(m)
    s, a, r, s2, d = _full_batch(32)
    agent._train_step_eager((s, a, r, s2, d, None, None))
    assert seen == [agent.critic, agent.actor]
