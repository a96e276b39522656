"""GPU tests for the multi-GPU plumbing (run on the GPU box).

Pre-flights the pieces the driver's 8-GPU scaling run will execute
(VERDICT r1 next-steps #1/#3), on a single MI355X:

  * the nccl(=RCCL) init path + a collective on hardware (world_size 1 —
    multiple ranks cannot share one GPU under NCCL semantics, so the
    multi-rank exchange is rehearsed over gloo below and runs over real
    RCCL only on the driver's multi-GPU node);
  * zero-copy device_slab views (torch.from_blob over engine HBM) that
    the RCCL collectives operate on;
  * the engine split-step API (PH_* masks) used by gradient DP;
  * two-process LocalSGD parameter averaging with both engines on one
    GPU, exchanging over gloo;
  * bench.py's multirank path end-to-end via torchrun.
"""

import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

O, A, H, K, B = 3, 1, 256, 51, 64
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def make_engine(seed=0, capacity=4096, init_params=True):
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    eng = FusedEngine(obs_dim=O, act_dim=A, hidden=H, n_atoms=K, batch=B,
                      capacity=capacity, v_min=-300.0, v_max=0.0,
                      gamma_n=0.99 ** 5, tau=0.001, lr_actor=1e-4,
                      lr_critic=1e-4, seed=seed)
    if init_params:
        torch.manual_seed(seed)
        a = actor(O, A, hidden=H)
        c = critic(O, A, {"type": "categorical", "v_min": -300.0,
                          "v_max": 0.0, "n_atoms": K}, hidden=H)
        eng.load_from_modules(a, a, c, c)
    return eng


def fill(eng, seed=3, n=1024):
    rng = np.random.default_rng(seed)
    eng.ingest(torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy(rng.uniform(-1, 1, (n, A)).astype("f")),
               torch.from_numpy(rng.uniform(-30, 0, n).astype("f")),
               torch.from_numpy(rng.standard_normal((n, O)).astype("f")),
               torch.from_numpy(np.zeros(n, np.float32)))


def test_rccl_world1_init_and_collectives():
    """The nccl backend (RCCL on ROCm) must initialize and run collectives
    on hardware — the branch no round-1 run ever executed."""
    import torch.distributed as dist
    assert dist.is_nccl_available()
    store = dist.TCPStore("127.0.0.1", 29741, 1, True)
    dist.init_process_group("nccl", store=store, rank=0, world_size=1)
    try:
        t = torch.arange(8, dtype=torch.float32, device="cuda")
        dist.all_reduce(t)
        dist.broadcast(t, src=0)
        torch.cuda.synchronize()
        np.testing.assert_allclose(t.cpu().numpy(), np.arange(8, dtype="f"))
        # the collectives must also accept the engine's from_blob slab views
        eng = make_engine(seed=5)
        v = eng.device_slab("actor")
        before = eng.store_slab("actor").numpy().copy()
        dist.all_reduce(v)          # world 1: identity
        dist.broadcast(v, src=0)
        torch.cuda.synchronize()
        np.testing.assert_allclose(eng.store_slab("actor").numpy(), before)
    finally:
        dist.destroy_process_group()


def test_device_slab_is_zero_copy_view():
    eng = make_engine(seed=1)
    v = eng.device_slab("actor")
    assert v.is_cuda and v.dtype == torch.float32
    host = eng.store_slab("actor")
    np.testing.assert_allclose(v.cpu().numpy(), host.numpy())
    # writes through the torch view must land in engine memory
    v.mul_(2.0)
    torch.cuda.synchronize()
    np.testing.assert_allclose(eng.store_slab("actor").numpy(),
                               2.0 * host.numpy(), rtol=1e-6)


def test_split_step_matches_full_step():
    """GRADS/APPLY phase sequence == one whole train step (same seed, same
    replay): the DP insertion points change nothing when no all-reduce
    happens between them."""
    e1, e2 = make_engine(seed=9), make_engine(seed=9)
    for e in (e1, e2):
        fill(e)
    # identical initial params
    for s in ("actor", "actor_target", "critic", "critic_target"):
        e2.load_slab(s, e1.store_slab(s))
    E = e2
    e1.step(1)
    E.step_part(E.PH_CRITIC_GRADS)
    E.step_part(E.PH_CRITIC_APPLY)
    E.step_part(E.PH_ACTOR_GRADS)
    E.step_part(E.PH_ACTOR_APPLY)
    for s in ("actor", "critic", "actor_target", "critic_target"):
        np.testing.assert_allclose(e1.store_slab(s).numpy(),
                                   e2.store_slab(s).numpy(),
                                   rtol=1e-6, atol=1e-7,
                                   err_msg=f"slab {s} diverged")


def test_manual_two_engine_grad_dp_consistency():
    """Two 'ranks' (two engines on one GPU) with manual gradient
    averaging: both must hold IDENTICAL parameters after every DP step —
    the invariant the RCCL all-reduce preserves on a real multi-GPU node."""
    ea, eb = make_engine(seed=11), make_engine(seed=12)
    fill(ea, seed=21)
    fill(eb, seed=22)              # different replay shards
    for s in ("actor", "actor_target", "critic", "critic_target"):
        eb.load_slab(s, ea.store_slab(s))
    for _ in range(3):
        for e in (ea, eb):
            e.step_part(e.PH_CRITIC_GRADS)
        g = 0.5 * (ea.store_slab("g_critic") + eb.store_slab("g_critic"))
        ea.load_slab("g_critic", g)
        eb.load_slab("g_critic", g)
        for e in (ea, eb):
            e.step_part(e.PH_CRITIC_APPLY)
            e.step_part(e.PH_ACTOR_GRADS)
        g = 0.5 * (ea.store_slab("g_actor") + eb.store_slab("g_actor"))
        ea.load_slab("g_actor", g)
        eb.load_slab("g_actor", g)
        for e in (ea, eb):
            e.step_part(e.PH_ACTOR_APPLY)
    for s in ("actor", "critic", "actor_target", "critic_target"):
        np.testing.assert_allclose(ea.store_slab(s).numpy(),
                                   eb.store_slab(s).numpy(),
                                   rtol=1e-6, atol=1e-7,
                                   err_msg=f"slab {s} diverged across ranks")
    # and training actually moved the params
    assert not np.allclose(ea.store_slab("actor").numpy(),
                           make_engine(seed=11).store_slab("actor").numpy()), \
        "3 DP steps left the actor slab at its initial values"


def _localsgd_rank(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    from d4pg_amd.parallel.dp import LocalSGDSync
    dist.init_process_group("gloo", rank=rank, world_size=world)
    eng = make_engine(seed=100 + rank)
    fill(eng, seed=200 + rank)
    sync = LocalSGDSync(eng)
    sync.broadcast_initial(src=0)
    for _ in range(2):
        eng.step(4)
        sync.average()
    q.put((rank, eng.store_slab("actor").numpy(),
           eng.store_slab("critic").numpy()))
    dist.barrier()
    dist.destroy_process_group()


def test_localsgd_two_proc_one_gpu():
    """Two processes, two engines, ONE GPU, param averaging over gloo —
    the full LocalSGDSync code path minus the RCCL transport (which
    test_rccl_world1 covers; the combination runs on the driver's node)."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_localsgd_rank, args=(r, 2, 29743, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(2):
        r, a, c = q.get(timeout=600)
        out[r] = (a, c)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    np.testing.assert_allclose(out[0][0], out[1][0], rtol=1e-6, atol=1e-7)
    np.testing.assert_allclose(out[0][1], out[1][1], rtol=1e-6, atol=1e-7)


def test_bench_multirank_gloo_end_to_end():
    """bench.py's N>1 localsgd path (param averaging + transition
    all_gather inside the timed region) via torchrun, 2 ranks on 1 GPU."""
    env = dict(os.environ, BENCH_DIST_BACKEND="gloo")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29745", "--no-python-warning",
           os.path.join(REPO, "bench.py"),
           "--gpus", "2", "--steps", "48", "--warmup", "16",
           "--sync_every", "16"]
    cmd.remove("--no-python-warning")
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                         env=env, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    line = [l for l in res.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"].startswith("hogwild-rccl x2")
    assert out["value"] > 0


def test_split_step_matches_full_step_wide_mfma_path():
    """Split-step (DP insertion points) parity on the WIDE path too —
    B >= 512 takes the per-layer MFMA kernels, a different code path from
    the row-block form covered above."""
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine

    def mk(seed):
        eng = FusedEngine(obs_dim=17, act_dim=6, hidden=256, n_atoms=51,
                          batch=1024, capacity=8192, v_min=-300.0,
                          v_max=0.0, gamma_n=0.99 ** 5, tau=0.001,
                          lr_actor=1e-4, lr_critic=1e-4, seed=seed)
        torch.manual_seed(seed)
        a = actor(17, 6, hidden=256)
        c = critic(17, 6, {"type": "categorical", "v_min": -300.0,
                           "v_max": 0.0, "n_atoms": 51}, hidden=256)
        eng.load_from_modules(a, a, c, c)
        eng.synth_fill(8192, seed=seed + 3)
        return eng

    e1, e2 = mk(31), mk(31)
    e1.step(1)
    E = e2
    E.step_part(E.PH_CRITIC_GRADS)
    E.step_part(E.PH_CRITIC_APPLY)
    E.step_part(E.PH_ACTOR_GRADS)
    E.step_part(E.PH_ACTOR_APPLY)
    for s in ("actor", "critic", "actor_target", "critic_target"):
        np.testing.assert_allclose(e1.store_slab(s).numpy(),
                                   e2.store_slab(s).numpy(),
                                   rtol=1e-6, atol=1e-7,
                                   err_msg=f"wide slab {s} diverged")
