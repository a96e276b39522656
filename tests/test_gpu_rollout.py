"""GPU-resident rollout parity tests (device Pendulum + K11 noise + K15
n-step fold vs the numpy oracles VectorPendulum / VecNStep, and the torch
actor forward).  Reference semantics being re-expressed: the per-env-step
actor loop /root/reference/main.py:142-152, GaussianNoise
random_process.py:4-21, n-step fold replay_memory.py:38-45."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

O, A, H, K = 3, 1, 256, 51
DIST = {"type": "categorical", "v_min": -300.0, "v_max": 0.0, "n_atoms": K}


def make_engine(capacity, seed=0, batch=64):
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    eng = FusedEngine(obs_dim=O, act_dim=A, hidden=H, n_atoms=K,
                      batch=batch, capacity=capacity, v_min=-300.0,
                      v_max=0.0, gamma_n=0.99 ** 5, tau=0.001,
                      lr_actor=1e-4, lr_critic=1e-4, seed=seed)
    torch.manual_seed(seed)
    a = actor(O, A, hidden=H)
    c = critic(O, A, DIST, hidden=H)
    eng.load_from_modules(a, a, c, c)
    return eng, a


def cpu_rollout(net, th0, td0, M, n_steps, horizon, gamma=0.99):
    """Oracle: VectorPendulum + VecNStep with the SAME start state and a
    noiseless policy; returns transitions in the device emit order
    (tick-major, env-minor)."""
    from d4pg_amd.envs.vector import VecNStep, VectorPendulum
    env = VectorPendulum(M, seed=0, horizon=horizon)
    env.th = np.asarray(th0, np.float64).copy()
    env.thdot = np.asarray(td0, np.float64).copy()
    env.t = 0
    fold = VecNStep(M, O, A, n_steps, gamma)
    obs = env._obs()
    outs = []
    with torch.no_grad():
        for t in range(horizon):
            a = net(torch.from_numpy(obs)).numpy()
            a = np.clip(a, -1.0, 1.0).astype(np.float32)
            obs2, r, done = env.step(a)
            out = fold.push(obs, a, r, obs2, done)
            if out is not None:
                outs.append(out)
            obs = obs2
    return [np.concatenate([o[i] for o in outs]) for i in range(5)]


def test_rollout_dynamics_and_fold_parity():
    M, N, HOR = 8, 3, 10
    cap = M * (HOR - N + 1)
    eng, net = make_engine(capacity=cap, seed=4)
    eng.rollout_alloc(M, N, horizon=HOR, gamma=0.99, eps=0.0, seed=9)
    rng = np.random.default_rng(11)
    th0 = rng.uniform(-np.pi, np.pi, M).astype(np.float32)
    td0 = rng.uniform(-1, 1, M).astype(np.float32)
    eng.rollout_set_state(th0, td0)
    env_steps, emitted = eng.rollout_run(1, reset=False, use_graph=False)
    assert env_steps == M * HOR and emitted == cap
    s, a, r, s2, d = eng.replay_rows()
    S, Aa, R, S2, D = cpu_rollout(net, th0, td0, M, N, HOR)
    assert s.shape == S.shape
    np.testing.assert_allclose(a, Aa, rtol=2e-4, atol=2e-4)
    np.testing.assert_allclose(s, S, rtol=2e-4, atol=2e-4)
    np.testing.assert_allclose(s2, S2, rtol=2e-4, atol=3e-3)
    np.testing.assert_allclose(r, R, rtol=2e-3, atol=5e-3)
    np.testing.assert_allclose(d, D)
    # final-tick transitions are flagged done (horizon semantics)
    assert d[-M:].sum() == M and d[:-M].sum() == 0


def test_rollout_gaussian_noise_moments():
    M, N, HOR = 256, 1, 6
    cap = M * HOR
    eng, net = make_engine(capacity=cap, seed=5)
    rng = np.random.default_rng(3)
    th0 = rng.uniform(-np.pi, np.pi, M).astype(np.float32)
    td0 = rng.uniform(-1, 1, M).astype(np.float32)

    def actions_with_eps(eps):
        eng.rollout_alloc(M, N, horizon=HOR, eps=eps, seed=77)
        eng.rollout_set_state(th0, td0)
        eng.rollout_run(1, reset=False, use_graph=False)
        _, a, _, _, _ = eng.replay_rows()
        return a.ravel().copy()

    a0 = actions_with_eps(0.0)
    a1 = actions_with_eps(0.3)
    # the first tick's M actions differ by exactly clip(tanh+0.3*N)-tanh;
    # unclipped samples should look ~N(0, 0.3)
    diff = a1[:M] - a0[:M]
    inner = (np.abs(a1[:M]) < 0.999) & (np.abs(a0[:M]) < 0.999)
    assert inner.sum() > 50
    std = diff[inner].std()
    assert 0.2 < std < 0.42, f"noise std {std} not ~0.3"


def test_rollout_reset_and_throughput_then_train():
    """Graph-replayed episodes with device reset fill the replay; the
    learner then trains from those device-generated transitions."""
    import time
    M, N, HOR = 1024, 5, 200
    eng, net = make_engine(capacity=1 << 20, seed=6)
    eng.rollout_alloc(M, N, horizon=HOR, eps=0.3, seed=13)
    eng.rollout_run(1)                      # capture + warm
    t0 = time.perf_counter()
    env_steps, emitted = eng.rollout_run(4)
    dt = time.perf_counter() - t0
    sps = env_steps / dt
    print(f"\n[gpu-rollout] {sps/1e6:.2f}M env-steps/s "
          f"(M={M}, horizon={HOR})")
    assert sps > 1e5
    c = eng.counters()
    assert c["size"] == 5 * M * (HOR - N + 1)
    s, a, r, s2, d = eng.replay_rows()
    # physical sanity of device-generated transitions
    assert np.all(np.abs(a) <= 1.0) and np.all(r <= 0.0)
    norm = s[:, 0] ** 2 + s[:, 1] ** 2
    np.testing.assert_allclose(norm, np.ones_like(norm), atol=1e-3)
    assert np.all(np.abs(s[:, 2]) <= 8.0 + 1e-5)
    # two graph replays give different episodes (device epoch advances)
    # and the learner trains straight off the device-filled replay+tree
    eng.step(20)
    c2 = eng.counters()
    assert c2["adam_t_actor"] == 20
    assert np.isfinite(c2["loss_critic"])


def test_rollout_ou_noise_runs():
    M, N, HOR = 64, 2, 8
    eng, net = make_engine(capacity=M * (HOR - N + 1), seed=7)
    eng.rollout_alloc(M, N, horizon=HOR, noise="ou", eps=1.0,
                      ou_theta=0.15, ou_sigma=0.2, seed=21)
    eng.rollout_run(1, use_graph=False)
    _, a, _, _, _ = eng.replay_rows()
    assert np.all(np.abs(a) <= 1.0)
    assert a.std() > 1e-4
