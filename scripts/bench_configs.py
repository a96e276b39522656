#!/usr/bin/env python3
"""Benchmark all five BASELINE.json configs on one MI355X; prints one JSON
line per config (learner grad-steps/s; actor env-steps/s where relevant).

  1. Pendulum-v1 DDPG, uniform replay, CPU eager          (plumbing)
  2. Pendulum-v1 D4PG (51 atoms, n-step 5, PER) GPU learner  (flagship)
  3. HalfCheetah-spec D4PG (obs 17 / act 6) GPU learner
  4. Humanoid-spec D4PG (obs 376 / act 17) GPU learner
  5. synthetic wide-batch B=4096 / H=1024 pure MFMA throughput
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def bench_engine(obs, act, hidden, batch, cap, steps, name):
    from d4pg_amd.models import actor, critic
    from d4pg_amd.ops import FusedEngine
    eng = FusedEngine(obs_dim=obs, act_dim=act, hidden=hidden, n_atoms=51,
                      batch=batch, capacity=cap, v_min=-300.0, v_max=0.0,
                      gamma_n=0.99 ** 5, tau=0.001, lr_actor=1e-4,
                      lr_critic=1e-3, seed=0)
    torch.manual_seed(0)
    a = actor(obs, act, hidden=hidden)
    c = critic(obs, act, {"type": "categorical", "v_min": -300.0,
                          "v_max": 0.0, "n_atoms": 51}, hidden=hidden)
    eng.load_from_modules(a, a, c, c)
    eng.synth_fill(cap, seed=7)
    eng.train_steps(max(200, steps // 10))
    t0 = time.perf_counter()
    eng.train_steps(steps)
    dt = time.perf_counter() - t0
    print(json.dumps({"config": name, "grad_steps_per_sec": steps / dt,
                      "ms_per_step": dt / steps * 1e3,
                      "obs": obs, "act": act, "hidden": hidden,
                      "batch": batch, "replay": cap}), flush=True)
    del eng


def main():
    # 1: CPU eager DDPG, uniform replay
    from d4pg_amd.algo.d4pg import DDPG
    agent = DDPG(3, 1, memory_size=20000, batch_size=64,
                 prioritized_replay=False,
                 critic_dist_info={"type": "categorical", "v_min": -300.0,
                                   "v_max": 0.0, "n_atoms": 51},
                 n_steps=1, device="cpu", backend="eager", seed=0)
    rng = np.random.default_rng(0)
    for _ in range(2000):
        agent.replayBuffer.add(rng.standard_normal(3), rng.uniform(-1, 1, 1),
                               -rng.random(), rng.standard_normal(3), 0.0)
    for _ in range(5):
        agent.train()
    t0 = time.perf_counter()
    for _ in range(100):
        agent.train()
    dt = time.perf_counter() - t0
    print(json.dumps({"config": "1_pendulum_ddpg_cpu_uniform",
                      "grad_steps_per_sec": 100 / dt}), flush=True)

    if not torch.cuda.is_available():
        print(json.dumps({"note": "no GPU; configs 2-5 skipped"}))
        return

    # 2: flagship Pendulum D4PG
    bench_engine(3, 1, 256, 64, 1_000_000, 50000,
                 "2_pendulum_d4pg_gpu")
    # 3: HalfCheetah-spec
    bench_engine(17, 6, 256, 64, 1_000_000, 50000,
                 "3_halfcheetah_spec_d4pg_gpu")
    # 4: Humanoid-spec (obs 376 — exercises the wide fan-in path)
    bench_engine(376, 17, 256, 64, 1_000_000, 20000,
                 "4_humanoid_spec_d4pg_gpu")
    # 5: wide-batch MFMA
    bench_engine(17, 6, 1024, 4096, 1_000_000, 1000,
                 "5_wide_b4096_h1024_mfma")

    # actor-side env throughput (CPU side of the box)
    from d4pg_amd.envs.vector import VectorPendulum
    from d4pg_amd.models import actor as actor_net
    env = VectorPendulum(64, seed=0)
    net = actor_net(3, 1)
    net.eval()
    obs = env.reset()
    with torch.no_grad():
        t0 = time.perf_counter()
        for _ in range(1500):
            a2 = net(torch.from_numpy(obs)).numpy()
            obs, r, done = env.step(np.clip(a2, -1, 1))
            if done:
                obs = env.reset()
        dt = time.perf_counter() - t0
    print(json.dumps({"config": "actor_vector64",
                      "env_steps_per_sec": 1500 * 64 / dt}), flush=True)


if __name__ == "__main__":
    main()
