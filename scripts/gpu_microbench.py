#!/usr/bin/env python3
"""Micro-benchmark of the fused learner on one MI355X.

Compares:
  1. eager-torch GPU train step (the naive port baseline we must beat)
  2. fused engine, uncaptured launches
  3. fused engine, hipGraph replay at several steps-per-graph

Run on the GPU box:  python scripts/gpu_microbench.py [--steps N]
Prints one JSON line per variant.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def bench_eager_gpu(steps):
    """Eager torch-on-GPU D4PG step, CPU PER replay (what a straight port
    of the reference to ROCm would do)."""
    from d4pg_amd.algo.d4pg import DDPG
    agent = DDPG(3, 1, memory_size=100000, batch_size=64,
                 critic_dist_info={"type": "categorical", "v_min": -300.0,
                                   "v_max": 0.0, "n_atoms": 51},
                 n_steps=5, gamma=0.99, device="cuda", backend="eager",
                 seed=0)
    rng = np.random.default_rng(0)
    for i in range(5000):
        agent.replayBuffer.add(rng.standard_normal(3), rng.uniform(-1, 1, 1),
                               -rng.random(), rng.standard_normal(3), 0.0)
    for _ in range(10):
        agent.train()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        agent.train()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return steps / dt


def make_engine(batch=64, hidden=256, obs=3, act=1, cap=1000000):
    from d4pg_amd.ops import FusedEngine
    from d4pg_amd.models import actor, critic
    eng = FusedEngine(obs_dim=obs, act_dim=act, hidden=hidden, n_atoms=51,
                      batch=batch, capacity=cap, v_min=-300.0, v_max=0.0,
                      gamma_n=0.99 ** 5, tau=0.001, lr_actor=1e-4,
                      lr_critic=1e-4, seed=0)
    torch.manual_seed(0)
    a = actor(obs, act, hidden=hidden)
    c = critic(obs, act, {"type": "categorical", "v_min": -300.0,
                          "v_max": 0.0, "n_atoms": 51}, hidden=hidden)
    eng.load_from_modules(a, a, c, c)
    eng.synth_fill(min(cap, 1000000), seed=7)
    return eng


def bench_engine_uncaptured(eng, steps):
    eng.step(20)
    t0 = time.perf_counter()
    eng.step(steps)
    dt = time.perf_counter() - t0
    return steps / dt


def bench_engine_graph(eng, steps, spg):
    eng.train_steps(spg * 2, steps_per_graph=spg)
    n = (steps // spg) * spg
    t0 = time.perf_counter()
    eng.train_steps(n, steps_per_graph=spg)
    dt = time.perf_counter() - t0
    return n / dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2000)
    ap.add_argument("--skip-eager", action="store_true")
    ap.add_argument("--wide", action="store_true",
                    help="also run the B=4096/H=1024 MFMA config")
    args = ap.parse_args()

    out = {}
    if not args.skip_eager:
        r = bench_eager_gpu(min(args.steps, 300))
        out["eager_gpu_steps_per_sec"] = r
        print(json.dumps({"variant": "eager_gpu", "steps_per_sec": r}),
              flush=True)

    eng = make_engine()
    r = bench_engine_uncaptured(eng, min(args.steps, 1000))
    out["engine_uncaptured"] = r
    print(json.dumps({"variant": "engine_uncaptured", "steps_per_sec": r}),
          flush=True)
    for spg in (1, 8, 32):
        r = bench_engine_graph(eng, args.steps, spg)
        out[f"engine_graph_{spg}"] = r
        print(json.dumps({"variant": f"engine_graph_spg{spg}",
                          "steps_per_sec": r}), flush=True)
    del eng

    if args.wide:
        eng = make_engine(batch=4096, hidden=1024, obs=17, act=6, cap=1000000)
        r = bench_engine_graph(eng, max(100, args.steps // 20), 4)
        print(json.dumps({"variant": "engine_wide_b4096_h1024",
                          "steps_per_sec": r,
                          "gflops_per_step_approx": 4096 * 1024 * 1024 * 2
                          * 14 / 1e9}), flush=True)
        del eng


if __name__ == "__main__":
    main()
