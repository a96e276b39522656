#!/usr/bin/env python3
"""Learning-quality evidence: train Pendulum-v1 D4PG end-to-end through the
product path (Worker + fused HIP engine) and report eval returns per cycle.
Random policy scores ~-1200..-1500; a learning agent reaches -150..-400.

Usage: python scripts/gpu_train_convergence.py [cycles]
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from d4pg_amd.algo.d4pg import DDPG  # noqa: E402
from d4pg_amd.config import configure_env_params, make_parser  # noqa: E402
from d4pg_amd.envs import make, obs_act_dims  # noqa: E402
from d4pg_amd.parallel.worker import Worker  # noqa: E402

cycles = int(sys.argv[1]) if len(sys.argv) > 1 else 30

args = make_parser().parse_args(
    ["--env", "Pendulum-v1", "--max_steps", "200", "--warmup", "5",
     "--rmsize", "1000000", "--bsize", "64", "--n_steps", "5",
     "--n_eps", "1000", "--cycles_per_epoch", "1000000", "--debug", "0",
     "--episodes_per_cycle", "8", "--train_steps_per_cycle", "600",
     "--eval_trials", "3", "--seed", "0"])
configure_env_params(args)

env = make(args.env, seed=0)
env._max_episode_steps = args.max_steps
obs_dim, act_dim = obs_act_dims(env)
device = "cuda" if torch.cuda.is_available() else "cpu"
backend = "hip" if device == "cuda" else "eager"
agent = DDPG(obs_dim, act_dim, env=env, memory_size=args.rmsize,
             batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
             prioritized_replay=True,
             critic_dist_info={"type": "categorical", "v_min": -300.0,
                               "v_max": 0.0, "n_atoms": 51},
             n_steps=args.n_steps, lr_actor=1e-4, lr_critic=1e-3,
             device=device, backend=backend, seed=0)
w = Worker("conv", args, agent, env, run_dir="")
w.warmup()
t0 = time.perf_counter()
best = -1e9
for cyc in range(cycles):
    w.collect_cycle()
    w.train_cycle()
    avg_r, _ = w.evaluate()
    best = max(best, avg_r)
    print(f"cycle {cyc:3d}  t={time.perf_counter() - t0:6.1f}s  "
          f"grad_steps={agent.train_steps_done:6d}  "
          f"eval_return={avg_r:8.1f}  best={best:8.1f}", flush=True)
print(f"FINAL best_eval_return {best:.1f} "
      f"({'LEARNING OK' if best > -700 else 'NOT CONVERGED'})")
