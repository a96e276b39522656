#!/usr/bin/env python3
"""CLI entry point — command-line parity with ``python main.py --flags`` of
the reference (/root/reference/main.py:371-405).

  --multithread 0 (default): a single Worker trains in-process
                             (reference main.py:390-392).
  --multithread 1: N HogWild worker processes + evaluator against
                   shared-memory global state (reference main.py:394-405).

MI355X-native distributed training (one process per GPU over RCCL) is
launched via torch.distributed.run against bench.py / d4pg_amd.parallel
instead — see README.
"""

from __future__ import annotations

from d4pg_amd.algo.d4pg import DDPG
from d4pg_amd.config import (configure_env_params, critic_dist_info,
                             make_parser, run_dir_name)
from d4pg_amd.envs import make, obs_act_dims
from d4pg_amd.parallel.hogwild import run_hogwild
from d4pg_amd.parallel.worker import Worker
from d4pg_amd.utils.logging import SummaryWriter


def main(argv=None):
    args = make_parser().parse_args(argv)
    configure_env_params(args)

    if args.multithread:
        run_hogwild(args)
        return

    env = make(args.env, seed=args.seed)
    env._max_episode_steps = args.max_steps
    obs_dim, act_dim = obs_act_dims(env, her=bool(args.her))
    device = args.device
    if device == "auto":
        import torch
        device = "cuda" if torch.cuda.is_available() else "cpu"
    backend = args.backend
    if backend == "auto":
        backend = "hip" if device == "cuda" else "eager"
    from d4pg_amd.config import noise_kwargs
    agent = DDPG(obs_dim, act_dim, env=env, memory_size=args.rmsize,
                 batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
                 lr_actor=args.lr_actor, lr_critic=args.lr_critic,
                 prioritized_replay=bool(args.p_replay),
                 critic_dist_info=critic_dist_info(args),
                 n_steps=args.n_steps, device=device, backend=backend,
                 seed=args.seed, **noise_kwargs(args))
    rd = run_dir_name(args)
    writer = SummaryWriter(rd)
    worker = Worker("1", args, agent, env, writer=writer, run_dir=rd)
    worker.work()


if __name__ == "__main__":
    main()
