"""HogWild shared-memory multi-worker mode (reference-parity, CPU).

Reproduces the reference's asynchronous-data-parallel topology
(/root/reference/main.py:382-405 + SURVEY.md §2b): a global DDPG whose
actor/critic parameters live in POSIX shared memory, two SharedAdam
optimizers with lr = 1e-3 / n_workers, a shared global step-count tensor,
N worker processes running Worker.work against the shared state (lock-free,
races tolerated by design) and one evaluator process.

This mode exists for capability parity and CPU-cluster use; the MI355X
scaling path is parallel/learner.py (one process per GPU over RCCL).
"""

from __future__ import annotations

import torch
import torch.multiprocessing as mp

from ..algo.d4pg import DDPG
from ..algo.shared_adam import SharedAdam
from ..config import critic_dist_info, run_dir_name
from ..envs import make, obs_act_dims
from ..utils.logging import SummaryWriter
from .worker import Worker, global_model_eval


def _build_agent(args, env, seed):
    obs_dim, act_dim = obs_act_dims(env, her=bool(args.her))
    from ..config import noise_kwargs
    return DDPG(obs_dim, act_dim, env=env, memory_size=args.rmsize,
                batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
                prioritized_replay=bool(args.p_replay),
                critic_dist_info=critic_dist_info(args),
                n_steps=args.n_steps, seed=seed, **noise_kwargs(args))


def _worker_main(name, args, global_model, global_count, opt_actor,
                 opt_critic, max_cycles):
    env = make(args.env, seed=args.seed + int(name))
    env._max_episode_steps = args.max_steps
    agent = _build_agent(args, env, seed=args.seed + int(name))
    agent.assign_global_optimizer(opt_actor, opt_critic)
    writer = SummaryWriter(run_dir_name(args)) if int(name) == 0 else None
    w = Worker(name, args, agent, env, writer=writer,
               run_dir=run_dir_name(args))
    w.work(global_model=global_model, global_count=global_count,
           max_cycles=max_cycles)


def run_hogwild(args, max_cycles: int | None = None,
                with_evaluator: bool = True):
    """Spawn N workers (+ evaluator) against shared-memory global state."""
    env = make(args.env, seed=args.seed)
    env._max_episode_steps = args.max_steps
    global_model = _build_agent(args, env, seed=args.seed)
    global_model.share_memory()
    opt_actor = SharedAdam(global_model.actor.parameters(),
                           lr=1e-3 / args.n_workers)
    opt_critic = SharedAdam(global_model.critic.parameters(),
                            lr=1e-3 / args.n_workers)
    global_count = torch.zeros(1)
    global_count.share_memory_()

    ctx = mp.get_context("spawn")
    procs = []
    if with_evaluator:
        env_factory = _EnvFactory(args)
        pe = ctx.Process(target=global_model_eval,
                         args=(global_model, global_count, args, env_factory),
                         daemon=True)
        pe.start()
        procs.append(pe)
    workers = []
    for i in range(args.n_workers):
        p = ctx.Process(target=_worker_main,
                        args=(str(i), args, global_model, global_count,
                              opt_actor, opt_critic, max_cycles))
        p.start()
        workers.append(p)
    for p in workers:
        p.join()
    for p in procs:
        p.terminate()
    return global_model, int(global_count.item())


class _EnvFactory:
    """Picklable env factory for the evaluator process."""

    def __init__(self, args):
        self.args = args

    def __call__(self):
        env = make(self.args.env, seed=self.args.seed + 999)
        env._max_episode_steps = self.args.max_steps
        return env
