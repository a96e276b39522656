"""Worker: the per-process actor+learner loop, and the evaluator.

Schedule parity with the reference's ``Worker.work`` and
``global_model_eval`` (/root/reference/main.py:188-368, 103-134):

  per epoch: ``cycles_per_epoch`` cycles of
    { collect ``episodes_per_cycle`` episodes -> ``train_steps_per_cycle``
      gradient steps -> ``eval_trials`` greedy rollouts -> scalars
      ``avg_test_reward``/``success_rate`` -> actor.pth/critic.pth save }

with a warmup fill of ``warmup`` episodes before training
(main.py:200-243).  In multithread mode each Worker runs in its own process
against a shared-memory global model (parallel/hogwild.py); the evaluator
copies global weights every 10 s and rolls one greedy episode, maintaining
the 0.95/0.05 EWMA return (main.py:103-134).
"""

from __future__ import annotations

import os
import time

import numpy as np

from ..her import add_experience, rollout_episode
from ..utils.logging import Meter, SummaryWriter


class Worker:
    def __init__(self, name, args, agent, env, writer: SummaryWriter | None = None,
                 run_dir: str | None = None):
        self.name = str(name)
        self.args = args
        self.agent = agent
        self.env = env
        self.run_dir = run_dir or "."
        self.writer = writer
        self.grad_meter = Meter()
        self.env_meter = Meter()
        self.rng = np.random.default_rng(
            None if args.seed is None else args.seed + hash(self.name) % 1000)

    # -- warmup fill (reference main.py:200-243) --
    def warmup(self) -> None:
        for _ in range(self.args.warmup):
            ep, _, _ = rollout_episode(self.agent, self.env, noise=True)
            self.env_meter.add(len(ep))
            add_experience(self.agent.replayBuffer, self.env, ep,
                           her=bool(self.args.her), n_steps=self.args.n_steps,
                           gamma=self.args.gamma, rng=self.rng)

    def collect_cycle(self) -> None:
        for _ in range(self.args.episodes_per_cycle):
            ep, _, _ = rollout_episode(self.agent, self.env, noise=True)
            self.env_meter.add(len(ep))
            add_experience(self.agent.replayBuffer, self.env, ep,
                           her=bool(self.args.her), n_steps=self.args.n_steps,
                           gamma=self.args.gamma, rng=self.rng)

    def train_cycle(self, global_model=None, global_count=None) -> None:
        n = self.args.train_steps_per_cycle
        if global_model is None and self.agent.backend == "hip":
            # fused multi-step: one engine launch for the whole cycle
            self.agent.train()              # builds the bridge on first use
            if n > 1:
                self.agent.engine.step(n=n - 1)
            self.grad_meter.add(n)
            if global_count is not None:
                global_count += n
            return
        for _ in range(n):
            self.agent.train(global_model)
            self.grad_meter.add()
            if global_count is not None:
                global_count += 1     # HogWild-tolerated non-atomic increment

    def evaluate(self):
        returns, successes = [], []
        for _ in range(self.args.eval_trials):
            _, R, s = rollout_episode(self.agent, self.env, noise=False)
            returns.append(R)
            successes.append(float(s))
        return float(np.mean(returns)), float(np.mean(successes))

    def work(self, global_model=None, global_count=None,
             max_cycles: int | None = None) -> None:
        """The main loop.  ``global_model``/``global_count`` engage
        HogWild-parity shared-memory training (main.py:245-307);
        ``max_cycles`` bounds the run (tests/bench)."""
        if global_model is not None:
            self.agent.sync_local_global(global_model)
            self.agent.hard_update()
        self.warmup()
        cycle_idx = 0
        for epoch in range(self.args.n_eps):
            for _ in range(self.args.cycles_per_epoch):
                self.collect_cycle()
                self.train_cycle(global_model, global_count)
                avg_r, succ = self.evaluate()
                step = (global_count.item() if global_count is not None
                        else self.agent.train_steps_done)
                if self.writer is not None:
                    self.writer.add_scalar("avg_test_reward", avg_r, step)
                    self.writer.add_scalar("success_rate", succ, step)
                    self.writer.add_scalar("grad_steps_per_sec",
                                           self.grad_meter.rate(), step)
                    self.writer.add_scalar("env_steps_per_sec",
                                           self.env_meter.rate(), step)
                if self.args.debug:
                    print(f"[worker {self.name}] epoch {epoch} step {step} "
                          f"avg_test_reward {avg_r:.2f} success {succ:.2f}",
                          flush=True)
                if self.run_dir:
                    os.makedirs(self.run_dir, exist_ok=True)
                    self.agent.save(self.run_dir)
                cycle_idx += 1
                if max_cycles is not None and cycle_idx >= max_cycles:
                    return


def global_model_eval(global_model, global_count, args, env_factory,
                      stop_at: float = 1e6, period: float = 10.0,
                      max_iters: int | None = None):
    """Evaluator process body (reference main.py:103-134): copy global
    weights, one greedy rollout, EWMA return, repeat every ``period`` s."""
    from ..algo.d4pg import DDPG
    from ..config import critic_dist_info

    env = env_factory()
    from ..envs import obs_act_dims
    obs_dim, act_dim = obs_act_dims(env, her=bool(args.her))
    agent = DDPG(obs_dim, act_dim, env=env, memory_size=1000,
                 batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
                 prioritized_replay=False,
                 critic_dist_info=critic_dist_info(args),
                 n_steps=args.n_steps, seed=args.seed)
    ewma = None
    iters = 0
    while float(global_count.item()) < stop_at:
        agent.actor.load_state_dict(global_model.actor.state_dict())
        agent.critic.load_state_dict(global_model.critic.state_dict())
        _, R, _ = rollout_episode(agent, env, noise=False, max_steps=500)
        ewma = R if ewma is None else 0.95 * ewma + 0.05 * R
        print(f"[eval] step {int(global_count.item())} return {R:.2f} "
              f"ewma {ewma:.2f}", flush=True)
        iters += 1
        if max_iters is not None and iters >= max_iters:
            return ewma
        time.sleep(period)
    return ewma
