"""End-to-end GPU training through the PRODUCT path (Worker + DDPG with
backend='hip' + GPUReplayAdapter + fused engine), not just the engine API:
collect real Pendulum episodes, train on-device, evaluate, checkpoint."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_worker_cycle_on_gpu(tmp_path):
    from d4pg_amd.algo.d4pg import DDPG
    from d4pg_amd.config import configure_env_params, make_parser
    from d4pg_amd.envs import make, obs_act_dims
    from d4pg_amd.parallel.worker import Worker

    args = make_parser().parse_args(
        ["--env", "Pendulum-v1", "--max_steps", "50", "--warmup", "2",
         "--rmsize", "20000", "--bsize", "64", "--n_steps", "5",
         "--n_eps", "1", "--cycles_per_epoch", "2", "--debug", "0",
         "--episodes_per_cycle", "2", "--train_steps_per_cycle", "10",
         "--eval_trials", "2", "--seed", "0"])
    configure_env_params(args)
    env = make(args.env, seed=0)
    env._max_episode_steps = args.max_steps
    obs_dim, act_dim = obs_act_dims(env)
    agent = DDPG(obs_dim, act_dim, env=env, memory_size=args.rmsize,
                 batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
                 prioritized_replay=True,
                 critic_dist_info={"type": "categorical", "v_min": -300.0,
                                   "v_max": 0.0, "n_atoms": 51},
                 n_steps=args.n_steps, device="cuda", backend="hip", seed=0)
    w = Worker("t", args, agent, env, run_dir=str(tmp_path))
    before = torch.cat([p.detach().reshape(-1).cpu()
                        for p in agent.actor.parameters()]).clone()
    w.work(max_cycles=2)
    # params must have moved on-device and synced back for eval/save
    agent.engine.sync_params_if_dirty()
    after = torch.cat([p.detach().reshape(-1).cpu()
                       for p in agent.actor.parameters()])
    assert not torch.allclose(before, after)
    assert agent.train_steps_done == 20
    assert len(agent.replayBuffer) > 0
    # checkpoint written in reference format
    assert os.path.exists(tmp_path / "actor.pth")
    st = torch.load(tmp_path / "actor.pth", weights_only=True)
    assert set(st.keys()) == {"fc1.weight", "fc1.bias", "fc2.weight",
                              "fc2.bias", "fc2_2.weight", "fc2_2.bias",
                              "fc3.weight", "fc3.bias"}
    # device counters line up with the python-side step count
    cnt = agent.engine.engine.counters()
    assert cnt["adam_t_actor"] == 20
    assert np.isfinite(cnt["loss_critic"])


def test_distributed_learner_train_marks_params_dirty():
    """Regression: the distributed learner's _train must leave the engine
    bridge params-dirty so the next broadcast ships TRAINED weights (a
    direct engine.train_steps call once bypassed the flag and silently
    broadcast the initial actor forever)."""
    from d4pg_amd.config import configure_env_params, make_parser
    from d4pg_amd.ops import pack_net
    from d4pg_amd.parallel.learner import DistributedD4PG

    args = make_parser().parse_args(
        ["--env", "Pendulum-v1", "--max_steps", "50", "--warmup", "0",
         "--rmsize", "10000", "--bsize", "64", "--n_steps", "5",
         "--debug", "0", "--train_steps_per_cycle", "5", "--seed", "1"])
    configure_env_params(args)
    node = DistributedD4PG(args, rank=0, world=1, device="cuda",
                           with_evaluator=False)
    rng = np.random.default_rng(0)
    for _ in range(500):
        node.agent.replayBuffer.add(
            rng.standard_normal(3).astype("f"),
            rng.uniform(-1, 1, 1).astype("f"), -rng.random(),
            rng.standard_normal(3).astype("f"), 0.0)
    before = pack_net(node.agent.actor).clone()
    node._train()
    assert node.agent.engine is not None
    node.agent.engine.sync_params_if_dirty()
    after = pack_net(node.agent.actor)
    assert not torch.equal(before, after), \
        "broadcast would ship stale params"
