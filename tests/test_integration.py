"""Integration tests: the worker schedule end-to-end on CPU, CLI parity,
HogWild multi-process mode, logging/plots."""

import os
import subprocess
import sys

import numpy as np
import pytest

from d4pg_amd.config import (configure_env_params, critic_dist_info,
                             make_parser, run_dir_name)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _args(extra=()):
    args = make_parser().parse_args(list(extra))
    configure_env_params(args)
    return args


def test_cli_flag_parity():
    """All 19 reference flags parse with reference defaults
    (/root/reference/main.py:31-56)."""
    args = _args([])
    assert args.n_workers == 4 and args.rmsize == 1000000
    assert args.tau == 0.001 and args.bsize == 64 and args.gamma == 0.99
    assert args.ou_theta == 0.15 and args.ou_sigma == 0.2 and args.ou_mu == 0.0
    assert args.env == "Pendulum-v1" and args.max_steps == 500
    assert args.n_eps == 20000 and args.warmup == 50 and args.p_replay == 1
    assert args.v_min == -300.0 and args.v_max == 0.0    # pendulum override
    assert args.n_atoms == 51 and args.multithread == 0 and args.n_steps == 1
    assert args.her == 0 and args.log_dir == "runs"
    d = critic_dist_info(args)
    assert d["type"] == "categorical" and d["n_atoms"] == 51


def test_run_dir_naming():
    args = _args(["--env", "Pendulum-v1", "--p_replay", "1", "--her", "1",
                  "--n_steps", "5", "--n_workers", "8"])
    assert run_dir_name(args) == \
        os.path.join("runs", "exp_Pendulum-v1_PER_HER_5N_8Workers")


def test_worker_cycle_end_to_end(tmp_path):
    """BASELINE config 1: Pendulum, 1 worker, a full (shrunk) cycle:
    warmup -> collect -> train -> eval -> scalars -> checkpoint."""
    from d4pg_amd.algo.d4pg import DDPG
    from d4pg_amd.envs import make, obs_act_dims
    from d4pg_amd.parallel.worker import Worker
    from d4pg_amd.utils.logging import SummaryWriter

    args = _args(["--warmup", "2", "--max_steps", "60",
                  "--log_dir", str(tmp_path), "--n_steps", "5"])
    args.episodes_per_cycle = 2
    args.train_steps_per_cycle = 4
    args.eval_trials = 2
    args.debug = 0
    env = make(args.env, seed=0)
    env._max_episode_steps = args.max_steps
    obs_dim, act_dim = obs_act_dims(env)
    agent = DDPG(obs_dim, act_dim, env=env, memory_size=args.rmsize,
                 batch_size=args.bsize, gamma=args.gamma, tau=args.tau,
                 prioritized_replay=True,
                 critic_dist_info=critic_dist_info(args),
                 n_steps=args.n_steps, seed=0)
    rd = run_dir_name(args)
    w = Worker("1", args, agent, env, writer=SummaryWriter(rd), run_dir=rd)
    w.work(max_cycles=2)
    assert agent.train_steps_done == 8
    assert os.path.exists(os.path.join(rd, "actor.pth"))
    assert os.path.exists(os.path.join(rd, "critic.pth"))
    assert os.path.exists(os.path.join(rd, "avg_test_reward.csv"))
    assert os.path.exists(os.path.join(rd, "success_rate.csv"))


def test_train_cli_smoke(tmp_path):
    """`python train.py` runs the reference CLI shape end-to-end."""
    cmd = [sys.executable, os.path.join(REPO, "train.py"),
           "--n_eps", "1", "--warmup", "1", "--max_steps", "30",
           "--log_dir", str(tmp_path), "--debug", "0", "--device", "cpu",
           "--backend", "eager"]
    env = dict(os.environ)
    env["D4PG_MAX_CYCLES"] = "1"
    # bound the run: patch via -c wrapper
    code = (
        "import sys; sys.argv = %r; "
        "import d4pg_amd.parallel.worker as W; "
        "orig = W.Worker.work; "
        "W.Worker.work = lambda self, *a, **k: orig(self, max_cycles=1); "
        "import runpy; runpy.run_path(%r, run_name='__main__')"
        % (cmd[1:], os.path.join(REPO, "train.py")))
    r = subprocess.run([sys.executable, "-c", code], cwd=REPO,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.timeout(300)
def test_hogwild_two_workers(tmp_path):
    """Shared-memory HogWild mode: 2 worker processes train against one
    global model; global step counter advances and weights change."""
    import torch
    from d4pg_amd.parallel.hogwild import run_hogwild

    args = _args(["--n_workers", "2", "--warmup", "1", "--max_steps", "30",
                  "--log_dir", str(tmp_path), "--debug", "0",
                  "--rmsize", "5000"])
    args.episodes_per_cycle = 1
    args.train_steps_per_cycle = 3
    args.eval_trials = 1
    global_model, count = run_hogwild(args, max_cycles=1,
                                      with_evaluator=False)
    assert count == 2 * 3          # both workers' steps hit the counter
    assert torch.isfinite(global_model.actor.fc1.weight).all()


def test_plots_from_csv(tmp_path):
    from d4pg_amd.plots import ewma_vectorized, plot_run
    csv = tmp_path / "avg_test_reward.csv"
    csv.write_text("step,value,walltime\n" + "\n".join(
        f"{i},{np.sin(i / 10)},0" for i in range(50)) + "\n")
    out = plot_run(str(tmp_path))
    assert len(out) == 1 and out[0].endswith(".png")
    sm = ewma_vectorized(np.ones(10), 5)
    np.testing.assert_allclose(sm, np.ones(10))


def test_logger_pickle_roundtrip(tmp_path):
    from d4pg_amd.utils.logging import Logger
    lg = Logger(str(tmp_path / "log.pkl"))
    lg.log("return", 1.5)
    lg.log("return", 2.5)
    lg.save()
    lg2 = Logger.load(str(tmp_path / "log.pkl"))
    assert [v for v, t in lg2.logs["return"]] == [1.5, 2.5]
