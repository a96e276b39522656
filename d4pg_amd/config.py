"""CLI & config layer.

Flag-name/default parity with the reference CLI (/root/reference/main.py:31-56):
every one of its 19 flags is accepted with the same name, type and default, so
reference command lines run unmodified.  MI355X-native flags (device placement,
distribution, broadcast cadence, kernel backend) are added on top.

Also carries the per-env value-range override table
(/root/reference/main.py:84-99 `configure_env_params`) and the run-directory
naming convention (/root/reference/main.py:59-64).
"""

from __future__ import annotations

import argparse
import os
from dataclasses import dataclass, field


def make_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="MI355X-native D4PG")

    # --- reference-parity flags (main.py:31-56) ---
    p.add_argument("--n_workers", default=4, type=int,
                   help="number of parallel actor workers")
    p.add_argument("--rmsize", default=1000000, type=int,
                   help="replay memory size")
    p.add_argument("--tau", default=0.001, type=float,
                   help="target-network soft-update rate")
    p.add_argument("--ou_theta", default=0.15, type=float, help="OU noise theta")
    p.add_argument("--ou_sigma", default=0.2, type=float, help="OU noise sigma")
    p.add_argument("--ou_mu", default=0.0, type=float, help="OU noise mu")
    p.add_argument("--bsize", default=64, type=int, help="minibatch size")
    p.add_argument("--gamma", default=0.99, type=float, help="discount factor")
    p.add_argument("--env", default="Pendulum-v1", type=str, help="environment id")
    p.add_argument("--max_steps", default=500, type=int,
                   help="max episode length override")
    p.add_argument("--n_eps", default=20000, type=int, help="number of epochs")
    p.add_argument("--debug", default=1, type=int, help="print debug output")
    p.add_argument("--warmup", default=50, type=int,
                   help="warmup episodes filled before training")
    p.add_argument("--p_replay", default=1, type=int,
                   help="1 = prioritized replay, 0 = uniform")
    p.add_argument("--v_min", default=-100.0, type=float,
                   help="C51 distribution support minimum")
    p.add_argument("--v_max", default=100.0, type=float,
                   help="C51 distribution support maximum")
    p.add_argument("--n_atoms", default=51, type=int, help="C51 atom count")
    p.add_argument("--multithread", default=0, type=int,
                   help="1 = spawn parallel workers + evaluator")
    p.add_argument("--n_steps", default=1, type=int, help="n-step return horizon")
    p.add_argument("--logfile", default="", type=str, help="log file name")
    p.add_argument("--log_dir", default="runs", type=str, help="log directory")
    p.add_argument("--her", default=0, type=int,
                   help="1 = hindsight experience replay")

    # --- MI355X-native flags ---
    p.add_argument("--device", default="auto", type=str,
                   help="'cuda' (=ROCm/HIP), 'cpu', or 'auto'")
    p.add_argument("--backend", default="auto", type=str,
                   choices=["auto", "hip", "eager"],
                   help="learner compute backend: hand-written HIP kernels or "
                        "eager torch (tests/CPU fallback)")
    p.add_argument("--replay_device", default="auto", type=str,
                   help="where the replay lives: 'cuda' = on-HBM sum tree, 'cpu'")
    p.add_argument("--broadcast_interval", default=1, type=int,
                   help="learner steps between parameter broadcasts to actors")
    p.add_argument("--train_steps_per_cycle", default=40, type=int,
                   help="gradient steps per cycle (reference main.py:303)")
    p.add_argument("--episodes_per_cycle", default=16, type=int,
                   help="episodes collected per cycle (reference main.py:299)")
    p.add_argument("--cycles_per_epoch", default=50, type=int,
                   help="cycles per epoch (reference main.py:299)")
    p.add_argument("--eval_trials", default=10, type=int,
                   help="greedy eval rollouts per cycle (reference main.py:309)")
    p.add_argument("--seed", default=0, type=int, help="RNG seed")
    p.add_argument("--lr_actor", default=1e-4, type=float,
                   help="actor Adam lr (reference local Adam 1e-4, "
                        "ddpg.py:67)")
    p.add_argument("--lr_critic", default=1e-3, type=float,
                   help="critic Adam lr (reference global Adam 1e-3, "
                        "main.py:384)")
    p.add_argument("--noise", default="gaussian", type=str,
                   choices=["gaussian", "ou"],
                   help="exploration noise process.  The reference ships "
                        "--ou_theta/sigma/mu flags but never constructs the "
                        "OU process (ddpg.py:74-75 commented out); here "
                        "'ou' wires them up")
    p.add_argument("--noise_eps", default=0.3, type=float,
                   help="Gaussian noise scale epsilon (reference "
                        "random_process.py:7 hardcodes 0.3)")
    p.add_argument("--vector_envs", default=0, type=int,
                   help="MI355X extension: batch M vectorized envs per "
                        "actor rank (one [M,obs] policy forward per tick) "
                        "instead of one env per process; 0 = scalar parity "
                        "mode (distributed learner only)")
    p.add_argument("--gpu_actors", default=-1, type=int,
                   help="device-resident rollout for vector actor ranks "
                        "(env dynamics + policy + noise + n-step fold as "
                        "HIP kernels): 1 = force, 0 = off, -1 = auto "
                        "(on when the rank sees a GPU)")
    return p


# Per-env C51 value-range overrides, mirroring the semantics of
# /root/reference/main.py:84-99 (Pendulum gets [-300, 0]); the rest use the
# flag defaults.  Entries are (v_min, v_max).
ENV_VALUE_RANGES = {
    "Pendulum-v0": (-300.0, 0.0),
    "Pendulum-v1": (-300.0, 0.0),
}


def configure_env_params(args) -> None:
    """Apply per-env v_min/v_max overrides in place (main.py:84-99 parity)."""
    rng = ENV_VALUE_RANGES.get(args.env)
    if rng is not None:
        args.v_min, args.v_max = rng


def noise_kwargs(args) -> dict:
    """DDPG ctor kwargs selecting/parameterizing the exploration noise
    from the CLI flags (--noise, --noise_eps, --ou_theta/sigma/mu)."""
    return {
        "noise": getattr(args, "noise", "gaussian"),
        "noise_eps": getattr(args, "noise_eps", 0.3),
        "ou_theta": getattr(args, "ou_theta", 0.15),
        "ou_sigma": getattr(args, "ou_sigma", 0.2),
        "ou_mu": getattr(args, "ou_mu", 0.0),
    }


def critic_dist_info(args) -> dict:
    """The derived distributional-critic config dict (main.py:373-376 parity)."""
    return {
        "type": "categorical",
        "v_min": args.v_min,
        "v_max": args.v_max,
        "n_atoms": args.n_atoms,
    }


def run_dir_name(args) -> str:
    """Run-directory naming convention (main.py:59-64 parity):
    encodes env, PER, HER, n-step and worker count."""
    name = "exp_" + args.env
    if args.p_replay:
        name += "_PER"
    if args.her:
        name += "_HER"
    name += "_%dN_%dWorkers" % (args.n_steps, args.n_workers)
    return os.path.join(args.log_dir, name)


@dataclass
class D4PGConfig:
    """Programmatic config (keyword equivalent of the CLI namespace)."""
    n_workers: int = 4
    rmsize: int = 1000000
    tau: float = 0.001
    ou_theta: float = 0.15
    ou_sigma: float = 0.2
    ou_mu: float = 0.0
    bsize: int = 64
    gamma: float = 0.99
    env: str = "Pendulum-v1"
    max_steps: int = 500
    n_eps: int = 20000
    debug: int = 1
    warmup: int = 50
    p_replay: int = 1
    v_min: float = -100.0
    v_max: float = 100.0
    n_atoms: int = 51
    multithread: int = 0
    n_steps: int = 1
    logfile: str = ""
    log_dir: str = "runs"
    her: int = 0
    device: str = "auto"
    backend: str = "auto"
    replay_device: str = "auto"
    broadcast_interval: int = 1
    train_steps_per_cycle: int = 40
    episodes_per_cycle: int = 16
    cycles_per_epoch: int = 50
    eval_trials: int = 10
    seed: int = 0
    lr_actor: float = 1e-4
    lr_critic: float = 1e-3
    noise: str = "gaussian"
    noise_eps: float = 0.3
    vector_envs: int = 0
    gpu_actors: int = -1
    extra: dict = field(default_factory=dict)

    @classmethod
    def from_args(cls, args) -> "D4PGConfig":
        known = {f for f in cls.__dataclass_fields__ if f != "extra"}
        kw, extra = {}, {}
        for k, v in vars(args).items():
            (kw if k in known else extra)[k] = v
        return cls(extra=extra, **kw)
