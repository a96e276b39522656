"""Round-2 hardening tests: ADVICE r1 fixes + VERDICT r1 weak items.

Covers (CPU-only):
  * _exchange push_cap truncation: advertised count is clamped to what
    _encode shipped, and the drop is counted (ADVICE low #2).
  * --noise selector: DDPG constructs the OU process from the ou_* flags
    (the reference parses but never uses them, ddpg.py:74-75).
  * load_state_dict replay-format guard: a CPU agent fed a GPU-format
    replay blob fails loudly instead of KeyError (ADVICE medium #1).
"""

import types

import numpy as np
import pytest
import torch

from d4pg_amd.algo.d4pg import DDPG
from d4pg_amd.config import make_parser, noise_kwargs
from d4pg_amd.noise import GaussianNoise, OrnsteinUhlenbeckProcess


def test_exchange_clamps_count_to_encoded(capsys):
    """A rank whose collection exceeds push_cap must advertise the
    truncated count (what the wire block actually carries), not len(lb)."""
    import torch.distributed as dist
    from d4pg_amd.parallel.learner import DistributedD4PG, _ListBuffer
    store = dist.TCPStore("127.0.0.1", 29931, 1, True)
    dist.init_process_group("gloo", store=store, rank=0, world_size=1)
    try:
        fake = types.SimpleNamespace(
            comm_device=torch.device("cpu"), push_cap=8, obs_dim=3,
            act_dim=1, world=1, rank=0, is_learner=True,
            dropped_transitions=0, agent=None)
        lb = _ListBuffer()
        rng = np.random.default_rng(0)
        for _ in range(13):                      # 13 > push_cap=8
            lb.add(rng.standard_normal(3), rng.standard_normal(1),
                   rng.random(), rng.standard_normal(3), 0.0)
        DistributedD4PG._exchange(fake, lb)
        assert fake.dropped_transitions == 5
        assert int(fake._last_counts[0].item()) == 8   # clamped, not 13
        assert "truncated" in capsys.readouterr().out
    finally:
        dist.destroy_process_group()


def test_noise_selector_ou():
    kw = noise_kwargs(make_parser().parse_args(
        ["--noise", "ou", "--ou_theta", "0.5", "--ou_sigma", "0.7",
         "--ou_mu", "0.1"]))
    agent = DDPG(3, 1, memory_size=100, prioritized_replay=False, seed=0,
                 **kw)
    assert isinstance(agent.noise, OrnsteinUhlenbeckProcess)
    assert agent.noise.theta == 0.5
    assert agent.noise.sigma == 0.7
    assert agent.noise.mu == 0.1
    a = agent.select_action(np.zeros(3), explore=True)
    assert a.shape == (1,) and np.all(np.abs(a) <= 1.0)


def test_noise_selector_gaussian_eps():
    kw = noise_kwargs(make_parser().parse_args(["--noise_eps", "0.05"]))
    agent = DDPG(3, 1, memory_size=100, prioritized_replay=False, seed=0,
                 **kw)
    assert isinstance(agent.noise, GaussianNoise)
    assert agent.noise.eps == 0.05


def test_batched_ou_shape():
    """OU over a [M, act] batch (the vector-actor noise path)."""
    ou = OrnsteinUhlenbeckProcess((16, 2), rng=np.random.default_rng(3))
    s = ou.sample()
    assert s.shape == (16, 2)
    samples = np.stack([ou.sample() for _ in range(500)])
    # OU mean-reverts to mu=0; long-run mean near 0, nonzero variance
    assert abs(samples.mean()) < 0.2
    assert samples.std() > 0.01


def test_load_state_dict_replay_format_guard():
    agent = DDPG(3, 1, memory_size=100, prioritized_replay=True, seed=0)
    rng = np.random.default_rng(0)
    for _ in range(70):
        agent.replayBuffer.add(rng.standard_normal(3).astype("f"),
                               rng.standard_normal(1).astype("f"),
                               0.1, rng.standard_normal(3).astype("f"), 0.0)
    agent.train()
    st = agent.state_dict()
    # simulate a GPU-engine checkpoint's replay blob landing on a CPU agent
    st["replay"] = {"sum_tree": torch.zeros(4), "min_tree": torch.zeros(4),
                    "s": torch.zeros(1, 3), "size": 1}
    fresh = DDPG(3, 1, memory_size=100, prioritized_replay=True, seed=0)
    with pytest.raises(ValueError, match="replay format"):
        fresh.load_state_dict(st)
    # load_replay=False skips the replay blob and succeeds
    fresh.load_state_dict(st, load_replay=False)
    assert fresh.train_steps_done == agent.train_steps_done


def test_config_dataclass_roundtrip():
    """D4PGConfig.from_args keeps every CLI flag (known fields + extras)."""
    from d4pg_amd.config import D4PGConfig, make_parser
    args = make_parser().parse_args(
        ["--noise", "ou", "--vector_envs", "16", "--gpu_actors", "1"])
    cfg = D4PGConfig.from_args(args)
    assert cfg.noise == "ou" and cfg.vector_envs == 16
    assert cfg.gpu_actors == 1
    # no flag silently dropped
    known = set(D4PGConfig.__dataclass_fields__) - {"extra"}
    assert set(vars(args)) <= known | set(cfg.extra)


def test_ingest_chunk_constant_matches_engine():
    """FusedEngine.INGEST_CHUNK must match engine.hip's ing_cap."""
    import re
    from d4pg_amd.ops import FusedEngine
    src = open("d4pg_amd/ops/hip/engine.hip").read()
    m = re.search(r"ing_cap = (\d+);", src)
    assert m and int(m.group(1)) == FusedEngine.INGEST_CHUNK
