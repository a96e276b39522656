"""SummaryWriter CSV shim, pickle Logger, Meter, run-dir naming."""

import csv
import time

from d4pg_amd.config import configure_env_params, make_parser, run_dir_name
from d4pg_amd.utils.logging import Logger, Meter, SummaryWriter


def test_summary_writer_csv(tmp_path):
    w = SummaryWriter(str(tmp_path / "run"))
    w.add_scalar("avg_test_reward", -150.5, 10)
    w.add_scalar("avg_test_reward", -120.25, 20)
    w.add_scalar("success_rate", 0.5, 10)
    w.close()
    with open(tmp_path / "run" / "avg_test_reward.csv") as f:
        rows = list(csv.DictReader(f))
    assert [float(r["value"]) for r in rows] == [-150.5, -120.25]
    assert [int(r["step"]) for r in rows] == [10, 20]
    assert (tmp_path / "run" / "success_rate.csv").exists()


def test_logger_roundtrip(tmp_path):
    lf = str(tmp_path / "log.pkl")
    lg = Logger(lf)
    lg.log("return", -100.0)
    lg.log("return", -90.0)
    lg.log("loss", 1.5)
    lg.save()
    lg2 = Logger.load(lf)
    assert [v for v, _ in lg2.logs["return"]] == [-100.0, -90.0]
    assert len(lg2.logs["loss"]) == 1
    # timestamps monotonic
    ts = [t for _, t in lg2.logs["return"]]
    assert ts[0] <= ts[1]


def test_meter_rate():
    m = Meter()
    m.add(100)
    time.sleep(0.05)
    r = m.rate()
    assert 0 < r < 100 / 0.05 + 1
    m.reset()
    assert m.count == 0


def test_run_dir_name_encodes_config():
    args = make_parser().parse_args(
        ["--env", "Pendulum-v1", "--p_replay", "1", "--her", "0",
         "--n_steps", "5", "--n_workers", "4"])
    configure_env_params(args)
    rd = run_dir_name(args)
    # the reference's run-dir convention encodes env/PER/n-step/workers
    # (main.py:59-64)
    assert "Pendulum" in rd
    assert "5N" in rd or "5" in rd


def test_d4pg_config_from_args_roundtrip():
    from d4pg_amd.config import D4PGConfig
    args = make_parser().parse_args(["--env", "Pendulum-v1", "--bsize",
                                     "128", "--lr_critic", "5e-4"])
    cfg = D4PGConfig.from_args(args)
    assert cfg.bsize == 128
    assert cfg.lr_critic == 5e-4
    assert cfg.env == "Pendulum-v1"
    assert cfg.extra == {}          # every CLI flag has a dataclass field
