"""The driver depends on bench.py's exact output contract: one JSON line
from rank 0 with the BASELINE metric/config fields.  Pin it (CPU fallback
path; the GPU and multirank paths emit through the same code)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "8", "--warmup", "2"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [l for l in res.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line expected"
    out = json.loads(lines[0])
    # driver-contract fields (brief): metric/value/unit/n_gpus/steps/
    # warmup/ms_per_step/higher_is_better/scaling/vs_baseline/dtype/data/
    # config{model, global_batch, seq_len, parallelism}
    assert out["metric"].startswith("learner grad-steps/sec")
    assert out["unit"] == "grad_steps/s"
    assert out["n_gpus"] == 1 and out["steps"] == 8 and out["warmup"] == 2
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["vs_baseline"] is None          # reference publishes none
    assert out["dtype"] == "fp32"
    assert "synthetic" in out["data"]
    cfg = out["config"]
    assert cfg["global_batch"] == 64 and cfg["seq_len"] is None
    assert cfg["n_step"] == 5 and cfg["prioritized_replay"] is True
    assert cfg["replay_capacity"] == 1_000_000
    assert out["value"] > 0 and out["ms_per_step"] > 0
    # timing identity: value == n_gpus * steps / elapsed
    ident = out["n_gpus"] * 1000.0 / out["ms_per_step"]
    assert abs(out["value"] - ident) < 1e-6 * out["value"]
