"""Metrics & logging.

The reference logs tensorboard scalars ``avg_test_reward`` and
``success_rate`` (/root/reference/main.py:66, 352-353) plus a pickle Logger
(plotUtil.ipynb cell 0).  tensorboard is not importable in this image, so
``SummaryWriter`` here is a drop-in shim with the same ``add_scalar`` API
writing CSV (one file per tag: step,value,walltime) — which plots/plots.py
consumes — and delegates to the real torch SummaryWriter when available.

``Logger`` reproduces the notebook's pickle append-log (name -> list of
(value, walltime)) used for offline comparison plots.
"""

from __future__ import annotations

import os
import pickle
import time


class SummaryWriter:
    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._files = {}
        self._tb = None
        try:
            from torch.utils.tensorboard import SummaryWriter as TBWriter
            self._tb = TBWriter(log_dir)
        except Exception:
            self._tb = None

    def _file(self, tag: str):
        f = self._files.get(tag)
        if f is None:
            path = os.path.join(self.log_dir,
                                tag.replace("/", "_") + ".csv")
            new = not os.path.exists(path)
            f = open(path, "a", buffering=1)
            if new:
                f.write("step,value,walltime\n")
            self._files[tag] = f
        return f

    def add_scalar(self, tag: str, value, global_step=None, walltime=None):
        walltime = walltime if walltime is not None else time.time()
        step = 0 if global_step is None else global_step
        self._file(tag).write(f"{step},{float(value)},{walltime}\n")
        if self._tb is not None:
            self._tb.add_scalar(tag, value, global_step=global_step,
                                walltime=walltime)

    def flush(self):
        for f in self._files.values():
            f.flush()
        if self._tb is not None:
            self._tb.flush()

    def close(self):
        for f in self._files.values():
            f.close()
        self._files.clear()
        if self._tb is not None:
            self._tb.close()


class Logger:
    """Pickle append-log with wall-clock timestamps (plotUtil.ipynb parity)."""

    def __init__(self, logfile: str):
        self.logfile = logfile
        self.logs = {}
        self.t0 = time.time()

    def log(self, name: str, value) -> None:
        self.logs.setdefault(name, []).append((value, time.time() - self.t0))

    def save(self) -> None:
        with open(self.logfile, "wb") as f:
            pickle.dump(self.logs, f)

    @classmethod
    def load(cls, logfile: str) -> "Logger":
        lg = cls(logfile)
        with open(logfile, "rb") as f:
            lg.logs = pickle.load(f)
        return lg


class Meter:
    """Throughput meter: windowed rate of a monotonically-increasing count
    (grad-steps/sec, env-steps/sec — the BASELINE.md metrics)."""

    def __init__(self):
        self.t0 = time.time()
        self.count = 0

    def add(self, n: int = 1):
        self.count += n

    def rate(self) -> float:
        dt = time.time() - self.t0
        return self.count / dt if dt > 0 else 0.0

    def reset(self):
        self.t0 = time.time()
        self.count = 0
