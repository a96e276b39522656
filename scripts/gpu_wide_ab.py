#!/usr/bin/env python3
"""Wide-config A/B: uncaptured step() loop vs hipGraph-replayed steps."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402


def rate(fn, n):
    t0 = time.perf_counter()
    fn(n)
    return n / (time.perf_counter() - t0)


def main():
    eng = make_engine(batch=4096, hidden=1024, obs=17, act=6, cap=1000000)
    eng.step(5)
    r_eager = rate(lambda n: eng.step(n), 100)
    print(f"wide step() uncaptured : {r_eager:7.1f} steps/s")
    eng.train_steps(8, steps_per_graph=8)      # capture
    r_graph = rate(lambda n: eng.train_steps(n, steps_per_graph=8), 104)
    print(f"wide hipGraph replayed : {r_graph:7.1f} steps/s")


if __name__ == "__main__":
    main()
