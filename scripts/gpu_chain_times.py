#!/usr/bin/env python3
"""Per-phase times of the chain-fused step kernel (stamps 32..42)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402

PHASES = [
    "C0 sample", "C1 fwd-chains x3", "C2 ct+proj+ce", "C3 c.dX chain",
    "C4 c.dW", "C5 c.adam", "C6 policy megachain (+tree)", "C7 a.dW",
    "C8 a.adam", "C9 tick",
]

eng = make_engine()
eng.step(50)
ts = eng.read("tstamp").numpy()[32:]
total = 0.0
for i, name in enumerate(PHASES):
    dt_us = (ts[i + 1] - ts[i]) / 100.0
    total += dt_us
    print(f"{name:28s} {dt_us:8.2f} us")
print(f"{'TOTAL':28s} {total:8.2f} us")
