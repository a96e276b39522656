#!/usr/bin/env python3
"""Print per-phase times of the persistent step kernel (100 MHz clock)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402

# stamp map after the round-2 quad restructure: fused regions stamp all
# their legacy indices back-to-back, so collapsed rows read ~0 us and the
# preceding row carries the whole block
PHASES = [
    "PH0 sample", "PH1 L1x4", "PH2 L2x3", "PH3 L3x3", "PH4 headsx3",
    "quadA ct+proj", "(fused)", "(fused)", "(fused)",
    "quadB c.dX", "(fused)", "(fused)", "c.dW", "c.adam",
    "quadC policy+tail", "(fused)", "(fused)", "(fused)", "(fused)",
    "(fused)", "(fused)", "(fused)", "(fused)", "(fused)",
    "(fused)", "a.dW", "a.adam", "(fused)",
]

eng = make_engine()
eng.step(50)          # warm; stamps overwritten each launch (s==0 only)
ts = eng.read("tstamp").numpy()
t0 = ts[0]
total = 0.0
for i, name in enumerate(PHASES):
    dt_us = (ts[i + 1] - ts[i]) / 100.0   # 100 MHz -> us
    total += dt_us
    print(f"{name:14s} {dt_us:8.2f} us")
print(f"{'TOTAL':14s} {total:8.2f} us (stamped step 0 incl. barriers)")
