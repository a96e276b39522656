#!/usr/bin/env python3
"""Print per-phase times of the persistent step kernel (100 MHz clock)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402

PHASES = [
    "PH0 sample", "PH1 L1x4", "PH2 L2x3", "PH3 L3x3", "PH4 headsx3",
    "PH5 ct.L2", "PH6 ct.L3", "PH7 ct.L4sm", "PH8 proj+ce",
    "PH9 c.dX4", "PH10 c.dX3", "PH11 c.dX2", "PH12 c.dW", "PH13 c.adam",
    "PH14 pc.L1", "PH15 pc.L2", "PH16 pc.L3", "PH17 pc.L4sm", "PH18 pgrad",
    "PH19 p.dX4", "PH20 p.dX3", "PH21 p.dX2a", "PH22 a.dX4", "PH23 a.dX3",
    "PH24 a.dX2", "PH25 a.dW", "PH26 a.adam+nx", "PH27 (fused)",
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
