#!/usr/bin/env python3
"""rocprofv3 driver: N steps of the wide-batch config (B=4096, H=1024)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402

n = int(sys.argv[1]) if len(sys.argv) > 1 else 30
eng = make_engine(batch=4096, hidden=1024, obs=17, act=6, cap=1000000)
eng.step(3)
eng.step(n)
print("done", eng.counters()["adam_t_actor"], "steps")
