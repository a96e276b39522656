#!/usr/bin/env python3
"""Tiny driver for rocprofv3: N uncaptured engine steps (flagship config)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.gpu_microbench import make_engine  # noqa: E402

n = int(sys.argv[1]) if len(sys.argv) > 1 else 200
eng = make_engine()
eng.step(10)      # warm
eng.step(n)
print("done", eng.counters()["adam_t_actor"], "steps")
