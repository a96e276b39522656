#!/usr/bin/env python3
"""Diagnose the IS-weight path: print bw, tree roots, beta inputs."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from tests.test_gpu_engine import make_engine, make_modules, random_transitions

eng = make_engine()
a, at, c, ct = make_modules()
eng.load_from_modules(a, at, c, ct)
tr = random_transitions(1024, seed=3)
eng.ingest(*[torch.from_numpy(x) for x in tr])
cap = eng.info()["tree_cap"]
st = eng.read("sum_tree").numpy()
mt = eng.read("min_tree").numpy()
print("after ingest: sum root", st[1], "min root", mt[1])
print("sum leaves[0:4]", st[cap:cap + 4], "min leaves[0:4]", mt[cap:cap + 4])
print("min leaves beyond size:", mt[cap + 1024:cap + 1028])
print("counters:", eng.counters())
eng.step(1)
bw = eng.read("bw").numpy()
print("bw min/max/mean:", bw.min(), bw.max(), bw.mean())
print("bw[:8]:", bw[:8])
st = eng.read("sum_tree").numpy()
mt = eng.read("min_tree").numpy()
print("after step: sum root", st[1], "min root", mt[1])
