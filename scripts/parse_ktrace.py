#!/usr/bin/env python3
"""Parse a rocprofv3 kernel-trace CSV: per-(kernel, grid) dispatch stats.
Usage: parse_ktrace.py <csv path> [top_n]"""

import csv
import sys
from collections import defaultdict


def main():
    path, top = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 25
    rows = list(csv.DictReader(open(path)))
    if not rows:
        print("empty trace"); return
    cols = rows[0].keys()

    def col(*cands):
        for c in cols:
            lc = c.lower()
            if any(k in lc for k in cands):
                return c
        raise KeyError(cands)

    kname = col("kernel_name", "name")
    start = col("start")
    end = col("end")
    try:
        gx = col("grid_size_x", "grid_size", "grid_x")
    except KeyError:
        gx = None
    agg = defaultdict(lambda: [0, 0.0])
    for r in rows:
        name = r[kname].split("(")[0].split("::")[-1]
        grid = r.get(gx, "?") if gx else "?"
        dt = (float(r[end]) - float(r[start])) / 1000.0  # us
        a = agg[(name, grid)]
        a[0] += 1
        a[1] += dt
    items = sorted(agg.items(), key=lambda kv: -kv[1][1])
    print(f"{'kernel':28s} {'grid':>10s} {'calls':>6s} {'total_ms':>9s} "
          f"{'avg_us':>8s}")
    for (name, grid), (n, tot) in items[:top]:
        print(f"{name:28s} {str(grid):>10s} {n:6d} {tot/1000.0:9.3f} "
              f"{tot/n:8.2f}")


if __name__ == "__main__":
    main()
