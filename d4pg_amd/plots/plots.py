"""Offline plotting (capability parity with /root/reference/plots/plots.py
and plotUtil.ipynb): read scalar CSV logs (step,value,walltime — the format
utils/logging.py writes, superset of the reference's step,avg,curr CSVs),
EWMA-smooth, and emit one PNG per env/run.  matplotlib is imported lazily so
headless training never needs it."""

from __future__ import annotations

import csv
import glob
import os

import numpy as np


def ewma_vectorized(data: np.ndarray, window: int) -> np.ndarray:
    """Exponentially-weighted moving average, alpha = 2/(window+1)
    (same smoothing the reference's numpy_ewma_vectorized_v2 computes,
    plots/plots.py:6-22), implemented as a stable recursive filter."""
    data = np.asarray(data, dtype=np.float64)
    if data.size == 0:
        return data
    alpha = 2.0 / (window + 1.0)
    out = np.empty_like(data)
    out[0] = data[0]
    for i in range(1, data.size):
        out[i] = alpha * data[i] + (1 - alpha) * out[i - 1]
    return out


def read_scalar_csv(path: str):
    steps, values = [], []
    with open(path) as f:
        r = csv.reader(f)
        header = next(r, None)
        for row in r:
            if len(row) >= 2:
                steps.append(float(row[0]))
                values.append(float(row[1]))
    return np.asarray(steps), np.asarray(values)


def plot_run(run_dir: str, window: int = 20, out_dir: str | None = None):
    """One PNG per scalar CSV in ``run_dir`` (smoothed + raw)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    out_dir = out_dir or run_dir
    os.makedirs(out_dir, exist_ok=True)
    made = []
    for path in sorted(glob.glob(os.path.join(run_dir, "*.csv"))):
        tag = os.path.splitext(os.path.basename(path))[0]
        steps, values = read_scalar_csv(path)
        if steps.size == 0:
            continue
        plt.figure(figsize=(8, 5))
        plt.plot(steps, values, alpha=0.3, label=tag)
        plt.plot(steps, ewma_vectorized(values, window),
                 label=f"{tag} (ewma{window})")
        plt.xlabel("step")
        plt.ylabel(tag)
        plt.legend()
        plt.grid(alpha=0.3)
        out = os.path.join(out_dir, tag + ".png")
        plt.savefig(out, dpi=110, bbox_inches="tight")
        plt.close()
        made.append(out)
    return made


def main(argv=None):
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("run_dirs", nargs="+")
    p.add_argument("--window", type=int, default=20)
    args = p.parse_args(argv)
    for rd in args.run_dirs:
        for f in plot_run(rd, window=args.window):
            print("wrote", f)


if __name__ == "__main__":
    main()
