#!/usr/bin/env python3
"""joinbench — BASELINE.json config 5: inner df.merge, left 1e9 rows ⋈
right 1e8 rows, int64 keys uniform in [0, 1e8).  Reports build/probe kernel
times (HIP events) and whole-op rows/s.
Run: python tools/joinbench.py [--left N] [--right M] [--steps K]
"""

import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import pandas
from modin_amd.core import lib
from modin_amd.core.dataframe import HipDataframe
from modin_amd.core.partition import DeviceBlock, HipDataframePartition
from modin_amd.query_compiler import HipQueryCompiler


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--left", type=int, default=1_000_000_000)
    ap.add_argument("--right", type=int, default=100_000_000)
    ap.add_argument("--steps", type=int, default=3)
    args = ap.parse_args()

    lib.ensure_ready(0)
    lib.profiling(True)
    rng = np.random.default_rng(42)
    keyspace = args.right  # keys uniform in [0, 1e8)

    def frame(n, cols):
        block = {"k": lib.put(rng.integers(0, keyspace, n).astype(np.int64))}
        for name in cols:
            block[name] = lib.put(rng.random(n))
        names = list(block)
        return HipQueryCompiler(HipDataframe(
            [HipDataframePartition(DeviceBlock(block, n))],
            pandas.RangeIndex(n), names, [n],
            pandas.Series({"k": np.dtype(np.int64),
                           **{c: np.dtype(np.float64) for c in cols}})))

    qL = frame(args.left, ["lv"])
    qR = frame(args.right, ["rv"])

    def step():
        out = qL.merge(qR, on="k")
        lib.sync()
        return out

    out = step()  # warm
    n_out = len(out)
    lib.kernel_stats_reset()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = step()
    lib.sync()
    dt = (time.perf_counter() - t0) / args.steps
    kernels = {}
    for k in ("join_hist", "join_scan", "join_fill", "join_fixup",
              "join_probe_count", "join_probe_emit", "gather"):
        nl, ms = lib.kernel_stats(k)
        if nl:
            kernels[k] = [nl, round(ms / args.steps, 3)]
    line = {
        "metric": "rows/sec inner-merge (left rows probed)",
        "left_rows": args.left,
        "right_rows": args.right,
        "out_rows": n_out,
        "ms_per_op": dt * 1e3,
        "value": args.left / dt,
        "kernel_ms_per_op": kernels,
    }
    print(json.dumps(line), flush=True)
    with open(os.path.join(REPO, "gpurun_out", "joinbench.json"), "w") as f:
        f.write(json.dumps(line) + "\n")


if __name__ == "__main__":
    os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
    main()
