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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--left", type=int, default=1_000_000_000)
    ap.add_argument("--right", type=int, default=100_000_000)
    ap.add_argument("--steps", type=int, default=3)
    args = ap.parse_args()

    import pandas
    from modin_amd.core import lib
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition
    from modin_amd.query_compiler import HipQueryCompiler

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
    print("v3 OK out_rows=", len(out))


if __name__ == "__main__":
    main()
