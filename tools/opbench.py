#!/usr/bin/env python3
"""opbench — measured throughput for the non-groupby hot ops
(BASELINE.json configs 2 and 3): Map (df+1 / fillna), Binary (df+df),
TreeReduce (df.sum / df.mean).  One JSON line per op with HIP-event kernel
timing and the algorithmic-bytes roofline fraction.
Run on an MI355X box:  python tools/opbench.py [--cols C] [--rows N]
"""

import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

HBM_PEAK = 8.0e12


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--map-rows", type=int, default=100_000_000)
    ap.add_argument("--map-cols", type=int, default=8)
    ap.add_argument("--red-rows", type=int, default=1_000_000_000)
    ap.add_argument("--red-cols", type=int, default=4)
    ap.add_argument("--steps", type=int, default=5)
    args = ap.parse_args()

    from modin_amd.core import lib
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition
    from modin_amd.query_compiler import HipQueryCompiler
    import pandas

    lib.ensure_ready(0)
    lib.profiling(True)
    rng = np.random.default_rng(42)

    def frame(rows, cols, nan_frac=0.0):
        block = {}
        for c in range(cols):
            v = rng.random(rows)
            if nan_frac:
                v[rng.random(rows) < nan_frac] = np.nan
            block[f"c{c}"] = lib.put(v)
            del v
        names = list(block)
        hf = HipDataframe([HipDataframePartition(DeviceBlock(block, rows))],
                          pandas.RangeIndex(rows), names, [rows],
                          pandas.Series({n: np.dtype(np.float64) for n in names}))
        return HipQueryCompiler(hf)

    results = []

    def report(op, kernel, elems, alg_bytes_per_elem, fn):
        fn()  # warm
        lib.sync()
        lib.kernel_stats_reset()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            fn()
        lib.sync()
        dt = (time.perf_counter() - t0) / args.steps
        agg = {}
        for k in (kernel if isinstance(kernel, list) else [kernel]):
            nl, ms = lib.kernel_stats(k)
            if nl:
                agg[k] = (nl, ms)
        total_ms = sum(ms for _, ms in agg.values()) / args.steps
        alg = elems * alg_bytes_per_elem
        line = {
            "op": op,
            "elems": elems,
            "ms_per_op": dt * 1e3,
            "kernel_ms_per_op": total_ms,
            "alg_GBps": alg / (total_ms / 1e3) / 1e9 if total_ms else None,
            "roofline_frac": alg / (total_ms / 1e3) / HBM_PEAK if total_ms else None,
            "kernels": {k: [v[0], round(v[1], 3)] for k, v in agg.items()},
        }
        results.append(line)
        print(json.dumps(line), flush=True)

    # config 2: Map — 1e8 x 8 f64
    qc = frame(args.map_rows, args.map_cols, nan_frac=0.01)
    ne = args.map_rows * args.map_cols
    report("map_add_scalar (df+1)", "map_f64", ne, 16,
           lambda: HipQueryCompiler.add(qc, 1.0)._modin_frame._partitions[0]
           .drain_call_queue())
    report("map_fillna (df.fillna(0))", "map_f64", ne, 16,
           lambda: HipQueryCompiler.fillna(qc, 0.0)._modin_frame._partitions[0]
           .drain_call_queue())
    qc2 = frame(args.map_rows, args.map_cols)  # distinct frame: honest 24 B/elem
    report("binary_add (df+df2)", "bin_f64", ne, 24,
           lambda: HipQueryCompiler.add(qc, qc2))
    del qc, qc2

    # config 3: TreeReduce — 1e9 x 4 f64 (fresh columns each op; reduce
    # results are cached per immutable column, so bypass the cache by
    # clearing it)
    qc = frame(args.red_rows, args.red_cols)
    block = qc._modin_frame._partitions[0].block()

    def clear_cache():
        for col in block.columns.values():
            col._reduce_cache = None

    ne = args.red_rows * args.red_cols
    def do_sum():
        clear_cache()
        HipQueryCompiler.sum(qc)
    report("tree_reduce_sum (df.sum)", "reduce_f64", ne, 8, do_sum)

    # sort_values: 1e9 rows, 1e6-key range (3 radix passes + gathers)
    del qc, block
    import oracle  # noqa: F401  (only to mirror bench imports)
    rng2 = np.random.default_rng(43)
    n = args.red_rows
    kcol = lib.put(rng2.integers(0, 10**6, n).astype(np.int64))
    lib.reduce(kcol)  # key-range metadata (cached)
    lib.sync()
    lib.kernel_stats_reset()
    t0 = time.perf_counter()
    for _ in range(3):
        perm = lib.sort_perm(kcol, True)
    lib.sync()
    dt = (time.perf_counter() - t0) / 3
    ks = {}
    for k in ("sort_pack", "sort_pass", "sort_unpack"):
        nl, ms = lib.kernel_stats(k)
        if nl:
            ks[k] = [nl, round(ms / 3, 3)]
    line = {"op": "sort_perm (1e9 rows, 1e6-key range)", "elems": n,
            "ms_per_op": dt * 1e3, "rows_per_s": n / dt, "kernels": ks}
    results.append(line)
    print(json.dumps(line), flush=True)

    with open(os.path.join(REPO, "gpurun_out", "opbench.json"), "w") as f:
        for line in results:
            f.write(json.dumps(line) + "\n")


if __name__ == "__main__":
    os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
    main()
