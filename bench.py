#!/usr/bin/env python3
"""bench.py — north-star benchmark: groupby-sum rows/s (BASELINE.json).

Workload (config.workload = "groupby-sum-1e9-int64key-f64val"): one "step" is
one full df.groupby('k').agg('sum') pass over a synthetic frame of
`--rows` rows (default 1e9) with int64 keys uniform in [0, 1e6) and fp64
values, inputs resident in HBM when the timed region starts.  Other
BASELINE.json configs are parity-test cases, not bench lines.

Usage:
  python bench.py [--gpus N] [--steps K] [--warmup W] [--rows R] [--keys K]
N>1 is launched by the driver as one rank per GPU via torch.distributed.run;
total rows stay fixed (the metric is quoted on 1e9 rows at 1/2/4/8 GPUs →
"scaling": "strong"); the dense table all-reduce over RCCL/xGMI is the
exchange step.

Output: ONE JSON line from rank 0 with the whole-job aggregate, a
`roofline` object for the dominant kernel (gb_accum) timed with HIP events
on the hipframe stream, and a `cpu_baseline` object (the numpy oracle timed
on this box's host cores, rank 0, N=1 only).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

HBM_PEAK = 8.0e12          # B/s, MI355X spec peak (MI355X_MICROARCH.md)
ALG_BYTES_PER_ROW = 16     # 8 B key + 8 B value read per row


def build_frame(rows: int, keys: int, rank: int, world: int, parts_per_rank: int):
    """Device-resident synthetic frame, built shard-by-shard to bound host RAM."""
    import oracle
    from modin_amd.core import lib
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition

    shard_counts = oracle.split_row_counts(rows, world, 1)
    while len(shard_counts) < world:
        shard_counts.append(0)
    local_n = shard_counts[rank]
    rng = np.random.default_rng([42, rank])
    chunk_counts = oracle.split_row_counts(local_n, parts_per_rank, 1)
    partitions = []
    for cn in chunk_counts:
        k = rng.integers(0, keys, cn).astype(np.int64)
        v = rng.random(cn)
        block = DeviceBlock({"k": lib.put(k), "v": lib.put(v)}, cn)
        partitions.append(HipDataframePartition(block))
        del k, v
    import pandas
    frame = HipDataframe(partitions, pandas.RangeIndex(local_n), ["k", "v"],
                         chunk_counts,
                         pandas.Series({"k": np.dtype(np.int64),
                                        "v": np.dtype(np.float64)}))
    return frame, local_n


def cpu_baseline_leg(keys_card: int, unit: str):
    """Time the oracle (numpy restatement, single-threaded bincount) on a
    bounded sample of the same workload: ~10-30 s of CPU work."""
    import oracle
    rng = np.random.default_rng(42)
    sample = 20_000_000
    k = rng.integers(0, keys_card, sample).astype(np.int64)
    v = rng.random(sample)
    # warm once
    oracle.groupby_agg(k[:100_000], {"v": v[:100_000]}, "sum")
    done_rows = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 10.0:
        oracle.groupby_agg(k, {"v": v}, "sum")
        done_rows += sample
    dt = time.perf_counter() - t0
    return {
        "value": done_rows / dt,
        "unit": unit,
        "cores": 1,
        "kind": "port",
        "sample": f"numpy-oracle groupby-sum, {sample} rows/pass x "
                  f"{done_rows // sample} passes in {dt:.1f}s",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows", type=int, default=1_000_000_000)
    ap.add_argument("--keys", type=int, default=1_000_000)
    ap.add_argument("--parts", type=int, default=1,
                    help="partitions per rank (gb_accum launches per step)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-verify", action="store_true",
                    help="skip the oracle parity gate before timing")
    args = ap.parse_args()

    import modin_amd.distributed as dmod
    from modin_amd.core import lib
    from modin_amd.query_compiler import HipQueryCompiler

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    is_dist = dmod.init_from_env()
    import torch
    if is_dist:
        device = dmod._state["device"]
    else:
        device = "cuda:0"
    lib.ensure_ready(int(os.environ.get("MODIN_AMD_GPU", "0")))
    lib.profiling(True)

    frame, local_n = build_frame(args.rows, args.keys, rank, world, args.parts)
    qc = HipQueryCompiler(frame)

    def step():
        out = HipQueryCompiler.groupby_sum(qc, "k")
        lib.sync()
        return out

    for _ in range(args.warmup):
        step()

    if world == 1 and not args.no_verify:
        # parity gate outside the timed region: the step's full 1e9-row
        # result must match the numpy oracle (regenerated from the same
        # seeds chunk-by-chunk to bound host RAM)
        import oracle
        res = step()._modin_frame
        got_keys = res.index.to_numpy()
        got_sums = res.to_pandas()["v"].to_numpy()
        # regenerate with EXACTLY build_frame's draw order (same chunking)
        vrng = np.random.default_rng([42, 0])
        acc = np.zeros(args.keys, dtype=np.float64)
        seen = np.zeros(args.keys, dtype=bool)
        for cn in oracle.split_row_counts(local_n, max(args.parts, 1), 1):
            kk = vrng.integers(0, args.keys, cn).astype(np.int64)
            vv = vrng.random(cn)
            acc += np.bincount(kk, weights=vv, minlength=args.keys)
            seen[kk] = True
        exp_keys = np.nonzero(seen)[0]
        np.testing.assert_array_equal(got_keys, exp_keys)
        np.testing.assert_allclose(got_sums, acc[exp_keys], rtol=1e-12,
                                   atol=1e-9)
        print(f"# verify ok: {exp_keys.size} groups match the oracle",
              file=sys.stderr)

    lib.kernel_stats_reset()
    if is_dist:
        dmod.barrier()
    torch.cuda.synchronize() if torch.cuda.is_available() else None
    lib.sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = step()
    lib.sync()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if is_dist:
        dmod.barrier()
    elapsed = time.perf_counter() - t0
    if is_dist:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=dmod._state["device"])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # dominant kernel of the groupby path this run (radix: gb_scatter;
    # dense: gb_dense; fallback: gb_accum)
    stats = {name: lib.kernel_stats(name)
             for name in ("gb_scatter", "gb_bucket_agg", "gb_dense",
                          "gb_accum", "gb_hist", "gb_compact_scatter")}
    dom = max(stats, key=lambda k: stats[k][1])
    n_launch, total_ms = stats[dom]
    rows_per_launch = local_n * args.steps / n_launch if n_launch else 0
    avg_ms = total_ms / n_launch if n_launch else float("nan")
    achieved = (ALG_BYTES_PER_ROW * rows_per_launch) / (avg_ms / 1e3) \
        if n_launch and avg_ms > 0 else 0.0
    # HBM traffic per launch of the dominant kernel, from rocprofv3 PMC
    # (profiles/r01c/pmc_traffic.txt: FETCH_SIZE x2-corrected + WRITE_SIZE =
    # 27.0 B/row for gb_scatter — reads exactly the algorithmic 16 B/row,
    # writes 1.1x of the 10 B/row payload from chunk-boundary partial
    # lines).  Env override wins.
    PMC_TRAFFIC_B_PER_ROW = {"gb_scatter": 27.0}
    traffic_env = os.environ.get("HF_TRAFFIC_BYTES_PER_LAUNCH")
    if not traffic_env and dom in PMC_TRAFFIC_B_PER_ROW:
        traffic_env = PMC_TRAFFIC_B_PER_ROW[dom] * rows_per_launch
    roofline = {
        "bound": "hbm",
        "kernel": dom,
        "achieved": achieved / 1e9,          # GB/s (algorithmic 16 B/row)
        "peak": HBM_PEAK / 1e9,
        "unit": "GB/s",
        "frac": achieved / HBM_PEAK,
        "traffic": float(traffic_env) if traffic_env else None,
        "kernel_ms": {k: [v[0], round(v[1], 3)] for k, v in stats.items()
                      if v[0]},
    }

    value = args.rows * args.steps / elapsed if elapsed > 0 else 0.0

    if rank == 0:
        cpu = None
        if world == 1 and not args.no_cpu_baseline:
            cpu = cpu_baseline_leg(args.keys, "rows/s")
        line = {
            "metric": "rows/sec groupby-sum",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world if is_dist else 1,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": "groupby-sum-1e9-int64key-f64val"
                if args.rows == 1_000_000_000 else
                f"groupby-sum-{args.rows}-int64key-f64val",
                "rows": args.rows,
                "key_cardinality": args.keys,
                "key_dtype": "int64",
                "val_dtype": "float64",
                "partitions_per_rank": args.parts,
                "parallelism": f"dp{world if is_dist else 1}",
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(line), flush=True)
    if is_dist:
        dmod.shutdown()


if __name__ == "__main__":
    main()
