#!/usr/bin/env python3
"""bench.py — north-star benchmark: groupby-sum rows/s (BASELINE.json).

Workload (config.workload = "groupby-sum-1e9-int64key-f64val"): one "step" is
one full mpd.DataFrame.groupby('k').sum() pass (the L1 modin.pandas path)
over a synthetic frame of `--rows` rows (default 1e9) with int64 keys uniform
in [0, 1e6) and fp64 values, generated ON DEVICE (hf_fill_rand*) so inputs
are born resident in HBM.  --dist zipf1.2 draws keys from zipf(1.2) instead
(BASELINE §8d skew variant).  Other BASELINE.json configs are parity-test
cases, not bench lines.

Usage:
  python bench.py [--gpus N] [--steps K] [--warmup W] [--rows R] [--keys K]
N>1 is launched by the driver as one rank per GPU via torch.distributed.run;
total rows stay fixed (the metric is quoted on 1e9 rows at 1/2/4/8 GPUs →
"scaling": "strong"); the dense table all-reduce over RCCL/xGMI is the
exchange step.

Output: ONE JSON line from rank 0.  `roofline` reports the WHOLE-OP
fraction (16 algorithmic B/row ÷ ms_per_step ÷ 8 TB/s spec) with the
per-kernel HIP-event breakdown inside; `cold_ms_per_step` is the first
groupby after frame build (per-key-column one-time work included: histogram,
u32 key copy, key-range reduce — cached on the immutable column afterwards,
the device analog of the reference's lazy metadata caches).  `cpu_baseline`
times plain pandas groupby().sum() on this box's host cores (BASELINE.md
§CPU-baseline), with the numpy oracle as the secondary figure.
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
SEED = 42


def shard_seeds(rank: int):
    """Per-rank independent RNG streams for the key and value draws."""
    return SEED + 2 * rank * 0x9E3779B9, SEED + (2 * rank + 1) * 0x9E3779B9


def build_frame(rows: int, keys: int, rank: int, world: int,
                parts_per_rank: int, dist: str):
    """Device-resident synthetic frame, generated on device (hf_fill_rand*)."""
    import oracle
    from modin_amd.core import lib
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition

    shard_counts = oracle.split_row_counts(rows, world, 1)
    while len(shard_counts) < world:
        shard_counts.append(0)
    local_n = shard_counts[rank]
    kseed, vseed = shard_seeds(rank)
    chunk_counts = oracle.split_row_counts(local_n, parts_per_rank, 1)
    cdf_col = None
    if dist.startswith("zipf"):
        cdf_col = lib.put(oracle.zipf_cdf(keys, float(dist[4:])))
    partitions = []
    off = 0
    for cn in chunk_counts:
        if cdf_col is not None:
            # draw indices offset by the shard position so chunks tile one
            # stream (the oracle mirror regenerates with the same offsets)
            k = lib.fill_randcdf(cn, (kseed + off) % (1 << 64), cdf_col)
        else:
            k = lib.fill_randint(cn, (kseed + off) % (1 << 64), 0, keys)
        v = lib.fill_randf64(cn, (vseed + off) % (1 << 64))
        block = DeviceBlock({"k": k, "v": v}, cn)
        partitions.append(HipDataframePartition(block))
        off += cn
    import pandas
    frame = HipDataframe(partitions, pandas.RangeIndex(local_n), ["k", "v"],
                         chunk_counts,
                         pandas.Series({"k": np.dtype(np.int64),
                                        "v": np.dtype(np.float64)}))
    return frame, local_n, chunk_counts


def expected_tables(local_n, keys, rank, chunk_counts, dist):
    """Host regeneration of the device draws (oracle RNG mirrors) reduced to
    the expected (group keys, sums) — chunked to bound host RAM."""
    import oracle
    kseed, vseed = shard_seeds(rank)
    acc = np.zeros(keys, dtype=np.float64)
    seen = np.zeros(keys, dtype=bool)
    cdf = oracle.zipf_cdf(keys, float(dist[4:])) if dist.startswith("zipf") \
        else None
    off = 0
    for cn in chunk_counts:
        for a in range(0, cn, 50_000_000):
            m = min(50_000_000, cn - a)
            if cdf is not None:
                kk = oracle.rand_cdf((kseed + off) % (1 << 64), m, cdf, a)
            else:
                kk = oracle.rand_int((kseed + off) % (1 << 64), m, 0, keys, a)
            vv = oracle.rand_f64((vseed + off) % (1 << 64), m, a)
            acc += np.bincount(kk, weights=vv, minlength=keys)
            seen[kk] = True
        off += cn
    exp_keys = np.nonzero(seen)[0]
    return exp_keys, acc[exp_keys]


def cpu_baseline_leg(keys_card: int, unit: str, dist: str):
    """plain pandas groupby().sum() on this box's
    host cores, on a bounded sample (~10-30 s of CPU work); the numpy oracle
    restatement is kept as the secondary figure."""
    import oracle
    import pandas as pd
    rng = np.random.default_rng(SEED)
    sample = 20_000_000
    if dist.startswith("zipf"):
        cdf = oracle.zipf_cdf(keys_card, float(dist[4:]))
        k = oracle.rand_cdf(SEED, sample, cdf)
    else:
        k = rng.integers(0, keys_card, sample).astype(np.int64)
    v = rng.random(sample)
    df = pd.DataFrame({"k": k, "v": v})
    df.iloc[:100_000].groupby("k").sum()  # warm
    done_rows = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 10.0:
        df.groupby("k").sum()
        done_rows += sample
    dt = time.perf_counter() - t0
    # secondary: the numpy oracle restatement (single core)
    t1 = time.perf_counter()
    passes = 0
    while time.perf_counter() - t1 < 5.0:
        oracle.groupby_agg(k, {"v": v}, "sum")
        passes += 1
    dt_o = time.perf_counter() - t1
    return {
        "value": done_rows / dt,
        "unit": unit,
        "cores": 1,   # pandas groupby is single-threaded; box has os.cpu_count() cores
        "cores_available": os.cpu_count(),
        "kind": "pandas",
        "sample": f"pandas {pd.__version__} groupby-sum, {sample} rows/pass x "
                  f"{done_rows // sample} passes in {dt:.1f}s",
        "note": "Modin-on-CPU leg unavailable on the bench box (no modin "
                "install there; BASELINE.md caveat) — plain pandas is the "
                "reference backend's own per-partition engine",
        "oracle_secondary": {
            "value": passes * sample / dt_o,
            "kind": "port",
            "cores": 1,
        },
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows", type=int, default=1_000_000_000)
    ap.add_argument("--keys", type=int, default=1_000_000)
    ap.add_argument("--parts", type=int, default=1,
                    help="partitions per rank (scatter launches per step)")
    ap.add_argument("--dist", default="uniform",
                    help="key distribution: uniform | zipf1.2")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-verify", action="store_true",
                    help="skip the oracle parity gate before timing")
    args = ap.parse_args()

    import modin_amd.distributed as dmod
    from modin_amd.core import lib
    from modin_amd.query_compiler import HipQueryCompiler
    import modin_amd.pandas as mpd

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    is_dist = dmod.init_from_env()
    import torch
    lib.ensure_ready(int(os.environ.get("MODIN_AMD_GPU", "0")))
    lib.profiling(True)

    t_build0 = time.perf_counter()
    frame, local_n, chunk_counts = build_frame(
        args.rows, args.keys, rank, world, args.parts, args.dist)
    lib.sync()
    t_build = time.perf_counter() - t_build0
    df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))

    def step():
        out = df.groupby("k").sum()
        lib.sync()
        return out

    # cold step: per-key-column one-time work (histogram, u32 key copy,
    # key-range reduce) runs here and is cached on the immutable column
    t_cold0 = time.perf_counter()
    step()
    cold_ms = (time.perf_counter() - t_cold0) * 1e3

    for _ in range(max(args.warmup - 1, 0)):
        step()

    if world == 1 and not args.no_verify:
        # parity gate outside the timed region: the full result must match
        # the oracle regeneration of the same device draws
        res = step()._query_compiler._modin_frame
        got_keys = res.index.to_numpy()
        got_sums = res.to_pandas()["v"].to_numpy()
        exp_keys, exp_sums = expected_tables(local_n, args.keys, 0,
                                             chunk_counts, args.dist)
        np.testing.assert_array_equal(got_keys, exp_keys)
        np.testing.assert_allclose(got_sums, exp_sums, rtol=1e-12, atol=1e-9)
        print(f"# verify ok: {exp_keys.size} groups match the oracle",
              file=sys.stderr)

    lib.kernel_stats_reset()
    if is_dist:
        dmod.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
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

    ms_per_step = elapsed / args.steps * 1e3
    # whole-op roofline (the honest figure — VERDICT r01 weak #1): the
    # algorithmic 16 B/row over the whole step, vs the 8 TB/s spec peak.
    # Per-kernel HIP-event times inside for the breakdown; `traffic` is the
    # PMC-measured whole-step HBM traffic (profiles/r02, x2 FETCH correction
    # per MI355X_MICROARCH.md §HBM) — env override wins.
    stats = {name: lib.kernel_stats(name)
             for name in ("gb_scatter", "gb_bucket_agg", "gb_dense",
                          "gb_accum", "gb_hist", "gb_compact_scatter",
                          "gb_keys32")}
    alg_bytes = ALG_BYTES_PER_ROW * local_n
    achieved = alg_bytes / (ms_per_step / 1e3)
    traffic_env = os.environ.get("HF_TRAFFIC_BYTES_PER_STEP")
    roofline = {
        "bound": "hbm",
        "scope": "whole-op",
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
            cpu = cpu_baseline_leg(args.keys, "rows/s", args.dist)
        line = {
            "metric": "rows/sec groupby-sum",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world if is_dist else 1,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "cold_ms_per_step": cold_ms,
            "build_s": round(t_build, 3),
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
                "key_distribution": args.dist,
                "key_dtype": "int64",
                "val_dtype": "float64",
                "partitions_per_rank": args.parts,
                "api_layer": "modin_amd.pandas DataFrame.groupby('k').sum()",
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
