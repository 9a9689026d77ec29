"""ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py header).

Numpy restatement of the reference's partition-operator semantics, each
function citing the reference code it follows.  No pandas in the compute
path of the oracle itself (pandas appears only in golden-vector generation,
which runs the real reference); numpy is the arithmetic substrate, matching
where the reference's own arithmetic lives (pandas/numpy inside each
partition — SURVEY.md §2 "Implementation language").
"""

from __future__ import annotations

import math

import numpy as np


# ---------------------------------------------------------------------------
# ingestion / chunking — modin/core/storage_formats/pandas/utils.py:28
# (compute_chunksize) + partition_manager.py:1029
# (split_pandas_df_into_partitions): chunk = ceil(n / num_splits), floored at
# MinRowPartitionSize; row slices [i*chunk, (i+1)*chunk).
# ---------------------------------------------------------------------------

def split_row_counts(n: int, num_splits: int, min_size: int = 32) -> list:
    chunk = max(math.ceil(n / num_splits) if num_splits else n, min_size)
    out = []
    start = 0
    while start < n:
        stop = min(start + chunk, n)
        out.append(stop - start)
        start = stop
    return out or [0]


# ---------------------------------------------------------------------------
# Map — modin/core/dataframe/algebra/map.py:28 (function applied blockwise;
# blockwise == global for elementwise ops).  Ops per
# query_compiler.py bindings: add/sub/mul/div scalars (:535 Binary scalar
# branch -> lazy map), fillna (:2710), abs (:2036).
# ---------------------------------------------------------------------------

def map_op(op: str, x: np.ndarray, scalar=None) -> np.ndarray:
    if op == "add":
        return x + scalar
    if op == "sub":
        return x - scalar
    if op == "rsub":
        return scalar - x
    if op == "mul":
        return x * scalar
    if op == "div":
        return x / scalar
    if op == "rdiv":
        return scalar / x
    if op == "fillna":
        y = x.astype(np.float64, copy=True)
        y[np.isnan(y)] = scalar
        return y
    if op == "abs":
        return np.abs(x)
    if op == "neg":
        return -x
    raise ValueError(op)


# ---------------------------------------------------------------------------
# Binary frame op — algebra/binary.py:420 frame branch -> n_ary_op
# (dataframe.py:3851): row-aligned elementwise zip.
# ---------------------------------------------------------------------------

def binary_op(op: str, a: np.ndarray, b: np.ndarray) -> np.ndarray:
    if op == "add":
        return a + b
    if op == "sub":
        return a - b
    if op == "mul":
        return a * b
    if op == "div":
        return a.astype(np.float64) / b.astype(np.float64)
    raise ValueError(op)


# ---------------------------------------------------------------------------
# TreeReduce — algebra/tree_reduce.py:63 -> dataframe.py:2208: per-partition
# map (pandas DataFrame.sum & friends: NaN-skipping), then axis reduce.
# pandas semantics restated:
#   sum:   nansum, 0.0 for all-NaN/empty (min_count=0)
#   count: non-NaN count
#   mean:  nansum/count, NaN if count==0
#   min/max: NaN-skipping, NaN if count==0
# int64 columns have no NaN; sum wraps like numpy int64.
# ---------------------------------------------------------------------------

def reduce_op(op: str, x: np.ndarray):
    if x.dtype == np.int64:
        if op == "sum":
            return np.add.reduce(x, dtype=np.int64) if x.size else np.int64(0)
        if op == "count":
            return int(x.size)
        if op == "mean":
            return float(np.add.reduce(x, dtype=np.float64) / x.size) if x.size \
                else float("nan")
        if op == "min":
            return np.min(x) if x.size else float("nan")
        if op == "max":
            return np.max(x) if x.size else float("nan")
        raise ValueError(op)
    valid = ~np.isnan(x)
    cnt = int(valid.sum())
    if op == "count":
        return cnt
    if op == "sum":
        return float(x[valid].sum()) if cnt else 0.0
    if op == "mean":
        return float(x[valid].sum() / cnt) if cnt else float("nan")
    if op == "min":
        return float(x[valid].min()) if cnt else float("nan")
    if op == "max":
        return float(x[valid].max()) if cnt else float("nan")
    raise ValueError(op)


# ---------------------------------------------------------------------------
# GroupByReduce — algebra/groupby.py:124 (map: per-partition
# groupby(by, as_index=True, observed=True).<agg>()) and :211 (reduce:
# concat partials, groupby(level=0).<agg'>()), with the map/reduce pairing
# of storage_formats/pandas/groupby.py:237-248:
#   sum  -> (sum, sum);  count -> (count, sum);
#   mean -> map {sum,count}, reduce sum then divide (:87-113).
# pandas result shape: ascending unique keys as the index; a key whose
# values are all NaN is PRESENT with sum 0.0 / count 0.
# Restated over dense int64 keys with np.bincount (the arithmetic is the
# same reassociated per-key summation).
# ---------------------------------------------------------------------------

def groupby_agg(keys: np.ndarray, vals: dict, agg: str):
    """Global groupby over int64 keys.  Returns (unique_keys, {name: agg})."""
    assert keys.dtype == np.int64
    if keys.size == 0:
        return np.empty(0, np.int64), {
            n: np.empty(0, np.int64 if agg == "count" else np.float64)
            for n in vals
        }
    kmin = keys.min()
    shifted = keys - kmin
    n_slots = int(shifted.max()) + 1
    rowcnt = np.bincount(shifted, minlength=n_slots)
    present = rowcnt > 0
    out_keys = (np.nonzero(present)[0] + kmin).astype(np.int64)
    out = {}
    for name, v in vals.items():
        v = np.asarray(v, dtype=np.float64)
        valid = ~np.isnan(v)
        sums = np.bincount(shifted[valid], weights=v[valid], minlength=n_slots)
        cnts = np.bincount(shifted[valid], minlength=n_slots)
        if agg == "sum":
            out[name] = sums[present]
        elif agg == "count":
            out[name] = cnts[present].astype(np.int64)
        elif agg == "mean":
            with np.errstate(invalid="ignore", divide="ignore"):
                out[name] = sums[present] / cnts[present]
        elif agg in ("min", "max"):
            acc = np.full(n_slots, np.inf if agg == "min" else -np.inf)
            ufunc = np.minimum if agg == "min" else np.maximum
            ufunc.at(acc, shifted[valid], v[valid])
            res = acc[present]
            res[cnts[present] == 0] = np.nan  # all-NaN group -> NaN
            out[name] = res
        else:
            raise ValueError(agg)
    return out_keys, out


# ---------------------------------------------------------------------------
# Compare + boolean filter — query_compiler comparison bindings and
# `df[mask]` (`PandasDataframe.filter`/`mask`, partition.py:224).
# pandas/numpy agree: NaN compares False for every op except != (True).
# ---------------------------------------------------------------------------

def compare_op(op: str, x: np.ndarray, s) -> np.ndarray:
    if op == "gt":
        return x > s
    if op == "ge":
        return x >= s
    if op == "lt":
        return x < s
    if op == "le":
        return x <= s
    if op == "eq":
        return x == s
    if op == "ne":
        return x != s
    raise ValueError(op)


def filter_rows(mask: np.ndarray, cols: dict):
    """Returns (kept_positions, filtered cols) preserving row order."""
    keep = np.asarray(mask).astype(bool)
    pos = np.nonzero(keep)[0].astype(np.int64)
    return pos, {n: np.asarray(v)[keep] for n, v in cols.items()}


# ---------------------------------------------------------------------------
# Sort — PandasDataframe.sort_by (dataframe.py:2742; stable order ==
# pandas sort_values(kind="stable"); descending keeps ties in original
# order, i.e. stable on the negated key).
# ---------------------------------------------------------------------------

def sort_perm(keys: np.ndarray, ascending: bool = True) -> np.ndarray:
    keys = np.asarray(keys)
    if keys.size == 0:
        return np.empty(0, np.int64)
    if ascending:
        return np.argsort(keys, kind="stable").astype(np.int64)
    return np.argsort(keys.max() - keys, kind="stable").astype(np.int64)


# ---------------------------------------------------------------------------
# Merge — MergeImpl.row_axis_merge (storage_formats/pandas/merge.py:104-178):
# materialize the right frame once (combine(), dataframe.py:2918), broadcast
# to every left partition, per-partition pandas.merge(how="inner").
# pandas inner-merge semantics restated: output rows ordered by left row,
# then right row within a key; result index is a fresh RangeIndex; key
# column keeps its left position; colliding value columns get _x/_y.
# ---------------------------------------------------------------------------

def inner_join(lk: np.ndarray, lvals: dict, rk: np.ndarray, rvals: dict):
    """Returns (keys, lidx, out_lvals, out_rvals) in pandas match order."""
    lk = np.asarray(lk)
    rk = np.asarray(rk)
    if rk.size == 0 or lk.size == 0:
        z = np.empty(0, np.int64)
        return (z, z, {n: np.asarray(v)[:0] for n, v in lvals.items()},
                {n: np.asarray(v)[:0] for n, v in rvals.items()})
    kmin = int(rk.min())
    slots = int(rk.max()) - kmin + 1
    cnt = np.bincount(rk - kmin, minlength=slots)
    csr = np.zeros(slots + 1, dtype=np.int64)
    np.cumsum(cnt, out=csr[1:])
    order = np.argsort(rk, kind="stable")  # right rows grouped, row-ordered
    lkk = lk - kmin
    inr = (lkk >= 0) & (lkk < slots)
    clipped = np.clip(lkk, 0, slots - 1)
    lcnt = np.where(inr, cnt[clipped], 0)
    total = int(lcnt.sum())
    lidx = np.repeat(np.arange(lk.size, dtype=np.int64), lcnt)
    starts = np.repeat(csr[clipped], lcnt)
    base = np.repeat(np.cumsum(lcnt) - lcnt, lcnt)
    rpos = starts + (np.arange(total, dtype=np.int64) - base)
    ridx = order[rpos]
    keys = lk[lidx]
    return (keys, lidx,
            {n: np.asarray(v)[lidx] for n, v in lvals.items()},
            {n: np.asarray(v)[ridx] for n, v in rvals.items()})


def partitioned_groupby_agg(keys: np.ndarray, vals: dict, agg: str,
                            num_splits: int, min_size: int = 32):
    """The two-phase form the reference actually runs (map per partition,
    reduce across partitions) — used to check that partitioning does not
    change the result beyond fp reassociation."""
    counts = split_row_counts(len(keys), num_splits, min_size)
    offs = np.cumsum([0] + counts)
    # map phase: per-partition partials.  sum & count cover sum/count/mean
    # (GroupbyReduceImpl pairs, storage_formats/pandas/groupby.py:237-248:
    # mean = map concat(sum,count) / reduce divide); min/max reduce with
    # themselves ({min: ("min","min")}) over NaN-skipping partials.
    partial = {}
    for i in range(len(counts)):
        sl = slice(offs[i], offs[i + 1])
        sub = {n: v[sl] for n, v in vals.items()}
        uk, psums = groupby_agg(keys[sl], sub, "sum")
        _, pcnts = groupby_agg(keys[sl], sub, "count")
        if agg in ("min", "max"):
            _, pext = groupby_agg(keys[sl], sub, agg)
        for j, k in enumerate(uk):
            acc = partial.setdefault(
                int(k), {n: [0.0, 0, float("nan")] for n in vals})
            for n in vals:
                acc[n][0] += psums[n][j]
                acc[n][1] += pcnts[n][j]
                if agg in ("min", "max"):
                    e = pext[n][j]
                    cur = acc[n][2]
                    if np.isnan(cur):
                        acc[n][2] = e
                    elif not np.isnan(e):
                        acc[n][2] = min(cur, e) if agg == "min" \
                            else max(cur, e)
    out_keys = np.array(sorted(partial.keys()), dtype=np.int64)
    out = {}
    for n in vals:
        if agg == "sum":
            out[n] = np.array([partial[int(k)][n][0] for k in out_keys])
        elif agg == "count":
            out[n] = np.array([partial[int(k)][n][1] for k in out_keys],
                              dtype=np.int64)
        elif agg in ("min", "max"):
            out[n] = np.array([partial[int(k)][n][2] for k in out_keys])
        else:
            out[n] = np.array([
                partial[int(k)][n][0] / partial[int(k)][n][1]
                if partial[int(k)][n][1] else float("nan")
                for k in out_keys
            ])
    return out_keys, out


def shuffle_dest(keys, splitters):
    """Restates csrc k_shuffle_dest (the RangePartitioning bin rule,
    reference partition_manager.py:1937 shuffle_partitions / range-partition
    sampling): dest[i] = #{j : splitters[j] <= key[i]}.  Depends only on the
    key value, so equal keys always land on one destination."""
    keys = np.asarray(keys, dtype=np.int64)
    spl = np.asarray(splitters, dtype=np.int64)
    return np.searchsorted(spl, keys, side="right").astype(np.int64)


def pick_splitters(samples, world):
    """Restates distributed.sample_splitters: sorted global sample ->
    world-1 quantile splitters (deterministic; identical on every rank)."""
    s = np.sort(np.asarray(samples, dtype=np.int64))
    if s.size == 0 or world <= 1:
        return np.empty(0, dtype=np.int64)
    qs = [(i * s.size) // world for i in range(1, world)]
    return s[qs]


# ---- deterministic RNG mirrors of the device fills (hf_fill_rand*) --------
# splitmix64 finalizer over (seed + index), bit-exact vs hipframe.hip's
# rng_mix64 — the verify gate regenerates device-born bench frames here.

def _mix64(x: np.ndarray) -> np.ndarray:
    with np.errstate(over="ignore"):
        x = (x + np.uint64(0x9E3779B97F4A7C15))
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        return x ^ (x >> np.uint64(31))


def _rand_bits(seed: int, n: int, offset: int = 0) -> np.ndarray:
    idx = np.arange(offset, offset + n, dtype=np.uint64)
    with np.errstate(over="ignore"):
        return _mix64(np.uint64(seed % (1 << 64)) + idx)


def rand_int(seed: int, n: int, lo: int, hi: int,
             offset: int = 0) -> np.ndarray:
    """Mirror of hf_fill_randint: lo + splitmix64(seed+i) % (hi-lo)."""
    return (lo + (_rand_bits(seed, n, offset) %
                  np.uint64(hi - lo)).astype(np.int64))


def rand_f64(seed: int, n: int, offset: int = 0) -> np.ndarray:
    """Mirror of hf_fill_randf64: (bits >> 11) * 2^-53 in [0,1)."""
    return ((_rand_bits(seed, n, offset) >> np.uint64(11)).astype(np.float64)
            * (1.0 / 9007199254740992.0))


def rand_cdf(seed: int, n: int, cdf: np.ndarray,
             offset: int = 0) -> np.ndarray:
    """Mirror of hf_fill_randcdf: searchsorted(cdf, U[0,1), side='right')."""
    u = rand_f64(seed, n, offset)
    return np.searchsorted(cdf, u, side="right").astype(np.int64)


def zipf_cdf(n_keys: int, s: float) -> np.ndarray:
    """CDF of zipf(s) over ranks 1..n_keys (key k has p ~ 1/(k+1)^s) —
    BASELINE §8d's skew variant; shared by bench.py and the device draw."""
    w = 1.0 / np.power(np.arange(1, n_keys + 1, dtype=np.float64), s)
    return np.cumsum(w) / w.sum()
