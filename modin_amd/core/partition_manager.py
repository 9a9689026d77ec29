"""Partition manager: classmethods over the partition list.

Mirrors ``modin/core/dataframe/pandas/partitioning/partition_manager.py``:
``map_partitions`` (:708), ``lazy_map_partitions`` (:773),
``groupby_reduce`` (:303 — map phase per partition, reduce phase across),
``from_pandas``/``split_pandas_df_into_partitions`` (:1070/:1029) with the
reference's chunking rule (``compute_chunksize``,
modin/core/storage_formats/pandas/utils.py:28), and the BenchmarkMode
barrier (:52-92, here a hipStreamSynchronize).

MI355X redesign of GroupByReduce: the reference's map phase
runs pandas ``groupby().sum()`` per partition and its reduce phase concats
partials and groupbys again (algebra/groupby.py:124/:211).  Here the map
phase accumulates every partition into ONE dense key-indexed device table
(hardware f64 atomics), the cross-GPU reduce is an RCCL all-reduce of that
table (distributed.py), and the final "reduce" is an on-device compaction to
ascending present keys — no concat copy, no second hash pass.
"""

from __future__ import annotations

import math

import pandas

from .. import config
from ..distributed import maybe_allreduce_table
from . import lib
from .partition import HipDataframePartition


def compute_chunksize(n: int, num_splits: int, min_size: int) -> int:
    """Reference: modin/core/storage_formats/pandas/utils.py:28."""
    chunk = math.ceil(n / num_splits) if num_splits else n
    return max(chunk, min_size)


def wait_if_benchmark_mode(fn):
    """Reference: partition_manager.py:52-92 decorator."""
    def wrapper(*args, **kwargs):
        out = fn(*args, **kwargs)
        if config.BenchmarkMode.get():
            lib.sync()
        return out
    return wrapper


class GroupbyTable:
    """Dense key-indexed aggregation table (device buffers).

    Layout matches hf_groupby_accum: sums f64[nvals][n_slots],
    rowcnt u64[n_slots], counts u64[nvals][n_slots] (optional).
    Buffers may be library-allocated (single process) or torch-allocated
    (multi-GPU, so RCCL can all-reduce them in place) — see distributed.py.
    """

    def __init__(self, nvals: int, key_min: int, n_slots: int,
                 want_counts: bool, agg_op: int = 0):
        self.nvals = nvals
        self.key_min = key_min
        self.n_slots = n_slots
        self.want_counts = want_counts
        self.agg_op = agg_op
        init = lib.AGG_IDENTITY[agg_op]
        self._torch_tensors = None
        from ..distributed import is_active, alloc_table_torch
        if is_active():
            self._torch_tensors, self.sums, self.rowcnt, self.counts = \
                alloc_table_torch(nvals, n_slots, want_counts, init)
        else:
            self.sums = lib.alloc_raw(8 * nvals * n_slots)
            self.rowcnt = lib.alloc_raw(8 * n_slots)
            self.counts = lib.alloc_raw(8 * nvals * n_slots) if want_counts else 0
            if init == 0.0:
                lib.memset_raw(self.sums, 0, 8 * nvals * n_slots)
            else:
                lib.fill_f64(self.sums, init, nvals * n_slots)
            lib.memset_raw(self.rowcnt, 0, 8 * n_slots)
            if want_counts:
                lib.memset_raw(self.counts, 0, 8 * nvals * n_slots)

    def free(self):
        if self._torch_tensors is not None:
            self._torch_tensors = None
        else:
            lib.free_raw(self.sums)
            lib.free_raw(self.rowcnt)
            if self.counts:
                lib.free_raw(self.counts)
        self.sums = self.rowcnt = self.counts = 0


class HipDataframePartitionManager:
    """Classmethod namespace over lists of HipDataframePartition (p×1 grid)."""

    _partition_class = HipDataframePartition

    # ---- ingestion (partition_manager.py:1070) ----
    @classmethod
    def from_pandas(cls, df: pandas.DataFrame, num_splits=None, cats=None):
        if num_splits is None:
            num_splits = config.NPartitions.get()
        n = len(df)
        chunk = compute_chunksize(n, num_splits, config.MinRowPartitionSize.get())
        parts = []
        row_lengths = []
        start = 0
        while start < n:
            stop = min(start + chunk, n)
            parts.append(cls._partition_class.put(df.iloc[start:stop], cats))
            row_lengths.append(stop - start)
            start = stop
        if not parts:  # empty frame: keep one empty partition
            parts = [cls._partition_class.put(df, cats)]
            row_lengths = [0]
        return parts, row_lengths

    @classmethod
    def to_pandas(cls, parts, index=None, columns=None):
        frames = [p.get() for p in parts]
        out = pandas.concat(frames, ignore_index=True) if len(frames) > 1 else frames[0]
        if index is not None:
            out.index = index
        if columns is not None:
            out = out[list(columns)]
        return out

    # ---- Map (partition_manager.py:708/:773) ----
    @classmethod
    @wait_if_benchmark_mode
    def map_partitions(cls, parts, func):
        return [p.apply(func) for p in parts]

    @classmethod
    def lazy_map_partitions(cls, parts, func):
        return [p.add_to_apply_calls(func) for p in parts]

    # ---- Binary / n-ary zip (dataframe.py:3851 n_ary_op device form) ----
    @classmethod
    @wait_if_benchmark_mode
    def binary_partitions(cls, left, right, func):
        if len(left) != len(right):
            raise lib.HfError(
                "n_ary_op requires co-partitioned frames (same row splits) — "
                "repartition first"
            )
        out = []
        for lp, rp in zip(left, right):
            lp.drain_call_queue()
            rp.drain_call_queue()
            out.append(HipDataframePartition(func(lp.block(), rp.block())))
        return out

    # ---- TreeReduce (partition_manager.py map+map_axis_partitions shape,
    #      dataframe.py:2244-2247) ----
    @classmethod
    @wait_if_benchmark_mode
    def reduce_partitions(cls, parts, col_names):
        """Per-partition single-pass partials, combined on host.

        Returns dict name -> dict(sum, count, mn, mx, isum, imn, imx) of the
        across-partition combination (the reference's reduce phase over a
        p×1 grid is a p-way combine of 1-row partials).
        """
        partials = {name: [] for name in col_names}
        for p in parts:
            block = p.block()
            for name in col_names:
                partials[name].append(lib.reduce(block.columns[name]))
        out = {}
        for name, rs in partials.items():
            tot_cnt = sum(r.count for r in rs)
            have = [r for r in rs if r.count > 0]
            dtype = None
            out[name] = {
                "sum": float(sum(r.sum for r in have)) if have else 0.0,
                "isum": int(sum(r.isum for r in have)) if have else 0,
                "count": int(tot_cnt),
                "mn": min((r.mn for r in have), default=float("nan")),
                "mx": max((r.mx for r in have), default=float("nan")),
                "imn": min((r.imn for r in have), default=0),
                "imx": max((r.imx for r in have), default=0),
            }
        return out

    # ---- GroupByReduce (partition_manager.py:303) ----
    @classmethod
    @wait_if_benchmark_mode
    def groupby_reduce(cls, parts, by_name, val_names, want_counts,
                       agg_op=0):
        """Dense-table groupby: returns (keys_col, sum_cols, count_cols, n).

        Map phase (algebra/groupby.py:124 device form): hf_groupby_accum per
        partition into one shared table.  Reduce phase (:211 device form):
        RCCL all-reduce across ranks (if distributed) + on-device compaction.
        """
        # key range via the i64 reduce kernel
        kmin, kmax, total_rows = None, None, 0
        key_cols = []
        raw_cols_per_part = []
        for p in parts:
            block = p.block()
            kcol = block.columns[by_name]
            if kcol.dtype_code != lib.HF_INT64:
                raise lib.HfError(
                    f"groupby key column {by_name!r} must be int64 (dense-key "
                    "path); hashed/float keys are a later round"
                )
            key_cols.append(kcol)
            raw_cols_per_part.append([block.columns[v] for v in val_names])
            if kcol.length:
                r = lib.reduce(kcol)
                kmin = r.imn if kmin is None else min(kmin, r.imn)
                kmax = r.imx if kmax is None else max(kmax, r.imx)
            total_rows += kcol.length
        # int64 value columns: the f64 accumulators are exact below 2^53.
        # For MIN/MAX a global per-column BIAS (subtract the column min)
        # keeps huge-magnitude ints (datetime64 ns views) exact as long as
        # the RANGE fits 2^53 — the caller adds the bias back int-side.
        biases = [0] * len(val_names)
        from ..distributed import is_active
        for ci in range(len(val_names)):
            imn = imx = None
            is_int = False
            for raw in raw_cols_per_part:
                c = raw[ci]
                if c.dtype_code != lib.HF_INT64:
                    continue
                is_int = True
                if c.length:
                    r = lib.reduce(c)
                    imn = r.imn if imn is None else min(imn, r.imn)
                    imx = r.imx if imx is None else max(imx, r.imx)
            if not is_int:
                continue
            if is_active():
                imn, imx = maybe_allreduce_keyrange(imn, imx)
            if imn is None:
                continue
            if max(abs(imn), abs(imx)) < 1 << 53:
                continue
            if agg_op in (lib.AGG_MIN, lib.AGG_MAX) and \
                    imx - imn < 1 << 53:
                biases[ci] = imn
                continue
            raise lib.HfError(
                f"groupby over int64 column {val_names[ci]!r} with values "
                "beyond 2^53: f64 accumulation would round — exact int "
                "accumulation covers min/max ranges below 2^53 only")
        val_cols_per_part = []
        for raw in raw_cols_per_part:
            vals = []
            for ci, c in enumerate(raw):
                if biases[ci] and c.length:
                    c = lib.map_scalar(lib.MAP_SUB, c, biases[ci])
                vals.append(lib.cast_f64(c))
            val_cols_per_part.append(vals)

        kmin, kmax = maybe_allreduce_keyrange(kmin, kmax)
        if kmin is None:  # globally empty
            kmin, kmax = 0, -1
        n_slots = kmax - kmin + 1
        if n_slots < 0:
            n_slots = 0
        if n_slots > config.MaxGroupbySlots.get():
            if is_active():
                k4, s4, c4, n4 = cls._groupby_shuffle(
                    key_cols, val_cols_per_part, want_counts, agg_op)
            else:
                k4, s4, c4, n4 = cls._groupby_hash(
                    key_cols, val_cols_per_part, total_rows, want_counts,
                    agg_op)
            return k4, s4, c4, n4, biases
        n_slots = max(n_slots, 1)
        table = GroupbyTable(len(val_names), kmin, n_slots, want_counts,
                             agg_op)
        for kcol, vals in zip(key_cols, val_cols_per_part):
            if kcol.length:
                lib.groupby_accum(kcol, vals, agg_op, kmin, n_slots,
                                  table.sums, table.rowcnt, table.counts)
        maybe_allreduce_table(table)
        keys, sums, counts, n = lib.groupby_compact(
            table.sums, table.rowcnt, table.counts, len(val_names), kmin, n_slots
        )
        table.free()
        return keys, sums, counts, n, biases


    @classmethod
    def _groupby_hash(cls, key_cols, val_cols_per_part, total_rows,
                      want_counts, agg_op):
        """Open-addressing hash groupby for unbounded key ranges (single
        rank); grows the table 4x and retries on overflow.  Near-unique
        keys route straight to the sort-based path: a strided 64K-key
        sample estimating the distinct ratio avoids a doomed (and
        expensive) full-table attempt when the cardinality clearly
        exceeds the 2^27-slot cap."""
        if total_rows > (1 << 27):
            S = 65536
            import numpy as np
            samples = []
            need = S
            for kcol in key_cols:
                if need <= 0 or not kcol.length:
                    continue
                take = min(need, S * kcol.length // max(total_rows, 1) + 1)
                idx = np.linspace(0, kcol.length - 1, take).astype(np.int64)
                samples.append(lib.get(lib.gather(kcol, lib.put(idx))))
                need -= take
            if samples:
                # birthday estimator: s draws from D uniform distinct give
                # ~s^2/2D collisions, so D^ = s^2 / (2 * collisions); only
                # clearly-over-cap cardinalities skip the hash attempt
                sarr = np.concatenate(samples)
                ssz = len(sarr)
                coll = ssz - len(np.unique(sarr))
                d_est = float("inf") if coll == 0 \
                    else ssz * ssz / (2.0 * coll)
                if d_est > 1.5 * (1 << 27):
                    return cls._groupby_sorted(key_cols, val_cols_per_part,
                                               want_counts, agg_op)
        INT64_MIN = -(1 << 63)
        nv = len(val_cols_per_part[0]) if val_cols_per_part else 0
        # start near 2x the row count (capped): avoids doomed attempts on
        # high-cardinality inputs
        H = 1 << max(12, min(27, (2 * total_rows - 1).bit_length()
                             if total_rows else 12))
        while True:
            L = H + 1
            tkey = lib.alloc_raw(8 * L)
            sums = lib.alloc_raw(8 * nv * L) if nv else lib.alloc_raw(8)
            rowcnt = lib.alloc_raw(8 * L)
            counts = lib.alloc_raw(8 * nv * L) if (want_counts and nv) else 0
            try:
                lib.fill_i64(tkey, INT64_MIN, L)
                init = lib.AGG_IDENTITY[agg_op]
                if nv:
                    if init == 0.0:
                        lib.memset_raw(sums, 0, 8 * nv * L)
                    else:
                        lib.fill_f64(sums, init, nv * L)
                lib.memset_raw(rowcnt, 0, 8 * L)
                if counts:
                    lib.memset_raw(counts, 0, 8 * nv * L)
                for kcol, vals in zip(key_cols, val_cols_per_part):
                    if kcol.length:
                        lib.groupby_hash_accum(kcol, vals, agg_op, H, tkey,
                                               sums, rowcnt, counts)
                return (*lib.groupby_hash_compact(tkey, sums, rowcnt, counts,
                                                  nv, H),)
            except lib.HfError as e:
                if "hash table full" in str(e):
                    if H < (1 << 27):
                        H <<= 2
                        continue
                    return cls._groupby_sorted(key_cols, val_cols_per_part,
                                               want_counts, agg_op)
                raise
            finally:
                lib.free_raw(tkey)
                lib.free_raw(sums)
                lib.free_raw(rowcnt)
                if counts:
                    lib.free_raw(counts)

    @classmethod
    def _groupby_sorted(cls, key_cols, val_cols_per_part, want_counts,
                        agg_op):
        """Sort-based groupby: the unbounded-cardinality path (>~134M
        groups).  Concatenate the partitions' rows, key-sort them (stable LSD
        radix, hf_sort_perm), then one segmented-aggregation pass over the
        equal-key runs (hf_groupby_sorted).  Mirrors the reference's
        fallthrough from the Map-Reduce groupby to the full-column groupby
        (algebra/groupby.py: GroupByReduce -> GroupByDefault)."""
        nv = len(val_cols_per_part[0]) if val_cols_per_part else 0
        keys = (key_cols[0] if len(key_cols) == 1
                else lib.concat(key_cols))
        vals = [(val_cols_per_part[0][c] if len(key_cols) == 1
                 else lib.concat([vp[c] for vp in val_cols_per_part]))
                for c in range(nv)]
        perm = lib.sort_perm(keys)
        skeys = lib.gather(keys, perm)
        svals = [lib.gather(v, perm) for v in vals]
        return lib.groupby_sorted(skeys, svals, agg_op, want_counts)


    @classmethod
    def _groupby_local(cls, key_cols, val_cols_per_part, want_counts, agg_op):
        """Single-rank groupby over explicit column lists, no collectives:
        dense table when the local key range is bounded, hash/sorted beyond
        (used for the post-shuffle per-rank aggregation)."""
        kmin = kmax = None
        total_rows = 0
        for kcol in key_cols:
            if kcol.length:
                r = lib.reduce(kcol)
                kmin = r.imn if kmin is None else min(kmin, r.imn)
                kmax = r.imx if kmax is None else max(kmax, r.imx)
            total_rows += kcol.length
        if kmin is None:
            kmin, kmax = 0, -1
        n_slots = max(kmax - kmin + 1, 0)
        if n_slots > config.MaxGroupbySlots.get():
            return cls._groupby_hash(key_cols, val_cols_per_part, total_rows,
                                     want_counts, agg_op)
        n_slots = max(n_slots, 1)
        nv = len(val_cols_per_part[0]) if val_cols_per_part else 0
        table = GroupbyTable(nv, kmin, n_slots, want_counts, agg_op)
        for kcol, vals in zip(key_cols, val_cols_per_part):
            if kcol.length:
                lib.groupby_accum(kcol, vals, agg_op, kmin, n_slots,
                                  table.sums, table.rowcnt, table.counts)
        out = lib.groupby_compact(table.sums, table.rowcnt, table.counts,
                                  nv, kmin, n_slots)
        table.free()
        return out

    @classmethod
    def _groupby_shuffle(cls, key_cols, val_cols_per_part, want_counts,
                         agg_op):
        """Multi-GPU unbounded-key groupby: the range-partitioning shuffle
        (SURVEY §8e option (ii); the device form of the reference's
        ``shuffle_partitions`` / range-partitioning groupby,
        partition_manager.py:1937).  Sampled splitters assign each key RANGE
        to one rank (a group never straddles ranks), rows move once
        (exchange_splits: RCCL all_to_all over xGMI), every rank aggregates
        its range locally with the single-rank engine, and the disjoint
        per-rank results — ascending in rank order by construction — are
        all-gathered so each rank returns the identical replicated result
        (the dense-table path's convention)."""
        import numpy as np
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        nv = len(val_cols_per_part[0]) if val_cols_per_part else 0
        keys = key_cols[0] if len(key_cols) == 1 else lib.concat(key_cols)
        vals = [val_cols_per_part[0][c] if len(key_cols) == 1
                else lib.concat([vp[c] for vp in val_cols_per_part])
                for c in range(nv)]
        # deterministic strided local sample -> identical splitters everywhere
        n = keys.length
        S = min(n, 4096)
        if S:
            idx = np.linspace(0, n - 1, S).astype(np.int64)
            sample = lib.get(lib.gather(keys, lib.put(idx)))
        else:
            sample = np.empty(0, dtype=np.int64)
        splitters = dist_mod.sample_splitters(sample)
        dest = lib.shuffle_dest(keys, splitters)
        send_counts = []
        send_k, send_v = [], [[] for _ in range(nv)]
        for d in range(P):
            mask = lib.compare_scalar(lib.CMP_EQ, dest, float(d))
            plan = lib.filter_plan(mask)
            send_counts.append(plan.n_kept)
            send_k.append(lib.filter_apply(plan, keys))
            for c in range(nv):
                send_v[c].append(lib.filter_apply(plan, vals[c]))
        rk = dist_mod.exchange_column(lib.concat(send_k), send_counts)
        rv = [dist_mod.exchange_column(lib.concat(send_v[c]), send_counts)
              for c in range(nv)]
        k2, s2, c2, _n2 = cls._groupby_local([rk], [rv], want_counts, agg_op)
        gk, gs, gc = dist_mod.allgather_groupby(
            lib.get(k2), [lib.get(c) for c in s2],
            [lib.get(c) for c in c2] if c2 is not None else None)
        keys_col = lib.put(gk)
        sums_cols = [lib.put(a) for a in gs]
        counts_cols = ([lib.put(a) for a in gc] if gc is not None else None)
        return keys_col, sums_cols, counts_cols, len(gk)


def maybe_allreduce_keyrange(kmin, kmax):
    from ..distributed import allreduce_minmax, is_active
    if not is_active():
        return kmin, kmax
    return allreduce_minmax(kmin, kmax)
