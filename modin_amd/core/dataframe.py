"""HipDataframe — the core dataframe (reference L4) on device partitions.

Mirrors ``modin/core/dataframe/pandas/dataframe/dataframe.py``:
``PandasDataframe(partitions, index, columns, row_lengths, ..., dtypes)``
(:82,161) with operators ``map`` (:2253), ``tree_reduce`` (:2208),
``n_ary_op`` (:3851), ``groupby_reduce`` (:4530), ``from_pandas`` (:4592),
and the lazy-metadata pattern of ``modin/core/dataframe/pandas/metadata/``
(ModinIndex): a groupby result's index is a device key column materialized
to a pandas.Index only on demand (and cached) — the D2H copy the reference
performs in ``PartitionManager.get_indices`` (partition_manager.py:1220).
"""

from __future__ import annotations

import numpy as np
import pandas

from . import lib
from .partition import DeviceBlock, HipDataframePartition
from .partition_manager import HipDataframePartitionManager


class DeviceIndex:
    """Lazy index backed by a device int64 column (groupby keys)."""

    def __init__(self, col: lib.ColumnRef, name=None):
        self.col = col
        self.name = name
        self._cache = None

    def materialize(self) -> pandas.Index:
        if self._cache is None:
            self._cache = pandas.Index(lib.get(self.col), name=self.name)
        return self._cache

    def __len__(self):
        return self.col.length


class HipDataframe:
    _partition_mgr_cls = HipDataframePartitionManager

    def __init__(self, partitions, index, columns, row_lengths, dtypes):
        self._partitions = partitions          # list[HipDataframePartition], p×1
        self._index = index                    # pandas.Index | DeviceIndex
        self.columns = pandas.Index(columns)
        self._row_lengths = row_lengths
        self.dtypes = dtypes                   # pandas.Series name -> np.dtype

    # ---- metadata ----
    @property
    def index(self) -> pandas.Index:
        if isinstance(self._index, DeviceIndex):
            return self._index.materialize()
        return self._index

    def __len__(self):
        return sum(self._row_lengths)

    # ---- ingestion (dataframe.py:4592) ----
    @classmethod
    def from_pandas(cls, df: pandas.DataFrame) -> "HipDataframe":
        parts, row_lengths = cls._partition_mgr_cls.from_pandas(df)
        return cls(parts, df.index, df.columns, row_lengths,
                   df.dtypes.copy())

    def to_pandas(self) -> pandas.DataFrame:
        out = self._partition_mgr_cls.to_pandas(self._partitions)
        out.index = self.index
        out = out[list(self.columns)]
        return out.astype(dict(self.dtypes))

    # ---- Map (dataframe.py:2253) ----
    def map(self, block_fn, lazy: bool = False, dtypes=None) -> "HipDataframe":
        mgr = self._partition_mgr_cls
        parts = (mgr.lazy_map_partitions(self._partitions, block_fn) if lazy
                 else mgr.map_partitions(self._partitions, block_fn))
        new_dtypes = dtypes if dtypes is not None else _peek_dtypes(parts, self)
        return HipDataframe(parts, self._index, self.columns,
                            self._row_lengths, new_dtypes)

    # ---- Binary zip (dataframe.py:3851) ----
    def n_ary_op(self, zip_fn, other: "HipDataframe") -> "HipDataframe":
        if self._row_lengths != other._row_lengths:
            if len(self) != len(other):
                raise lib.HfError("n_ary_op: length mismatch")
            # co-partition (dataframe.py:3709 _copartition): re-split the rhs
            # to the lhs row splits via host round trip is NOT offered — the
            # deterministic from_pandas chunking makes equal-length frames
            # align; anything else is a later round.
            raise lib.HfError("n_ary_op: frames are not co-partitioned")
        parts = self._partition_mgr_cls.binary_partitions(
            self._partitions, other._partitions, zip_fn
        )
        return HipDataframe(parts, self._index, self.columns,
                            self._row_lengths, _peek_dtypes(parts, self))

    # ---- TreeReduce (dataframe.py:2208) ----
    def tree_reduce(self, col_names):
        partials = self._partition_mgr_cls.reduce_partitions(
            self._partitions, col_names
        )
        from ..distributed import allreduce_partials, is_active
        if is_active():
            partials = allreduce_partials(partials, list(col_names))
        return partials

    # ---- GroupByReduce (dataframe.py:4530) ----
    def groupby_reduce(self, by: str, agg: str) -> "HipDataframe":
        val_names = [c for c in self.columns if c != by]
        want_counts = agg in ("count", "mean", "min", "max")
        agg_op = lib.AGG_OP_OF[agg]
        keys, sums, counts, n = self._partition_mgr_cls.groupby_reduce(
            self._partitions, by, val_names, want_counts, agg_op
        )
        if agg == "sum":
            cols = {name: sums[i] for i, name in enumerate(val_names)}
            dtypes = pandas.Series({n_: np.dtype(np.float64) for n_ in val_names})
        elif agg == "count":
            cols = {name: counts[i] for i, name in enumerate(val_names)}
            dtypes = pandas.Series({n_: np.dtype(np.int64) for n_ in val_names})
        elif agg in ("min", "max"):
            # empty (all-NaN) groups hold the agg identity; pandas says NaN
            cols = {name: lib.fixup_empty(sums[i], counts[i])
                    for i, name in enumerate(val_names)}
            dtypes = pandas.Series({n_: np.dtype(np.float64) for n_ in val_names})
        else:  # mean = sums / counts (GroupbyReduceImpl mean shape, groupby.py:87)
            cols = {}
            for i, name in enumerate(val_names):
                cnt_f = lib.cast_f64(counts[i])
                cols[name] = lib.binary(lib.BIN_DIV, sums[i], cnt_f)
            dtypes = pandas.Series({n_: np.dtype(np.float64) for n_ in val_names})
        block = DeviceBlock(cols, n)
        part = HipDataframePartition(block)
        return HipDataframe([part], DeviceIndex(keys, name=by), val_names,
                            [n], dtypes)

    # ---- broadcast inner join (MergeImpl.row_axis_merge device form,
    #      merge.py:104-178: combine() the right frame once, probe per left
    #      partition; pandas suffix rules "_x"/"_y" on collisions) ----
    def broadcast_join(self, other: "HipDataframe", on: str) -> "HipDataframe":
        if on not in self.columns or on not in other.columns:
            raise lib.HfError(f"merge: key column {on!r} missing")
        left_names = [c for c in self.columns if c != on]
        right_names = [c for c in other.columns if c != on]
        common = set(left_names) & set(right_names)
        lout = {n: (n + "_x" if n in common else n) for n in left_names}
        rout = {n: (n + "_y" if n in common else n) for n in right_names}

        # materialize right device-side (combine(), dataframe.py:2918)
        def concat_col(frame, name):
            cols = [p.block().columns[name] for p in frame._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        rkeys = concat_col(other, on)
        if rkeys.dtype_code != lib.HF_INT64:
            raise lib.HfError("merge: key column must be int64 (dense-range "
                              "CSR join; hashed keys are a later round)")
        rvals = [concat_col(other, n) for n in right_names]
        from ..distributed import is_active
        if is_active():
            # the broadcast join's combine() across ranks: every rank builds
            # from the FULL right table (all-gather of the right shards),
            # probes only its own left shard — result stays left-sharded
            from .. import distributed as dist_mod
            rdt = [c.dtype_code for c in rvals]
            gathered = dist_mod.allgather_arrays(
                [lib.get(rkeys)] + [lib.get(c) for c in rvals])
            rkeys = lib.put(gathered[0])
            rvals = [lib.put(a) for a in gathered[1:]]
            del rdt
        if rkeys.length:
            r = lib.reduce(rkeys)
            kmin, n_slots = r.imn, r.imx - r.imn + 1
        else:
            kmin, n_slots = 0, 1
        # cache the build side on the right frame: its columns are immutable,
        # so repeated merges with the same right frame skip hist/scan/fill
        # (the lazy-metadata pattern again; a real broadcast join caches its
        # build side)
        cache_key = (on, tuple(right_names), kmin, n_slots)
        cached = getattr(other, "_join_build_cache", None)
        if cached is not None and cached[0] == cache_key:
            j = cached[1]
        else:
            j = lib.join_build(rkeys, rvals, kmin, n_slots)
            other._join_build_cache = (cache_key, j)

        out_parts, lengths = [], []
        for p in self._partitions:
            block = p.block()
            lkeys = block.columns[on]
            if lkeys.dtype_code != lib.HF_INT64:
                raise lib.HfError("merge: key column must be int64")
            keys_c, lidx, rcols, nout = lib.join_probe(j, lkeys)
            cols = {}
            for name in self.columns:  # left column order, key in place
                if name == on:
                    cols[on] = keys_c
                else:
                    cols[lout[name]] = lib.gather(block.columns[name], lidx)
            for i, rn in enumerate(right_names):
                cols[rout[rn]] = rcols[i]
            out_parts.append(HipDataframePartition(DeviceBlock(cols, nout)))
            lengths.append(nout)
        out_columns = ([on if c == on else lout[c] for c in self.columns]
                       + [rout[c] for c in right_names])
        dtypes = {}
        for c in self.columns:
            dtypes[on if c == on else lout[c]] = self.dtypes[c]
        for c in right_names:
            dtypes[rout[c]] = other.dtypes[c]
        total = sum(lengths)
        return HipDataframe(out_parts, pandas.RangeIndex(total), out_columns,
                            lengths, pandas.Series(dtypes))

    # ---- row filter (PandasDataframe.filter / mask device form,
    #      partition.py:224) ----
    def filter_rows(self, mask_frame: "HipDataframe") -> "HipDataframe":
        """Keep rows where the 1-column int64 mask is nonzero.  Result index
        = kept original positions (pandas boolean-mask semantics over the
        RangeIndex these frames carry)."""
        if self._row_lengths != mask_frame._row_lengths:
            raise lib.HfError("filter: mask is not co-partitioned with frame")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError(
                "filter: only RangeIndex frames this round (groupby results "
                "etc. need index gather — later round)")
        out_parts, lengths, idx_cols = [], [], []
        base = 0
        for p, mp, length in zip(self._partitions, mask_frame._partitions,
                                 self._row_lengths):
            mblock = mp.block()
            (mask_col,) = mblock.columns.values()
            plan = lib.filter_plan(mask_col)
            block = p.block()
            cols = {name: lib.filter_apply(plan, col)
                    for name, col in block.columns.items()}
            idx_cols.append(lib.filter_iota(plan, base))
            out_parts.append(HipDataframePartition(DeviceBlock(cols,
                                                               plan.n_kept)))
            lengths.append(plan.n_kept)
            base += length
        idx_col = idx_cols[0] if len(idx_cols) == 1 else lib.concat(idx_cols)
        return HipDataframe(out_parts, DeviceIndex(idx_col, name=None),
                            self.columns, lengths, self.dtypes)

    # ---- comparison map (mask column) ----
    def compare_scalar(self, op_code: int, scalar) -> "HipDataframe":
        def block_fn(block: DeviceBlock) -> DeviceBlock:
            out = {name: lib.compare_scalar(op_code, col, scalar)
                   for name, col in block.columns.items()}
            return DeviceBlock(out, block.length)
        return self.map(block_fn)

    # ---- row range (head/tail; mask/partition.py:224 row-slice form) ----
    def take_row_range(self, start: int, stop: int) -> "HipDataframe":
        start = max(0, min(start, len(self)))
        stop = max(start, min(stop, len(self)))
        out_parts, lengths = [], []
        off = 0
        for p, ln in zip(self._partitions, self._row_lengths):
            lo, hi = max(start - off, 0), min(stop - off, ln)
            if lo < hi:
                block = p.block()
                cols = {n: lib.col_slice(c, lo, hi - lo)
                        for n, c in block.columns.items()}
                out_parts.append(HipDataframePartition(
                    DeviceBlock(cols, hi - lo)))
                lengths.append(hi - lo)
            off += ln
        if not out_parts:
            block = self._partitions[0].block()
            cols = {n: lib.col_slice(c, 0, 0)
                    for n, c in block.columns.items()}
            out_parts, lengths = [HipDataframePartition(DeviceBlock(cols, 0))], [0]
        idx = self.index[start:stop]
        return HipDataframe(out_parts, idx, self.columns, lengths, self.dtypes)

    # ---- astype over all columns ----
    def astype_all(self, dtype) -> "HipDataframe":
        code = {np.dtype(np.float64): lib.MAP_CAST_F64,
                np.dtype(np.int64): lib.MAP_CAST_I64}[np.dtype(dtype)]

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            return DeviceBlock({n: lib.map_scalar(code, c, 0)
                                if c.dtype_code != (lib.HF_FLOAT64 if code == lib.MAP_CAST_F64 else lib.HF_INT64)
                                else c
                                for n, c in block.columns.items()},
                               block.length)
        return self.map(block_fn)

    # ---- row concat (PartitionManager.concat device form,
    #      partition_manager.py:943: stack the partition lists; no device
    #      copy — partitions are immutable) ----
    def concat_rows(self, others: list) -> "HipDataframe":
        cols = list(self.columns)
        for o in others:
            if list(o.columns) != cols:
                raise lib.HfError(
                    "concat: all frames must share the same columns this "
                    "round (NaN-fill alignment is a later round)")
        frames = [self] + list(others)
        parts = [p for f in frames for p in f._partitions]
        lengths = [ln for f in frames for ln in f._row_lengths]
        idx = pandas.Index(np.concatenate([np.asarray(f.index) for f in frames]))
        return HipDataframe(parts, idx, cols, lengths, self.dtypes)

    # ---- sort (PandasDataframe.sort_by device form, dataframe.py:2742;
    #      SURVEY §8f.2): stable radix permutation + column gathers ----
    def sort_rows(self, by: str, ascending: bool = True) -> "HipDataframe":
        if by not in self.columns:
            raise lib.HfError(f"sort_values: column {by!r} missing")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError(
                "sort_values: only RangeIndex frames this round")

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        from ..distributed import is_active
        if is_active():
            return self._sort_rows_distributed(by, ascending, concat_col)
        kcol = concat_col(by)
        perm = lib.sort_perm(kcol, ascending)
        cols = {}
        for name in self.columns:
            src = kcol if name == by else concat_col(name)
            cols[name] = lib.gather(src, perm)
        n = perm.length
        part = HipDataframePartition(DeviceBlock(cols, n))
        return HipDataframe([part], DeviceIndex(perm, name=None),
                            self.columns, [n], self.dtypes)

    def _sort_rows_distributed(self, by, ascending, concat_col):
        """Distributed sort_values: the range-partitioning shuffle + local
        stable sort (SURVEY §8f.2 "reuses the shuffle"; reference
        sort_by -> _apply_func_to_range_partitioning, dataframe.py:2742).
        Sampled splitters put each key RANGE on one rank (rank order =
        global key order), rows move once (exchange_splits), each rank
        radix-sorts its range.  Stability: exchange output is source-rank
        (= global-position) ordered and the per-dest filter preserves
        order, so the local stable sort reproduces pandas order exactly.
        Result: the frame stays sharded, rank r holding globally-sorted
        slice r; the index carries the original global positions."""
        import numpy as np
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        kcol = concat_col(by)
        n = kcol.length
        S = min(n, 4096)
        if S:
            sidx = np.linspace(0, n - 1, S).astype(np.int64)
            sample = lib.get(lib.gather(kcol, lib.put(sidx)))
        else:
            sample = np.empty(0, dtype=np.int64)
        splitters = dist_mod.sample_splitters(sample)
        dest = lib.shuffle_dest(kcol, splitters)
        if not ascending:  # rank 0 takes the LARGEST key range
            dest = lib.map_scalar(lib.MAP_RSUB, dest, P - 1)
        base = dist_mod.global_row_base(n)
        names = list(self.columns)
        send_cols = {m: [] for m in names}
        send_pos, send_counts = [], []
        cols_cat = {m: (kcol if m == by else concat_col(m)) for m in names}
        for d in range(P):
            mask = lib.compare_scalar(lib.CMP_EQ, dest, float(d))
            plan = lib.filter_plan(mask)
            send_counts.append(plan.n_kept)
            send_pos.append(lib.filter_iota(plan, base))
            for m in names:
                send_cols[m].append(lib.filter_apply(plan, cols_cat[m]))
        recv = {m: dist_mod.exchange_column(lib.concat(send_cols[m]),
                                            send_counts)
                for m in names}
        rpos = dist_mod.exchange_column(lib.concat(send_pos), send_counts)
        perm = lib.sort_perm(recv[by], ascending)
        out_cols = {m: lib.gather(recv[m], perm) for m in names}
        pos_sorted = lib.gather(rpos, perm)
        ln = perm.length
        part = HipDataframePartition(DeviceBlock(out_cols, ln))
        return HipDataframe([part], DeviceIndex(pos_sorted, name=None),
                            names, [ln], self.dtypes)

    # ---- dropna mask: AND of per-column notna (pandas dropna(how="any")) ----
    def notna_all_mask(self) -> "HipDataframe":
        def block_fn(block: DeviceBlock) -> DeviceBlock:
            acc = None
            for col in block.columns.values():
                m = lib.compare_scalar(lib.CMP_NOTNA, col, 0.0)
                acc = m if acc is None else lib.binary(lib.BIN_MUL, acc, m)
            return DeviceBlock({"mask": acc}, block.length)
        return self.map(block_fn)

    # ---- column selection (getitem_column_array device form) ----
    def take_columns(self, names) -> "HipDataframe":
        def sel(block: DeviceBlock) -> DeviceBlock:
            return block.select(names)
        parts = [p.add_to_apply_calls(sel) for p in self._partitions]
        return HipDataframe(parts, self._index, names, self._row_lengths,
                            self.dtypes[list(names)])


def _peek_dtypes(parts, frame) -> pandas.Series:
    """Derive result dtypes from the first partition's (drained) block."""
    block = parts[0].block()
    return pandas.Series({name: col.np_dtype for name, col in block.columns.items()})


