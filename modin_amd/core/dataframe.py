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

import os

import numpy as np
import pandas

from . import lib
from .partition import DeviceBlock, HipDataframePartition
from .partition_manager import HipDataframePartitionManager


class DeviceIndex:
    """Lazy index backed by a device int64 column (groupby keys).  With
    ``cats`` the column holds dictionary codes and materializes to the
    category values (string groupby keys)."""

    def __init__(self, col: lib.ColumnRef, name=None, cats=None):
        self.col = col
        self.name = name
        self.cats = cats
        self._cache = None

    def materialize(self) -> pandas.Index:
        if self._cache is None:
            arr = lib.get(self.col)
            if self.cats is not None:
                from .partition import decode_dict
                arr = decode_dict(arr, self.cats)
            self._cache = pandas.Index(arr, name=self.name)
        return self._cache

    def __len__(self):
        return self.col.length


def recode_dict_col(codes: lib.ColumnRef, old_cats, new_cats,
                    missing: int = -1):
    """Re-express dictionary codes in a new dictionary: host LUT
    (new_cats.get_indexer(old_cats), −1 rows preserved via a +1-shifted
    sentinel slot) gathered on device — no string ever touches the GPU.
    `missing` is the code for old categories absent from new_cats (merge
    passes −2 so unmatched categories never collide with NaN's −1)."""
    import numpy as np
    lut = np.empty(len(old_cats) + 1, dtype=np.int64)
    lut[0] = -1
    idxr = new_cats.get_indexer(old_cats)
    if missing != -1:
        idxr = np.where(idxr < 0, missing, idxr)
    lut[1:] = idxr
    lut_col = lib.put(lut)
    shifted = lib.map_scalar(lib.MAP_ADD, codes, 1)
    return lib.gather(lut_col, shifted)


def _compare_dict_col(op_code, codes, cats, scalar):
    """Scalar comparison on a dictionary column IN CODE SPACE: sorted
    categories make order compares a rank threshold (searchsorted), equality
    a code lookup; NaN (code −1) compares False except ``ne`` (pandas
    semantics).  Never touches a string on device."""
    if op_code == lib.CMP_NOTNA:
        return lib.compare_scalar(lib.CMP_NE, codes, -1.0)
    if not isinstance(scalar, str):
        # pandas: str col == non-str -> all False; != -> all True; order
        # compares raise
        if op_code == lib.CMP_EQ:
            return lib.compare_scalar(lib.CMP_EQ, codes, -2.0)
        if op_code == lib.CMP_NE:
            return lib.compare_scalar(lib.CMP_NE, codes, -2.0)
        raise lib.HfError(
            f"ordering comparison between string column and {type(scalar)}")
    if op_code in (lib.CMP_EQ, lib.CMP_NE):
        locs = cats.get_indexer([scalar])
        loc = float(locs[0]) if locs[0] >= 0 else -2.0
        return lib.compare_scalar(op_code, codes, loc)
    left = float(cats.searchsorted(scalar, side="left"))
    right = float(cats.searchsorted(scalar, side="right"))
    if op_code == lib.CMP_GT:     # x > s  <=>  code >= rank_right
        return lib.compare_scalar(lib.CMP_GE, codes, right)
    if op_code == lib.CMP_GE:
        return lib.compare_scalar(lib.CMP_GE, codes, left)
    # LT/LE: exclude NaN's code −1 (pandas: NaN compares False)
    thr = left if op_code == lib.CMP_LT else right
    m = lib.compare_scalar(lib.CMP_LT, codes, thr)
    notna = lib.compare_scalar(lib.CMP_NE, codes, -1.0)
    return lib.binary(lib.BIN_MUL, m, notna)


def union_cats(a, b):
    if a.equals(b):
        return a
    u = a.union(b)
    try:
        return u.sort_values()
    except TypeError:  # mixed types — keep union order
        return u


INAT = np.iinfo(np.int64).min  # pandas' own NaT bit pattern in the ns view


class HipDataframe:
    _partition_mgr_cls = HipDataframePartitionManager

    def __init__(self, partitions, index, columns, row_lengths, dtypes):
        self._partitions = partitions          # list[HipDataframePartition], p×1
        self._index = index                    # pandas.Index | DeviceIndex
        self.columns = pandas.Index(columns)
        self._row_lengths = row_lengths
        self.dtypes = dtypes                   # pandas.Series name -> np.dtype

    # ---- datetime/NaT helpers ----
    def _dt_cols(self) -> set:
        """Names of datetime64[ns]-tagged columns (int64 ns on device)."""
        return {c for c in self.columns
                if isinstance(self.dtypes[c], np.dtype)
                and np.issubdtype(self.dtypes[c], np.datetime64)}

    def _col_has_nat(self, name: str) -> bool:
        """True if the tagged column holds any NaT (iNaT = INT64_MIN).
        One cached device reduce per partition column — O(1) after the
        first call on the same buffers."""
        for p in self._partitions:
            col = p.block().columns[name]
            if col.length and lib.reduce(col).imn == INAT:
                return True
        return False

    def _guard_nat(self, ctx: str, cols=None) -> None:
        """Loud gate for ops whose kernels would treat iNaT as a huge
        negative int (DESIGN.md NaT scope): raises if any datetime
        column in `cols` (default all) holds NaT."""
        dtc = self._dt_cols()
        for c in (cols if cols is not None else self.columns):
            if c in dtc and self._col_has_nat(c):
                raise lib.HfError(
                    f"{ctx}: column {c!r} holds NaT — this op does not "
                    "support NaT yet (dropna() it first)")

    # ---- metadata ----
    @property
    def index(self) -> pandas.Index:
        idx = (self._index.materialize()
               if isinstance(self._index, DeviceIndex) else self._index)
        dt = getattr(self, "_index_dtype", None)
        if dt is not None and idx.dtype != dt:
            idx = idx.astype(dt)
        return idx

    def __len__(self):
        return sum(self._row_lengths)

    # ---- ingestion (dataframe.py:4592) ----
    @classmethod
    def from_pandas(cls, df: pandas.DataFrame) -> "HipDataframe":
        """String/object/category columns dictionary-encode at ingestion
        (ONE frame-wide sorted dictionary — SURVEY §8f.3); numeric columns
        upload as-is.  At world>1 the dictionaries are unified across ranks
        here (all-gather + union + host recode of the codes) so every code
        means the same value on every rank — collectives then operate on
        codes alone."""
        from .partition import encode_dict
        dtypes = df.dtypes.copy()
        cats_map = {}
        edf = df
        for name in df.columns:
            dt = df.dtypes[name]
            if dt in (np.dtype(np.int64), np.dtype(np.float64)):
                continue
            if isinstance(dt, pandas.DatetimeTZDtype):
                raise lib.HfError(
                    f"column {name!r}: timezone-aware datetimes are a "
                    "later round (convert with tz_localize(None))")
            if isinstance(dt, np.dtype) and np.issubdtype(dt, np.datetime64):
                # typed-column tag (SURVEY §8f.3 gateway): the device
                # stores the int64 ns view (NaT = iNaT = INT64_MIN, the
                # same bits pandas uses); every sort/groupby/merge/
                # compare path runs on int64, to_pandas restores the tag
                if edf is df:
                    edf = df.copy()
                edf[name] = df[name].to_numpy().astype(
                    "datetime64[ns]").view(np.int64)
                dtypes[name] = np.dtype("datetime64[ns]")
                continue
            if (dt == np.dtype(object)
                    or isinstance(dt, pandas.CategoricalDtype)
                    or pandas.api.types.is_string_dtype(dt)):
                if edf is df:
                    edf = df.copy()
                codes, cats = encode_dict(df[name])
                edf[name] = codes
                cats_map[name] = cats
            else:
                raise lib.HfError(
                    f"column {name!r} has dtype {dt}: the HipNative backend "
                    "stores int64/float64 device columns (strings are "
                    "dictionary-encoded)")
        from ..distributed import is_active
        if cats_map and is_active():
            from .. import distributed as dist_mod
            for name, cats in cats_map.items():
                gathered = dist_mod.allgather_arrays(
                    [cats.to_numpy(dtype=object)])[0]
                union = pandas.Index(pandas.unique(gathered)).sort_values()
                if not union.equals(cats):
                    lut = union.get_indexer(cats)
                    c = edf[name].to_numpy()
                    edf[name] = np.where(c >= 0, lut[c], -1)
                cats_map[name] = union
        parts, row_lengths = cls._partition_mgr_cls.from_pandas(
            edf, cats=cats_map)
        return cls(parts, df.index, df.columns, row_lengths, dtypes)

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

    def repartition_like(self, row_lengths) -> "HipDataframe":
        """Device re-slice of this frame's columns to the given row
        splits (same total length) — the device form of the reference's
        `_copartition` re-split (dataframe.py:3709), no host round trip."""
        if list(self._row_lengths) == list(row_lengths):
            return self

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        cats = (self._partitions[0].block().cats
                if self._partitions else {})
        full = {c: concat_col(c) for c in self.columns}
        parts, off = [], 0
        for ln in row_lengths:
            cols = {c: lib.col_slice(col, off, ln)
                    for c, col in full.items()}
            parts.append(HipDataframePartition(
                DeviceBlock(cols, ln, cats)))
            off += ln
        return HipDataframe(parts, self._index, self.columns,
                            list(row_lengths), self.dtypes)

    # ---- Binary zip (dataframe.py:3851) ----
    def n_ary_op(self, zip_fn, other: "HipDataframe") -> "HipDataframe":
        if self._row_lengths != other._row_lengths:
            if len(self) != len(other):
                raise lib.HfError("n_ary_op: length mismatch")
            # co-partition (dataframe.py:3709 _copartition): re-slice the
            # rhs to the lhs row splits, device-side
            other = other.repartition_like(self._row_lengths)
        parts = self._partition_mgr_cls.binary_partitions(
            self._partitions, other._partitions, zip_fn
        )
        return HipDataframe(parts, self._index, self.columns,
                            self._row_lengths, _peek_dtypes(parts, self))

    # ---- TreeReduce (dataframe.py:2208) ----
    def tree_reduce(self, col_names):
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        bad = [c for c in col_names if c in blk_cats]
        if bad:
            raise lib.HfError(
                f"reduction over string column(s) {bad}: select numeric "
                "columns (string agg is a later round)")
        partials = self._partition_mgr_cls.reduce_partitions(
            self._partitions, col_names
        )
        from ..distributed import allreduce_partials, is_active
        if is_active():
            partials = allreduce_partials(partials, list(col_names))
        return partials

    KEYCOL = "\x00key\x00"  # internal combined multi-key column

    def _combined_key_frame(self, by_list, dropna: bool = True):
        """Multi-key groupby support: fold the key columns into ONE int64
        key (k1*span2*span3… + k2*span3… + …, mins subtracted — GLOBAL
        mins at world>1 so every rank combines identically), giving the
        existing single-key router (dense/radix/hash/sorted, shuffle) the
        whole multi-key case.  Rows with a NaN string key are dropped
        first (pandas dropna=True).  Returns (frame_with_KEYCOL,
        decode(keys_np) -> pandas.MultiIndex).

        The reference reaches the same point through pandas tuple keys
        (algebra/groupby.py by-list handling); combined ascending order ==
        pandas lexicographic order because the fold is monotone."""
        from ..distributed import is_active
        from .. import distributed as dist_mod
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        for b in by_list:
            if b not in self.columns:
                raise lib.HfError(f"groupby: key column {b!r} missing")
        parts = self._partitions
        dict_keys = [b for b in by_list if b in blk_cats]
        nan_dict_keys = set()
        if dict_keys:
            need = False
            for p in parts:
                for b in dict_keys:
                    c = p.block().columns[b]
                    if c.length and lib.reduce(c).imn < 0:
                        need = True
                        nan_dict_keys.add(b)
            if need and dropna:
                fparts = []
                for p in parts:
                    block = p.block()
                    acc = None
                    for b in dict_keys:
                        m = lib.compare_scalar(lib.CMP_GE,
                                               block.columns[b], 0.0)
                        acc = m if acc is None else lib.binary(
                            lib.BIN_MUL, acc, m)
                    plan = lib.filter_plan(acc)
                    cols = {m2: lib.filter_apply(plan, c)
                            for m2, c in block.columns.items()}
                    fparts.append(HipDataframePartition(
                        DeviceBlock(cols, plan.n_kept, block.cats)))
                parts = fparts
            elif need:
                # dropna=False: NaN string keys become one extra code at
                # the TOP of the column's code space (sorts last per
                # level, pandas' NaN placement); decode maps it back
                fparts = []
                for p in parts:
                    block = p.block()
                    cols = dict(block.columns)
                    for b in nan_dict_keys:
                        ncats = len(blk_cats[b])
                        m = lib.compare_scalar(lib.CMP_GE, cols[b], 0.0)
                        isna = lib.map_scalar(lib.MAP_RSUB, m, 1)
                        cols[b] = lib.binary(
                            lib.BIN_ADD, cols[b],
                            lib.map_scalar(lib.MAP_MUL, isna, ncats + 1))
                    fparts.append(HipDataframePartition(
                        DeviceBlock(cols, block.length, block.cats)))
                parts = fparts
        mins, spans = [], []
        for b in by_list:
            kmin = kmax = None
            for p in parts:
                c = p.block().columns[b]
                if c.dtype_code != lib.HF_INT64:
                    raise lib.HfError(
                        f"groupby: key column {b!r} must be int64 or "
                        "string")
                if c.length:
                    r = lib.reduce(c)
                    kmin = r.imn if kmin is None else min(kmin, r.imn)
                    kmax = r.imx if kmax is None else max(kmax, r.imx)
            if is_active():
                kmin, kmax = dist_mod.allreduce_minmax(kmin, kmax)
            if kmin is None:
                kmin, kmax = 0, 0
            mins.append(kmin)
            spans.append(kmax - kmin + 1)
        total = 1
        for sp in spans:
            total *= sp
            if total > (1 << 62):
                # unbounded combined span: dense-rank fallback — NO tuple
                # hash needed (the pinned sorted-heads composition,
                # test_host_logic.py): per-column stable sort + OR'd run
                # heads give each distinct key tuple a dense ordinal in
                # lexicographic order
                return self._combined_key_frame_sorted(by_list, parts,
                                                       blk_cats)
        strides = [1] * len(by_list)
        for i in range(len(by_list) - 2, -1, -1):
            strides[i] = strides[i + 1] * spans[i + 1]
        new_parts = []
        for p in parts:
            block = p.block()
            comb = None
            for b, mn, st in zip(by_list, mins, strides):
                t = lib.map_scalar(lib.MAP_SUB, block.columns[b], mn)
                if st != 1:
                    t = lib.map_scalar(lib.MAP_MUL, t, st)
                comb = t if comb is None else lib.binary(lib.BIN_ADD,
                                                         comb, t)
            cols = dict(block.columns)
            cols[self.KEYCOL] = comb
            new_parts.append(HipDataframePartition(
                DeviceBlock(cols, block.length, block.cats)))
        columns = list(self.columns) + [self.KEYCOL]
        dtypes = pandas.concat([self.dtypes, pandas.Series(
            {self.KEYCOL: np.dtype(np.int64)})])
        frame = HipDataframe(new_parts, self._index, columns,
                             [p.block().length for p in new_parts]
                             if new_parts else [0], dtypes)

        def decode(keys_np):
            from .partition import decode_dict
            levels = []
            for b, mn, st, sp in zip(by_list, mins, strides, spans):
                lv = (keys_np // st) % sp + mn
                if b in blk_cats:
                    if b in nan_dict_keys and not dropna:
                        lv = np.where(lv >= len(blk_cats[b]), -1, lv)
                    lv = decode_dict(lv, blk_cats[b])
                levels.append(lv)
            return pandas.MultiIndex.from_arrays(levels, names=by_list)

        return frame, decode

    def _combined_key_frame_sorted(self, by_list, parts, blk_cats):
        """Unbounded multi-key fold: dense group ordinals from one stable
        multi-column sort — per-column effective keys compose the sort,
        OR'd run-head flags delimit distinct tuples, cumsum(heads)-1 is
        the dense lexicographic group id, scattered back to row order.
        decode() gathers each original key column at the group's first
        sorted occurrence.  Covers ANY int64/string key combination (the
        pinned sorted-heads prototype); single-rank (group ids are
        rank-local ordinals)."""
        from ..distributed import is_active
        if is_active():
            raise lib.HfError(
                "multi-key groupby beyond a 2^62 combined span at world>1 "
                "is a later round")

        def concat_col(name):
            cs = [p.block().columns[name] for p in parts]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        n = sum(p.block().length for p in parts)
        eff_keys = [self._effective_sort_key(concat_col(b), b in blk_cats,
                                             True)
                    for b in by_list]
        perm = self._compose_sort_perm(eff_keys)
        khead = None
        for ekc, _ in eff_keys:
            h = self._run_head_col(lib.gather(ekc, perm), n)
            khead = h if khead is None else lib.binary(lib.BIN_ADD, khead,
                                                       h)
        if len(eff_keys) > 1:
            khead = lib.compare_scalar(lib.CMP_GE, khead, 1.0)
        gid_sorted = lib.map_scalar(lib.MAP_SUB, lib.cumsum(khead), 1)
        gid = lib.scatter(gid_sorted, perm)
        rep_sorted = lib.filter_iota(lib.filter_plan(khead), 0)
        rep_orig = lib.gather(perm, rep_sorted)   # [ngroups] original rows
        key_cols = {b: concat_col(b) for b in by_list}
        new_parts = []
        off = 0
        for p in parts:
            block = p.block()
            cols = dict(block.columns)
            cols[self.KEYCOL] = lib.col_slice(gid, off, block.length)
            off += block.length
            new_parts.append(HipDataframePartition(
                DeviceBlock(cols, block.length, block.cats)))
        columns = list(self.columns) + [self.KEYCOL]
        dtypes = pandas.concat([self.dtypes, pandas.Series(
            {self.KEYCOL: np.dtype(np.int64)})])
        frame = HipDataframe(new_parts, self._index, columns,
                             [p.block().length for p in new_parts]
                             if new_parts else [0], dtypes)

        def decode(keys_np):
            from .partition import decode_dict
            rows = lib.gather(rep_orig,
                              lib.put(np.asarray(keys_np, dtype=np.int64)))
            levels = []
            for b in by_list:
                lv = lib.get(lib.gather(key_cols[b], rows))
                if b in blk_cats:
                    lv = decode_dict(lv, blk_cats[b])
                levels.append(lv)
            return pandas.MultiIndex.from_arrays(levels, names=by_list)

        return frame, decode

    # ---- GroupByReduce (dataframe.py:4530) ----
    def groupby_reduce(self, by, agg: str,
                       dropna: bool = True) -> "HipDataframe":
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by),
                                                      dropna=dropna)
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_reduce(self.KEYCOL, agg)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        want_counts = agg in ("count", "mean", "min", "max")
        agg_op = lib.AGG_OP_OF[agg]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        dict_vals = [v for v in val_names if v in blk_cats]
        if dict_vals and agg not in ("min", "max", "count"):
            raise lib.HfError(
                f"groupby {agg} over string column(s) {dict_vals}: pandas "
                "supports min/max/count on strings (sum concatenation is "
                "a later round)")
        key_cats = blk_cats.get(by)
        parts = self._partitions
        float_key = (parts and key_cats is None
                     and parts[0].block().columns[by].dtype_code
                     == lib.HF_FLOAT64)
        NANCODE = 1 << 62  # dict NaN group key under dropna=False
        if float_key:
            # pandas drops NaN keys (dropna=True); valid float keys ride
            # the ordered f64->i64 bit transform (monotone, so output
            # order == pandas).  dropna=False keeps them: every NaN
            # canonicalizes to ONE ordered key above ordered(+inf), so
            # the NaN group lands LAST — pandas' dropna=False index order
            # — and the inverse transform decodes it back to NaN.
            fparts = []
            for p in parts:
                block = p.block()
                kc = block.columns[by]
                cols = dict(block.columns)
                if dropna and kc.length and \
                        lib.reduce(kc).count < kc.length:
                    mask = lib.compare_scalar(lib.CMP_NOTNA, kc, 0.0)
                    plan = lib.filter_plan(mask)
                    cols = {m: lib.filter_apply(plan, c)
                            for m, c in block.columns.items()}
                    kc = cols[by]
                cols[by] = lib.ordered_i64(kc)
                fparts.append(HipDataframePartition(
                    DeviceBlock(cols, cols[by].length, block.cats)))
            parts = fparts
        if key_cats is not None:
            has_nan = any(
                p.block().columns[by].length
                and lib.reduce(p.block().columns[by]).imn < 0
                for p in parts)
            if has_nan:
                # dropna=True: filter code == -1 rows (pandas drops NaN
                # groups).  dropna=False: remap -1 to a key ABOVE every
                # code so the NaN group sorts last like pandas; decoded
                # back to -1 (= NaN) on the result keys below.
                fparts = []
                for p in parts:
                    block = p.block()
                    mask = lib.compare_scalar(lib.CMP_GE,
                                              block.columns[by], 0.0)
                    if dropna:
                        plan = lib.filter_plan(mask)
                        cols = {m: lib.filter_apply(plan, c)
                                for m, c in block.columns.items()}
                        fparts.append(HipDataframePartition(
                            DeviceBlock(cols, plan.n_kept, block.cats)))
                    else:
                        isna = lib.map_scalar(lib.MAP_RSUB, mask, 1)
                        k2 = lib.binary(
                            lib.BIN_ADD, block.columns[by],
                            lib.map_scalar(lib.MAP_MUL, isna,
                                           NANCODE + 1))
                        cols = dict(block.columns)
                        cols[by] = k2
                        fparts.append(HipDataframePartition(
                            DeviceBlock(cols, block.length, block.cats)))
                parts = fparts
        if dict_vals:
            # sorted dictionaries: code order == lexicographic order, so
            # string min/max/count ride the numeric path on codes with
            # NaN (-1) masked to float-NaN (pandas skips NaN strings)
            fparts2 = []
            for p in parts:
                b2 = p.block()
                cols2 = dict(b2.columns)
                for v in dict_vals:
                    cols2[v] = lib.fixup_empty(
                        lib.cast_f64(cols2[v]),
                        lib.compare_scalar(lib.CMP_NE, cols2[v], -1.0))
                fparts2.append(HipDataframePartition(
                    DeviceBlock(cols2, b2.length, b2.cats)))
            parts = fparts2
        keys, sums, counts, n, biases = \
            self._partition_mgr_cls.groupby_reduce(
                parts, by, val_names, want_counts, agg_op)
        # pandas dtype preservation: sum/min/max of an int64 (or the int64
        # view of a datetime64) column stays typed — the f64 accumulators
        # are exact below 2^53, huge min/max ride the per-column bias the
        # partition manager applied (added back INT-side here).
        int_vals = {n_ for n_ in val_names
                    if (self.dtypes[n_] == np.dtype(np.int64)
                        or (isinstance(self.dtypes[n_], np.dtype)
                            and np.issubdtype(self.dtypes[n_],
                                              np.datetime64)))}

        def back_to_int(name, col):
            if name not in int_vals:
                return col, np.dtype(np.float64)
            bias = biases[val_names.index(name)]
            if col.length and not bias:
                r = lib.reduce(col)
                if max(abs(r.mn if r.mn == r.mn else 0.0),
                       abs(r.mx if r.mx == r.mx else 0.0)) >= 2.0**53:
                    raise lib.HfError(
                        f"groupby {agg} of int64 column {name!r} exceeds "
                        "2^53: exact int accumulation is a later round")
            out = lib.map_scalar(lib.MAP_CAST_I64, col, 0)
            if bias:
                out = lib.map_scalar(lib.MAP_ADD, out, bias)
            return out, self.dtypes[name]

        if agg == "sum":
            cols, dts = {}, {}
            for i, name in enumerate(val_names):
                cols[name], dts[name] = back_to_int(name, sums[i])
            dtypes = pandas.Series(dts)
        elif agg == "count":
            cols = {name: counts[i] for i, name in enumerate(val_names)}
            dtypes = pandas.Series({n_: np.dtype(np.int64) for n_ in val_names})
        elif agg in ("min", "max"):
            # empty (all-NaN) groups hold the agg identity; pandas says NaN
            # (int64 columns can't produce all-NaN groups, so the cast back
            # is safe)
            cols, dts = {}, {}
            out_val_cats = {}
            for i, name in enumerate(val_names):
                fixed = lib.fixup_empty(sums[i], counts[i])
                if name in blk_cats:
                    # decode: NaN -> code -1, back to int codes + the dict
                    cols[name] = lib.map_scalar(
                        lib.MAP_CAST_I64,
                        lib.map_scalar(lib.MAP_FILLNA, fixed, -1.0), 0)
                    dts[name] = np.dtype(object)
                    out_val_cats[name] = blk_cats[name]
                else:
                    cols[name], dts[name] = back_to_int(name, fixed)
            dtypes = pandas.Series(dts)
        else:  # mean = sums / counts (GroupbyReduceImpl mean shape, groupby.py:87)
            cols = {}
            for i, name in enumerate(val_names):
                cnt_f = lib.cast_f64(counts[i])
                cols[name] = lib.binary(lib.BIN_DIV, sums[i], cnt_f)
            dtypes = pandas.Series({n_: np.dtype(np.float64) for n_ in val_names})
        block = DeviceBlock(cols, n,
                            out_val_cats if agg in ("min", "max") and
                            dict_vals else None)
        part = HipDataframePartition(block)
        if float_key:
            idx = pandas.Index(lib.ordered_to_f64_np(lib.get(keys)),
                               name=by)
        else:
            if key_cats is not None and not dropna and keys.length and \
                    lib.reduce(keys).imx >= NANCODE:
                m2 = lib.compare_scalar(lib.CMP_GE, keys, float(NANCODE))
                keys = lib.binary(
                    lib.BIN_SUB, keys,
                    lib.map_scalar(lib.MAP_MUL, m2, NANCODE + 1))
            idx = DeviceIndex(keys, name=by, cats=key_cats)
        return HipDataframe([part], idx, val_names, [n], dtypes)

    def groupby_var(self, by: str, ddof: int = 1,
                    sqrt: bool = False) -> "HipDataframe":
        """groupby var/std: one extra squared-values pass through the SAME
        sum machinery (sums of x and x², non-NaN counts), composed on
        device as (Σx² − (Σx)²/n)/(n−ddof), clamped at 0 (cancellation)
        and fixed to NaN where n <= ddof (pandas nanvar shape, reference
        groupby var via GroupbyReduceImpl-style pairs)."""
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_var(self.KEYCOL, ddof,
                                                        sqrt)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        bad = [v for v in val_names if v in blk_cats]
        if bad:
            raise lib.HfError(
                f"groupby var/std over string column(s) {bad}")
        keys, key_cats, sums, sumsqs, counts, n = self._groupby_moments(
            by, val_names, blk_cats)
        cols = {}
        for i, name in enumerate(val_names):
            cnt_f = lib.cast_f64(counts[i])
            t = lib.binary(lib.BIN_MUL, sums[i], sums[i])
            t = lib.binary(lib.BIN_DIV, t, cnt_f)
            t = lib.binary(lib.BIN_SUB, sumsqs[i], t)
            denom_i = lib.map_scalar(lib.MAP_ADD, counts[i], -ddof)
            var = lib.binary(lib.BIN_DIV, t, lib.cast_f64(denom_i))
            # clamp tiny negative cancellation: (v+|v|)/2
            var = lib.map_scalar(
                lib.MAP_MUL,
                lib.binary(lib.BIN_ADD, var,
                           lib.map_scalar(lib.MAP_ABS, var, 0)), 0.5)
            # n <= ddof (incl. all-NaN groups) -> NaN
            okm = lib.compare_scalar(lib.CMP_GE, denom_i, 1.0)
            var = lib.fixup_empty(var,
                                  lib.binary(lib.BIN_MUL, denom_i, okm))
            if sqrt:
                var = lib.map_scalar(lib.MAP_SQRT, var, 0.0)
            cols[name] = var
        block = DeviceBlock(cols, n)
        part = HipDataframePartition(block)
        dtypes = pandas.Series({v: np.dtype(np.float64) for v in val_names})
        return HipDataframe([part],
                            DeviceIndex(keys, name=by, cats=key_cats),
                            val_names, [n], dtypes)

    def _groupby_moments(self, by: str, val_names, blk_cats):
        """(keys, key_cats, sums, sumsqs, counts, n) via one SUM pass over
        values and their squares."""
        SQ = "\x00sq\x00"

        def add_sq(block: DeviceBlock) -> DeviceBlock:
            cols = dict(block.columns)
            for v in val_names:
                f = lib.cast_f64(block.columns[v])
                cols[SQ + v] = lib.binary(lib.BIN_MUL, f, f)
            return DeviceBlock(cols, block.length, block.cats)

        parts = self._partition_mgr_cls.map_partitions(self._partitions,
                                                       add_sq)
        key_cats = blk_cats.get(by)
        if key_cats is not None:
            fparts = []
            for p in parts:
                block = p.block()
                if block.columns[by].length and \
                        lib.reduce(block.columns[by]).imn < 0:
                    mask = lib.compare_scalar(lib.CMP_GE,
                                              block.columns[by], 0.0)
                    plan = lib.filter_plan(mask)
                    cols = {m: lib.filter_apply(plan, c)
                            for m, c in block.columns.items()}
                    fparts.append(HipDataframePartition(
                        DeviceBlock(cols, plan.n_kept, block.cats)))
                else:
                    fparts.append(p)
            parts = fparts
        ext_names = list(val_names) + [SQ + v for v in val_names]
        keys, sums, counts, n, _biases = \
            self._partition_mgr_cls.groupby_reduce(
                parts, by, ext_names, True, lib.AGG_SUM)
        k = len(val_names)
        return keys, key_cats, sums[:k], sums[k:], counts[:k], n

    def groupby_size(self, by) -> "HipDataframe":
        """groupby().size(): group row counts INCLUDING NaN values (and
        NaN keys dropped) — a ones column through the sum path."""
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                res = cf.take_columns([self.KEYCOL]).groupby_size(
                    self.KEYCOL)
                res._index = decode(lib.get(res._index.col))
                return res
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})

        def ones(block: DeviceBlock) -> DeviceBlock:
            one = lib.map_scalar(
                lib.MAP_ADD,
                lib.map_scalar(lib.MAP_MUL,
                               lib.cast_f64(block.columns[by])
                               if block.columns[by].dtype_code
                               == lib.HF_INT64 else block.columns[by], 0.0),
                1.0)
            return DeviceBlock({by: block.columns[by], "\x00size\x00": one},
                               block.length, block.cats)

        ones_frame = HipDataframe(
            self._partition_mgr_cls.map_partitions(self._partitions, ones),
            self._index, [by, "\x00size\x00"], self._row_lengths,
            pandas.Series({by: self.dtypes[by],
                           "\x00size\x00": np.dtype(np.float64)}))
        res = ones_frame.groupby_reduce(by, "sum")
        # ones are never NaN, so the f64 sums are exact ints: cast
        block = res._partitions[0].block()
        col = lib.map_scalar(lib.MAP_CAST_I64,
                             block.columns["\x00size\x00"], 0)
        part = HipDataframePartition(DeviceBlock({"size": col}, res._row_lengths[0]))
        return HipDataframe([part], res._index, ["size"],
                            res._row_lengths,
                            pandas.Series({"size": np.dtype(np.int64)}))

    def distinct_stats(self, name: str):
        """Distinct values of one int64/dict column with first-appearance
        positions and group sizes — the engine under Series.unique /
        value_counts / nunique (reference: qc unique/value_counts).  One
        groupby-min pass over the global row position plus one size pass;
        NaN (dict code −1) tracked separately (pandas: unique keeps NaN,
        value_counts/nunique drop it)."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        self._guard_nat("unique/value_counts/nunique", [name])
        POS = "\x00pos\x00"
        parts2 = []
        base = 0
        nan_first = None
        nan_count = 0
        float_col = (self._partitions
                     and name not in blk_cats
                     and self._partitions[0].block().columns[name]
                     .dtype_code == lib.HF_FLOAT64)
        for p, ln in zip(self._partitions, self._row_lengths):
            block = p.block()
            col = block.columns[name]
            ones = lib.compare_scalar(lib.CMP_NOTNA, col, 0.0)
            plan = lib.filter_plan(ones)
            pos = lib.filter_iota(plan, base)
            cols = {name: col, POS: pos}
            if name in blk_cats and col.length and \
                    lib.reduce(col).imn < 0:
                nm = lib.compare_scalar(lib.CMP_EQ, col, -1.0)
                nplan = lib.filter_plan(nm)
                nan_count += nplan.n_kept
                if nplan.n_kept:
                    npos = lib.filter_iota(nplan, base)
                    first = lib.reduce(npos).imn
                    nan_first = first if nan_first is None \
                        else min(nan_first, first)
                # drop the NaN rows from the grouped pass
                keepm = lib.compare_scalar(lib.CMP_GE, col, 0.0)
                kplan = lib.filter_plan(keepm)
                cols = {name: lib.filter_apply(kplan, col),
                        POS: lib.filter_apply(kplan, pos)}
            elif float_col:
                # the NOTNA plan above already excluded NaN rows from
                # ``pos``; track the NaNs separately and transform the
                # kept keys to ordered-i64
                kc = lib.filter_apply(plan, col)
                dropped = col.length - plan.n_kept
                if dropped:
                    nan_count += dropped
                    nm_inv = lib.map_scalar(lib.MAP_RSUB, ones, 1)
                    nplan = lib.filter_plan(nm_inv)
                    npos = lib.filter_iota(nplan, base)
                    first = lib.reduce(npos).imn
                    nan_first = first if nan_first is None \
                        else min(nan_first, first)
                cols = {name: lib.ordered_i64(kc), POS: pos}
            elif col.dtype_code != lib.HF_INT64:
                raise lib.HfError(
                    f"unique/value_counts: column {name!r} has an "
                    "unsupported dtype")
            parts2.append(HipDataframePartition(
                DeviceBlock(cols, cols[name].length, block.cats)))
            base += ln
        frame2 = HipDataframe(
            parts2, pandas.RangeIndex(base), [name, POS],
            [pp.block().length for pp in parts2] if parts2 else [0],
            pandas.Series({name: self.dtypes[name],
                           POS: np.dtype(np.int64)}))
        fmin = frame2.groupby_reduce(name, "min")
        fsize = frame2.take_columns([name]).groupby_size(name)
        keys = lib.get(fmin._index.col)
        firstpos = lib.get(
            fmin._partitions[0].block().columns[POS]).astype(np.int64)
        counts = lib.get(fsize._partitions[0].block().columns["size"])
        if name in blk_cats:
            from .partition import decode_dict
            values = decode_dict(keys, blk_cats[name])
        elif float_col:
            values = lib.ordered_to_f64_np(keys)
        else:
            values = keys
        return {"values": values, "counts": counts, "firstpos": firstpos,
                "nan_count": int(nan_count),
                "nan_firstpos": nan_first}

    def median_columns(self):
        """Per-column median (NaN skipped): quantile_columns([0.5])."""
        return {name: v[0]
                for name, v in self.quantile_columns([0.5]).items()}

    def quantile_columns(self, qs):
        """Per-column quantiles (linear interpolation, NaN skipped):
        NOTNA filter -> ordered radix sort -> the two bracketing elements
        per q sliced on device (pandas nanquantile)."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        out = {}
        for name in self.columns:
            if name in blk_cats:
                raise lib.HfError(
                    f"median/quantile over string column {name!r}")
            cols = [p.block().columns[name] for p in self._partitions]
            col = cols[0] if len(cols) == 1 else lib.concat(cols)
            if col.dtype_code == lib.HF_FLOAT64 and col.length:
                mask = lib.compare_scalar(lib.CMP_NOTNA, col, 0.0)
                plan = lib.filter_plan(mask)
                if plan.n_kept < col.length:
                    col = lib.filter_apply(plan, col)
            n = col.length
            if n == 0:
                out[name] = [float("nan")] * len(qs)
                continue
            key = (lib.ordered_i64(col)
                   if col.dtype_code == lib.HF_FLOAT64 else col)
            perm = lib.sort_perm(key)
            sv = lib.gather(lib.cast_f64(col)
                            if col.dtype_code == lib.HF_INT64 else col,
                            perm)
            vals = []
            for q in qs:
                pos = (n - 1) * float(q)
                lo = int(np.floor(pos))
                hi = int(np.ceil(pos))
                pair = lib.get(lib.col_slice(sv, lo, hi - lo + 1))
                vals.append(float(pair[0] + (pair[-1] - pair[0])
                                  * (pos - lo)))
            out[name] = vals
        return out

    def encode_keys_ordered(self, names) -> "HipDataframe":
        """Copy with the named float64 key columns re-expressed as the
        order-preserving int64 transform (hf_ordered_i64): all NaNs become
        ONE canonical sentinel above every finite key — the device form of
        pandas groupby(dropna=False)'s NaN group, which sorts last."""
        parts = []
        for p in self._partitions:
            b = p.block()
            cols = dict(b.columns)
            for nm in names:
                cols[nm] = lib.ordered_i64(lib.cast_f64(cols[nm]))
            parts.append(HipDataframePartition(
                DeviceBlock(cols, b.length, dict(b.cats))))
        dts = dict(self.dtypes)
        for nm in names:
            dts[nm] = np.dtype(np.int64)
        return HipDataframe(parts, self._index, self.columns,
                            self._row_lengths, pandas.Series(dts))

    def _shuffle_frame_by_key(self, by, with_pos: bool = False):
        """Re-shard the whole frame by the groupby key's sampled ranges
        (hf_shuffle_dest + exchange_column): afterwards every key range —
        hence every GROUP — lives on exactly one rank, so the single-rank
        engines run unchanged on the local shard (the frame-level form of
        partition_manager._groupby_shuffle; reference shuffle_partitions,
        partition_manager.py:1937).  with_pos adds the original global row
        position as a hidden column (transforms route results back)."""
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})

        def concat_col(m):
            cs = [p.block().columns[m] for p in self._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        kcol = concat_col(by)
        ekey, _ = self._effective_sort_key(kcol, by in blk_cats, True)
        n = ekey.length
        S = min(n, 4096)
        if S:
            sidx = np.linspace(0, n - 1, S).astype(np.int64)
            sample = lib.get(lib.gather(ekey, lib.put(sidx)))
        else:
            sample = np.empty(0, dtype=np.int64)
        splitters = dist_mod.sample_splitters(sample)
        dest = lib.shuffle_dest(ekey, splitters)
        names = list(self.columns)
        cols_cat = {m: concat_col(m) for m in names}
        POS = "\x00gpos\x00"
        if with_pos:
            base = dist_mod.global_row_base(n)
            cols_cat[POS] = None  # filled per plan below
        send_cols = {m: [] for m in cols_cat}
        send_counts = []
        for d in range(P):
            plan = lib.filter_plan(
                lib.compare_scalar(lib.CMP_EQ, dest, float(d)))
            send_counts.append(plan.n_kept)
            for m in names:
                send_cols[m].append(lib.filter_apply(plan, cols_cat[m]))
            if with_pos:
                send_cols[POS].append(lib.filter_iota(plan, base))
        recv = {m: dist_mod.exchange_column(lib.concat(send_cols[m]),
                                            send_counts)
                for m in send_cols}
        pos_col = recv.pop(POS, None)
        ln = recv[names[0]].length if names else 0
        part = HipDataframePartition(DeviceBlock(recv, ln, dict(blk_cats)))
        shuf = HipDataframe([part], pandas.RangeIndex(ln), names, [ln],
                            self.dtypes)
        return (shuf, pos_col) if with_pos else shuf

    def _route_back_by_pos(self, res: "HipDataframe",
                           pos_col) -> "HipDataframe":
        """Return shuffled same-length results to their origin ranks: each
        result row travels to the rank owning its original global position
        (hf_shuffle_dest over the rank boundaries), then an inverse-
        permutation scatter restores the original local order.  `self` is
        the pre-shuffle frame (its shard lengths define the boundaries)."""
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        r = dist_mod.rank()
        lens = dist_mod.allgather_lengths(len(self))
        bounds = np.cumsum(lens)[:-1].astype(np.int64)  # P-1 splitters
        dest = lib.shuffle_dest(pos_col, bounds)
        names = list(res.columns)
        blk = res._partitions[0].block()
        send_cols = {m: [] for m in names}
        send_pos, send_counts = [], []
        for d in range(P):
            plan = lib.filter_plan(
                lib.compare_scalar(lib.CMP_EQ, dest, float(d)))
            send_counts.append(plan.n_kept)
            send_pos.append(lib.filter_apply(plan, pos_col))
            for m in names:
                send_cols[m].append(lib.filter_apply(plan, blk.columns[m]))
        recv = {m: dist_mod.exchange_column(lib.concat(send_cols[m]),
                                            send_counts)
                for m in names}
        rpos = dist_mod.exchange_column(lib.concat(send_pos), send_counts)
        my_base = sum(lens[:r])
        lpos = lib.map_scalar(lib.MAP_SUB, rpos, my_base)
        out_cols = {m: lib.scatter(recv[m], lpos) for m in names}
        ln = lens[r]
        part = HipDataframePartition(
            DeviceBlock(out_cols, ln, dict(blk.cats)))
        return HipDataframe([part], pandas.RangeIndex(ln), names, [ln],
                            res.dtypes)

    @staticmethod
    def _replicate_result_frame(res: "HipDataframe") -> "HipDataframe":
        """All-gather the per-rank disjoint (ascending-range) result shards
        in rank order so every rank returns the identical replicated frame
        — the output convention of the dense-table path."""
        from .. import distributed as dist_mod
        names = list(res.columns)
        blk = res._partitions[0].block() if res._partitions else None
        idx_vals = np.asarray(res.index)
        if idx_vals.dtype == object:
            raise lib.HfError("distributed groupby result with a "
                              "non-numeric index is a later round")
        datas = [lib.get(blk.columns[m]) for m in names] if blk else \
            [np.empty(0) for _ in names]
        as_f = idx_vals.dtype.kind == "f"
        g = dist_mod.allgather_arrays(
            datas + [idx_vals.astype(np.float64 if as_f else np.int64)])
        cols = {m: lib.put(g[i]) for i, m in enumerate(names)}
        idx = pandas.Index(g[-1])
        ln = len(idx)
        cats = dict(blk.cats) if blk else {}
        return HipDataframe(
            [HipDataframePartition(DeviceBlock(cols, ln, cats))], idx,
            names, [ln], res.dtypes)

    def groupby_median(self, by) -> "HipDataframe":
        return self.groupby_quantile(by, 0.5)

    def groupby_quantile(self, by, q: float = 0.5) -> "HipDataframe":
        """groupby().quantile(q) (median is q=0.5): sort by (key, value
        na-last) once per value column, then gather the per-group
        bracketing elements and interpolate linearly — offsets from
        groupby size, non-NaN counts from groupby count (pandas
        nanquantile per group)."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            key = by if isinstance(by, str) else \
                (by[0] if len(by) == 1 else None)
            if key is None:
                # multi-key: combine first (global span), then shuffle on
                # the combined key through the single-key path below
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_quantile(self.KEYCOL, q)
                res._index = decode(np.asarray(res.index).astype(np.int64))
                return res
            shuf = self._shuffle_frame_by_key(key)
            with dist_mod.local_mode():
                res = shuf.groupby_quantile(key, q)
            return self._replicate_result_frame(res)
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_quantile(self.KEYCOL, q)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        bad = [v for v in val_names if v in blk_cats]
        if bad:
            raise lib.HfError(f"groupby median over string column(s) {bad}")
        sizes_f = self.groupby_size(by)
        sizes = lib.get(sizes_f._partitions[0].block().columns["size"])
        offs = np.concatenate([[0], np.cumsum(sizes)[:-1]])
        cnt_res = self.groupby_reduce(by, "count")
        cnt_block = cnt_res._partitions[0].block()
        ng = len(sizes)

        def concat_col(m):
            cols = [p.block().columns[m] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        key_col = concat_col(by)
        key_cats = blk_cats.get(by)
        # NaN keys (dict -1 / float NaN) sort LAST via the effective key,
        # so the leading group runs line up with the (NaN-key-dropped)
        # size/count tables
        ekey, _ = self._effective_sort_key(key_col, key_cats is not None,
                                           True)
        out_cols = {}
        for v in val_names:
            vcol = concat_col(v)
            eval_, _ = self._effective_sort_key(
                lib.cast_f64(vcol) if vcol.dtype_code == lib.HF_INT64
                else vcol, False, True)
            perm = self._compose_sort_perm([(ekey, True), (eval_, True)])
            sv = lib.gather(lib.cast_f64(vcol)
                            if vcol.dtype_code == lib.HF_INT64 else vcol,
                            perm)
            cnt = lib.get(cnt_block.columns[v]) if ng else np.empty(0)
            cnt = np.asarray(cnt, dtype=np.int64)
            pos = np.maximum(cnt - 1, 0) * float(q)
            lo = offs + np.floor(pos).astype(np.int64)
            hi = offs + np.ceil(pos).astype(np.int64)
            lo_c = lib.put(lo.astype(np.int64))
            hi_c = lib.put(hi.astype(np.int64))
            a = lib.get(lib.gather(sv, lo_c)) if ng else np.empty(0)
            b = lib.get(lib.gather(sv, hi_c)) if ng else np.empty(0)
            frac = pos - np.floor(pos)
            med = a + (b - a) * frac
            med[cnt == 0] = np.nan
            out_cols[v] = lib.put(med)
        part = HipDataframePartition(DeviceBlock(out_cols, ng))
        dtypes = pandas.Series({v: np.dtype(np.float64)
                                for v in val_names})
        return HipDataframe([part], cnt_res._index, val_names, [ng],
                            dtypes)

    def groupby_firstlast(self, by, last: bool = False) -> "HipDataframe":
        """groupby().first()/last(): per value column, the value at the
        min/max ORIGINAL ROW POSITION among that column's non-NaN rows of
        each group (pandas skips NaN) — one groupby-min/max of a position
        column per value column, then a device gather; groups with no
        non-NaN value fill NaN (float) / code −1 (dict)."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            key = by if isinstance(by, str) else \
                (by[0] if len(by) == 1 else None)
            if key is None:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_firstlast(self.KEYCOL,
                                                              last)
                res._index = decode(np.asarray(res.index).astype(np.int64))
                return res
            shuf = self._shuffle_frame_by_key(key)
            with dist_mod.local_mode():
                res = shuf.groupby_firstlast(key, last)
            return self._replicate_result_frame(res)
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_firstlast(self.KEYCOL,
                                                              last)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        POS = "\x00pos\x00"
        res0 = self.groupby_size(by)
        gkeys = res0.index
        ng = len(gkeys)
        out_cols, dtypes, out_cats = {}, {}, {}
        for v in val_names:
            parts2 = []
            base = 0
            vcols = []
            for p, ln in zip(self._partitions, self._row_lengths):
                block = p.block()
                kcol = block.columns[by]
                vcol = block.columns[v]
                vcols.append(vcol)
                # keep rows whose key AND value are usable: float-NaN
                # keys (dropped groups) and NaN values (pandas first/last
                # skip NaN) both filter out; positions stay global
                ones = lib.compare_scalar(lib.CMP_NOTNA, kcol, 0.0)
                if v in blk_cats:
                    keep = lib.compare_scalar(lib.CMP_NE, vcol, -1.0)
                elif vcol.dtype_code == lib.HF_FLOAT64:
                    keep = lib.compare_scalar(lib.CMP_NOTNA, vcol, 0.0)
                else:
                    keep = None
                mask = (ones if keep is None
                        else lib.binary(lib.BIN_MUL, ones, keep))
                plan0 = lib.filter_plan(mask)
                pos = lib.filter_iota(plan0, base)
                cols = {by: lib.filter_apply(plan0, kcol), POS: pos}
                parts2.append(HipDataframePartition(
                    DeviceBlock(cols, cols[by].length, block.cats)))
                base += ln
            sub = HipDataframe(
                parts2, pandas.RangeIndex(base), [by, POS],
                [pp.block().length for pp in parts2] if parts2 else [0],
                pandas.Series({by: self.dtypes[by],
                               POS: np.dtype(np.int64)}))
            posres = sub.groupby_reduce(by, "max" if last else "min")
            pos_np = lib.get(posres._partitions[0].block().columns[POS])
            skeys = posres.index
            vcat = vcols[0] if len(vcols) == 1 else lib.concat(vcols)
            vals_at = lib.get(lib.gather(vcat,
                                         lib.put(pos_np.astype(np.int64)))) \
                if len(pos_np) else np.empty(0, dtype=vcat.np_dtype)
            idx = gkeys.get_indexer(skeys)
            if v in blk_cats:
                out = np.full(ng, -1, dtype=np.int64)
                out[idx] = vals_at
                dtypes[v] = np.dtype(object)
                out_cats[v] = blk_cats[v]
            elif vcat.dtype_code == lib.HF_INT64:
                out = np.empty(ng, dtype=np.int64)
                out[:] = 0
                out[idx] = vals_at  # int columns: every group has a value
                dtypes[v] = np.dtype(np.int64)
            else:
                out = np.full(ng, np.nan)
                out[idx] = vals_at
                dtypes[v] = np.dtype(np.float64)
            out_cols[v] = lib.put(out)
        part = HipDataframePartition(DeviceBlock(out_cols, ng,
                                                 out_cats))
        return HipDataframe([part], res0._index, val_names, [ng],
                            pandas.Series(dtypes))

    def shift_rows(self, periods: int) -> "HipDataframe":
        """pandas shift(axis=0): device slice + NaN block concat per
        column (int columns become float64, the pandas rule).  Operates on
        the frame as one sequence (partitions concatenated)."""
        if periods == 0:
            return self
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        n = len(self)
        k = min(abs(periods), n)

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        dtc = self._dt_cols()
        out_cols, dtypes, out_cats = {}, {}, {}
        for name in self.columns:
            col = concat_col(name)
            if name in blk_cats:
                # dictionary codes shift with a -1 (NaN) fill; the
                # dictionary itself is untouched
                nanb = lib.alloc(k, lib.HF_INT64)
                if k:
                    lib.fill_i64(nanb.dptr(), -1, k)
                dtypes[name] = self.dtypes[name]
                out_cats[name] = blk_cats[name]
            elif name in dtc:
                # pandas shift on datetime keeps the dtype, filling NaT
                nanb = lib.alloc(k, lib.HF_INT64)
                if k:
                    lib.fill_i64(nanb.dptr(), INAT, k)
                dtypes[name] = self.dtypes[name]
            else:
                if col.dtype_code == lib.HF_INT64:
                    col = lib.cast_f64(col)
                nanb = lib.alloc(k, lib.HF_FLOAT64)
                if k:
                    lib.fill_f64(nanb.dptr(), float("nan"), k)
                dtypes[name] = np.dtype(np.float64)
            if periods > 0:
                kept = lib.col_slice(col, 0, n - k)
                out_cols[name] = lib.concat([nanb, kept])
            else:
                kept = lib.col_slice(col, k, n - k)
                out_cols[name] = lib.concat([kept, nanb])
        part = HipDataframePartition(DeviceBlock(out_cols, n, out_cats))
        return HipDataframe([part], self._index, self.columns, [n],
                            pandas.Series(dtypes))

    def diff_rows(self, periods: int) -> "HipDataframe":
        """pandas diff(axis=0): x − x.shift(periods), composed on the
        concatenated columns (both operands single-partition aligned)."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        if blk_cats:
            raise lib.HfError("diff over string columns is a later round")
        if self._dt_cols():
            raise lib.HfError("diff over datetime columns yields "
                              "timedelta64 — a later round")
        shifted = self.shift_rows(periods) if periods else None
        n = len(self)
        out_cols = {}
        for name in self.columns:
            cols = [p.block().columns[name] for p in self._partitions]
            col = cols[0] if len(cols) == 1 else lib.concat(cols)
            if col.dtype_code == lib.HF_INT64:
                col = lib.cast_f64(col)
            sh = (shifted._partitions[0].block().columns[name]
                  if shifted is not None else col)
            out_cols[name] = lib.binary(lib.BIN_SUB, col, sh)
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        dtypes = pandas.Series({c: np.dtype(np.float64)
                                for c in self.columns})
        return HipDataframe([part], self._index, self.columns, [n],
                            dtypes)

    def cumsum_rows(self, agg_op: int = 0) -> "HipDataframe":
        """pandas cumsum/cummin/cummax (axis=0): the device three-phase
        scan per column (int64 exact, float64 NaN-skipping)."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        if blk_cats:
            raise lib.HfError("cumulative ops over string columns")
        dtc = self._dt_cols()
        if dtc:
            if agg_op not in (lib.AGG_MIN, lib.AGG_MAX):
                raise lib.HfError(
                    f"cumsum/cumprod over datetime columns {sorted(dtc)} "
                    "(pandas raises too — cummin/cummax are supported)")
            self._guard_nat("cummin/cummax", dtc)
        n = len(self)
        out_cols, dtypes = {}, {}
        for name in self.columns:
            cols = [p.block().columns[name] for p in self._partitions]
            col = cols[0] if len(cols) == 1 else lib.concat(cols)
            out_cols[name] = lib.cumsum(col, agg_op)
            dtypes[name] = self.dtypes[name]
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        return HipDataframe([part], self._index, self.columns, [n],
                            pandas.Series(dtypes))

    def clip_columns(self, lower, upper) -> "HipDataframe":
        """pandas clip(axis=None): per-column fmax(lower) then
        fmin(upper); int columns stay int64 for integral bounds (the
        pandas dtype rule rides the map wrapper's promotion)."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        if blk_cats:
            raise lib.HfError("clip over string columns")

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            out = {}
            for name, col in block.columns.items():
                c = col
                if lower is not None:
                    c = lib.map_scalar(lib.MAP_MAX, c, lower)
                if upper is not None:
                    c = lib.map_scalar(lib.MAP_MIN, c, upper)
                out[name] = c
            return DeviceBlock(out, block.length)
        return self.map(block_fn)

    def reduce_axis1(self, op: str) -> "HipDataframe":
        """Row-wise reductions (pandas axis=1): fold the columns with the
        elementwise kernels — sum/count fold ADD over NaN-zeroed values /
        NOTNA masks, min/max fold fmin/fmax (NaN-skipping), mean =
        sum/count.  Returns a 1-column frame aligned with the rows."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        if blk_cats:
            raise lib.HfError(f"{op}(axis=1) over string columns")
        all_int = all(self.dtypes[c] == np.dtype(np.int64)
                      for c in self.columns)

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            cols = list(block.columns.values())
            if op in ("min", "max"):
                bop = lib.BIN_MIN if op == "min" else lib.BIN_MAX
                acc = cols[0]
                for c in cols[1:]:
                    acc = lib.binary(bop, acc, c)
                return DeviceBlock({"\x00r\x00": acc}, block.length)
            sums = None
            cnts = None
            for c in cols:
                f = lib.cast_f64(c) if c.dtype_code == lib.HF_INT64 else c
                fz = lib.map_scalar(lib.MAP_FILLNA, f, 0.0)
                sums = fz if sums is None else lib.binary(lib.BIN_ADD,
                                                          sums, fz)
                if op in ("mean", "count"):
                    m = lib.compare_scalar(lib.CMP_NOTNA, c, 0.0)
                    cnts = m if cnts is None else lib.binary(lib.BIN_ADD,
                                                             cnts, m)
            if op == "count":
                return DeviceBlock({"\x00r\x00": cnts}, block.length)
            if op == "mean":
                return DeviceBlock(
                    {"\x00r\x00": lib.binary(lib.BIN_DIV, sums,
                                              lib.cast_f64(cnts))},
                    block.length)
            if all_int:  # exact: int sums (no NaN possible)
                sums = lib.map_scalar(lib.MAP_CAST_I64, sums, 0)
            return DeviceBlock({"\x00r\x00": sums}, block.length)

        mgr = self._partition_mgr_cls
        parts = mgr.map_partitions(self._partitions, block_fn)
        dt = (np.dtype(np.int64)
              if (op == "count" or (all_int and op in ("sum", "min",
                                                       "max")))
              else np.dtype(np.float64))
        return HipDataframe(parts, self._index, ["\x00r\x00"],
                            self._row_lengths,
                            pandas.Series({"\x00r\x00": dt}))

    def idx_extreme(self, maximum: bool) -> dict:
        """Per-column idxmax/idxmin: the FIRST original position holding
        the column's max/min (NaN skipped) — one cached reduce + an EQ
        filter + a position min."""
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        bad = [c for c in self.columns if c in blk_cats]
        if bad:
            raise lib.HfError(f"idxmax/idxmin over string column(s) {bad}")
        out = {}
        for name in self.columns:
            cols = [p.block().columns[name] for p in self._partitions]
            col = cols[0] if len(cols) == 1 else lib.concat(cols)
            if not col.length:
                out[name] = float("nan")
                continue
            r = lib.reduce(col)
            if r.count == 0:
                out[name] = float("nan")
                continue
            if col.dtype_code == lib.HF_INT64:
                target = float(r.imx if maximum else r.imn)
            else:
                target = r.mx if maximum else r.mn
            m = lib.compare_scalar(lib.CMP_EQ, col, target)
            plan = lib.filter_plan(m)
            pos = lib.filter_iota(plan, 0)
            out[name] = int(lib.reduce(pos).imn)
        return out

    def groupby_nunique(self, by) -> "HipDataframe":
        """groupby().nunique(): per value column, the count of DISTINCT
        non-NaN values in each group — sort by (key, value), mark run
        boundaries with exact device compares (int64 subtract for keys,
        f64 subtract for NaN-free values), filter to the distinct pairs
        and size-count them per key; groups whose values are all NaN
        report 0 (pandas dropna=True)."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            key = by if isinstance(by, str) else \
                (by[0] if len(by) == 1 else None)
            if key is None:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_nunique(self.KEYCOL)
                res._index = decode(np.asarray(res.index).astype(np.int64))
                return res
            shuf = self._shuffle_frame_by_key(key)
            with dist_mod.local_mode():
                res = shuf.groupby_nunique(key)
            return self._replicate_result_frame(res)
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_nunique(self.KEYCOL)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        res0 = self.groupby_size(by)
        gkeys = res0.index
        ng = len(gkeys)
        out_cols = {}
        for v in val_names:
            sub = self.take_columns([by, v])
            fparts = []
            for p in sub._partitions:
                block = p.block()
                vcol = block.columns[v]
                if v in blk_cats:
                    keep = lib.compare_scalar(lib.CMP_NE, vcol, -1.0)
                elif vcol.dtype_code == lib.HF_FLOAT64:
                    keep = lib.compare_scalar(lib.CMP_NOTNA, vcol, 0.0)
                else:
                    keep = None
                if keep is not None:
                    plan = lib.filter_plan(keep)
                    cols = {m: lib.filter_apply(plan, c)
                            for m, c in block.columns.items()}
                    fparts.append(HipDataframePartition(
                        DeviceBlock(cols, plan.n_kept, block.cats)))
                else:
                    fparts.append(p)
            sub = HipDataframe(fparts, pandas.RangeIndex(
                sum(pp.block().length for pp in fparts)),
                [by, v],
                [pp.block().length for pp in fparts] if fparts else [0],
                self.dtypes[[by, v]])
            srt = sub.sort_rows([by, v], True)
            sblock = srt._partitions[0].block()
            n2 = sblock.length
            if n2 == 0:
                counts_sub = np.empty(0, dtype=np.int64)
                skeys_idx = pandas.Index(np.empty(0, dtype=np.int64),
                                         name=by)
            else:
                def run_head(col):
                    # head[i] = (col[i] != col[i-1]); row 0 is ALWAYS a
                    # head: the shifted-in sentinel is col[0]^1 (i64) /
                    # NaN (f64), guaranteed != col[0]
                    prev_body = lib.col_slice(col, 0, n2 - 1)
                    first = lib.alloc(1, col.dtype_code)
                    if col.dtype_code == lib.HF_FLOAT64:
                        lib.fill_f64(first.dptr(), float("nan"), 1)
                    else:
                        v0 = int(lib.get(lib.col_slice(col, 0, 1))[0])
                        lib.fill_i64(first.dptr(), v0 ^ 1, 1)
                    prev = lib.concat([first, prev_body])
                    d = lib.binary(lib.BIN_SUB, col, prev)
                    return lib.compare_scalar(lib.CMP_NE, d, 0.0)

                hk = run_head(sblock.columns[by])
                hv = run_head(sblock.columns[v])
                head = lib.binary(lib.BIN_ADD, hk, hv)  # >0 == new pair
                head = lib.compare_scalar(lib.CMP_GE, head, 1.0)
                plan = lib.filter_plan(head)
                dkeys = lib.filter_apply(plan, sblock.columns[by])
                dpart = HipDataframePartition(DeviceBlock(
                    {by: dkeys}, plan.n_kept,
                    {by: blk_cats[by]} if by in blk_cats else {}))
                dframe = HipDataframe([dpart],
                                      pandas.RangeIndex(plan.n_kept),
                                      [by], [plan.n_kept],
                                      self.dtypes[[by]])
                dsz = dframe.groupby_size(by)
                counts_sub = lib.get(
                    dsz._partitions[0].block().columns["size"])
                skeys_idx = dsz.index
            out = np.zeros(ng, dtype=np.int64)
            idx = gkeys.get_indexer(skeys_idx)
            out[idx] = counts_sub
            out_cols[v] = lib.put(out)
        part = HipDataframePartition(DeviceBlock(out_cols, ng))
        dtypes = pandas.Series({v: np.dtype(np.int64) for v in val_names})
        return HipDataframe([part], res0._index, val_names, [ng], dtypes)

    # ---- groupby transforms (same-length results in ORIGINAL row order:
    # pandas DataFrameGroupBy.cumsum/cummin/cummax/cumcount/rank;
    # reference: modin/pandas/groupby.py fallback through
    # _wrap_aggregation / default2pandas transform) ----
    @staticmethod
    def _run_head_col(col, n):
        """head[i] = (col[i] != col[i-1]) over an EXACT-comparable (i64 or
        NaN-free f64) column; row 0 is always a head (the shifted-in
        sentinel differs by construction)."""
        prev_body = lib.col_slice(col, 0, n - 1)
        first = lib.alloc(1, col.dtype_code)
        if col.dtype_code == lib.HF_FLOAT64:
            lib.fill_f64(first.dptr(), float("nan"), 1)
        else:
            v0 = int(lib.get(lib.col_slice(col, 0, 1))[0])
            lib.fill_i64(first.dptr(), v0 ^ 1, 1)
        prev = lib.concat([first, prev_body])
        d = lib.binary(lib.BIN_SUB, col, prev)
        return lib.compare_scalar(lib.CMP_NE, d, 0.0)

    @staticmethod
    def _iota(n):
        """Device [0, n) int64 column (an all-ones filter plan's kept
        positions)."""
        ones = lib.alloc(n, lib.HF_INT64)
        lib.fill_i64(ones.dptr(), 1, n)
        plan = lib.filter_plan(ones)
        return lib.filter_iota(plan, 0)

    @staticmethod
    def _const_i64(value):
        c = lib.alloc(1, lib.HF_INT64)
        lib.fill_i64(c.dptr(), int(value), 1)
        return c

    def groupby_transform(self, by, how: str, ascending: bool = True,
                          method: str = "average", periods: int = 1,
                          dropna: bool = True,
                          na_option: str = "keep") -> "HipDataframe":
        """Same-length groupby transforms in original row order.

        how: 'cumsum' | 'cummin' | 'cummax' (segmented scan), 'cumcount'
        (position within group), 'rank' (method 'average'|'min'|'first',
        ascending per pandas; na_option='keep').

        Device composition: stable sort by key (LSD radix keeps original
        order within groups), key-run head flags delimit the segments, the
        segmented scan / position arithmetic runs in sorted order, and the
        inverse-permutation scatter restores original order.  Rows whose
        key is NaN belong to no group (pandas dropna=True) and come back
        NaN — which also forces float64 results, exactly pandas' dtype
        rule (int64 stays int64 only when every key is valid).

        Reference semantics pinned against pandas 2.3.3 in-container
        (NaN keys -> NaN in every transform incl. cumcount; NaN values
        stay NaN and don't advance cum* state; rank na_option='keep')."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            # shuffle by the FIRST key's ranges (every multi-key group
            # shares its first key, so groups stay rank-local), run the
            # single-rank transform locally, route rows back to their
            # origin rank by global position (reference: the same
            # shuffle_partitions recipe, applied row-preserving)
            by_l = [by] if isinstance(by, str) else list(by)
            shuf, pos_col = self._shuffle_frame_by_key(by_l[0],
                                                       with_pos=True)
            with dist_mod.local_mode():
                res = shuf.groupby_transform(by, how, ascending=ascending,
                                             method=method, periods=periods,
                                             dropna=dropna,
                                             na_option=na_option)
            if how == "ngroup":
                # global group ordinal = local ordinal + #groups on lower
                # ranks (rank ranges ascend, pandas numbers sorted groups)
                col = res._partitions[0].block().columns[how]
                r = lib.reduce(lib.cast_f64(col)) if col.length else None
                n_local = int(r.mx) + 1 if r is not None and r.count else 0
                offs = dist_mod.allgather_lengths(n_local)
                my_off = sum(offs[: dist_mod.rank()])
                if my_off:
                    newc = lib.map_scalar(lib.MAP_ADD, col, float(my_off))
                    blk = res._partitions[0].block()
                    cols2 = dict(blk.columns)
                    cols2[how] = newc
                    res = HipDataframe(
                        [HipDataframePartition(
                            DeviceBlock(cols2, blk.length, blk.cats))],
                        res._index, list(res.columns), [blk.length],
                        res.dtypes)
            return self._route_back_by_pos(res, pos_col)
        by_list = [by] if isinstance(by, str) else list(by)
        for b in by_list:
            if b not in self.columns:
                raise lib.HfError(f"groupby: key column {b!r} missing")
        if how not in ("cumsum", "cummin", "cummax", "cumprod", "cumcount",
                       "rank", "ngroup", "shift", "diff", "ffill", "bfill",
                       "bsum", "bmin", "bmax", "bcount", "bmean"):
            raise lib.HfError(f"groupby transform {how!r} not supported")
        if how == "rank" and method not in ("average", "min", "first"):
            raise lib.HfError(f"rank method {method!r} not supported")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError("groupby transforms: only RangeIndex frames "
                              "this round")
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        val_names = [c for c in self.columns if c not in by_list]
        n = len(self)
        needs_vals = how not in ("cumcount", "ngroup")
        if needs_vals:
            for v in val_names:
                if v in blk_cats:
                    raise lib.HfError(
                        f"groupby {how}: string column {v!r} unsupported "
                        "(pandas raises on non-numeric transforms)")
        if n == 0 or (needs_vals and not val_names):
            names = val_names if needs_vals else [how]
            dts = pandas.Series({v: np.dtype(np.float64) for v in names})
            part = HipDataframePartition(DeviceBlock(
                {v: lib.alloc(0, lib.HF_FLOAT64) for v in names}, 0))
            return HipDataframe([part], pandas.RangeIndex(0), names,
                                [0], dts)

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        # per-row key validity in ORIGINAL order (pandas dropna=True);
        # dropna=False keeps NaN keys as a real group (the canonical-NaN
        # effective key makes all NaNs one run) — duplicated() rides this
        valid = None
        for b in by_list if dropna else []:
            c = concat_col(b)
            if b in blk_cats:
                m = lib.compare_scalar(lib.CMP_GE, c, 0.0)
            elif c.dtype_code == lib.HF_FLOAT64:
                m = lib.compare_scalar(lib.CMP_NOTNA, c, 0.0)
            else:
                continue
            valid = m if valid is None else lib.binary(lib.BIN_MUL, valid, m)
        if valid is not None and lib.reduce(valid).isum == n:
            valid = None  # every key valid — int dtypes survive
        eff_keys = [self._effective_sort_key(concat_col(b), b in blk_cats,
                                             True)
                    for b in by_list]
        if how == "rank":
            return self._groupby_rank(by_list, val_names, eff_keys, valid,
                                      n, concat_col, ascending, method,
                                      na_option)
        perm = self._compose_sort_perm(eff_keys)
        head = None
        for ekc, _ in eff_keys:
            h = self._run_head_col(lib.gather(ekc, perm), n)
            head = h if head is None else lib.binary(lib.BIN_ADD, head, h)
        if len(eff_keys) > 1:
            head = lib.compare_scalar(lib.CMP_GE, head, 1.0)
        out_cols, dts = {}, {}
        if how in ("cumcount", "ngroup"):
            plan = lib.filter_plan(head)
            hp = lib.filter_iota(plan, 0)           # run start positions
            rid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(head), 1)
            if how == "cumcount":
                start = lib.gather(hp, rid)
                cc = lib.binary(lib.BIN_SUB, self._iota(n), start)
            else:
                # ngroup: 0-based group id in SORTED key order (pandas
                # sort=True default) == the run id; NaN-key runs sort last
                # so valid group numbering is unaffected before the fixup
                cc = rid
            res = lib.scatter(cc, perm)
            if valid is not None:
                res = lib.fixup_empty(lib.cast_f64(res), valid)
            name = how
            part = HipDataframePartition(DeviceBlock({name: res}, n))
            dt = np.dtype(np.int64 if valid is None else np.float64)
            return HipDataframe([part], pandas.RangeIndex(n), [name], [n],
                                pandas.Series({name: dt}))
        if how in ("shift", "diff"):
            periods = int(periods)
            plan = lib.filter_plan(head)
            ng = plan.n_kept
            hp = lib.filter_iota(plan, 0)
            rid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(head), 1)
            start = lib.gather(hp, rid)
            nxt = lib.concat([lib.col_slice(hp, 1, ng - 1),
                              self._const_i64(n)]) if ng > 1 \
                else self._const_i64(n)
            end = lib.gather(nxt, rid)              # run end (exclusive)
            pos = self._iota(n)
            idx = lib.map_scalar(lib.MAP_SUB, pos, periods)
            # source row must stay inside the run: start <= idx < end
            ok = lib.binary(
                lib.BIN_MUL,
                lib.compare_scalar(
                    lib.CMP_GE, lib.binary(lib.BIN_SUB, idx, start), 0.0),
                lib.compare_scalar(
                    lib.CMP_GE,
                    lib.binary(lib.BIN_SUB,
                               lib.map_scalar(lib.MAP_SUB, end, 1), idx),
                    0.0))
            cidx = lib.map_scalar(
                lib.MAP_MAX,
                lib.map_scalar(lib.MAP_MIN, idx, max(n - 1, 0)), 0)
            for v in val_names:
                sv = lib.gather(lib.cast_f64(concat_col(v)), perm)
                sh = lib.fixup_empty(lib.gather(sv, cidx), ok)
                if how == "diff":
                    sh = lib.binary(lib.BIN_SUB, sv, sh)
                res = lib.scatter(sh, perm)
                if valid is not None:
                    res = lib.fixup_empty(res, valid)
                out_cols[v] = res
                dts[v] = np.dtype(np.float64)
            part = HipDataframePartition(DeviceBlock(out_cols, n))
            return HipDataframe([part], pandas.RangeIndex(n), val_names,
                                [n], pandas.Series(dts))
        if how in ("ffill", "bfill"):
            # pinned prototype (test_host_logic.py
            # test_ffill_bfill_composition_prototype): segmented MAX over
            # (valid ? sorted position : −1) finds each row's last valid
            # source within its key run; bfill = ffill over reversed rows
            # (the reversal permutation is its own inverse).
            rstate = None
            if how == "bfill":
                # reversal permutation (its own inverse) + reversed sort
                # state, built once; the index upload is 8n B H2D —
                # a device iota-reversal is a later micro-optimization
                ridx = lib.put(np.arange(n - 1, -1, -1, dtype=np.int64))
                reff = [(lib.gather(ekc, ridx), ea)
                        for ekc, ea in eff_keys]
                rperm = self._compose_sort_perm(reff)
                rhead = None
                for ekc, _ in reff:
                    h = self._run_head_col(lib.gather(ekc, rperm), n)
                    rhead = h if rhead is None else lib.binary(
                        lib.BIN_ADD, rhead, h)
                if len(reff) > 1:
                    rhead = lib.compare_scalar(lib.CMP_GE, rhead, 1.0)
                rstate = (ridx, rperm, rhead)
            for v in val_names:
                vc = concat_col(v)
                src_int = vc.dtype_code == lib.HF_INT64
                if src_int:
                    # int columns hold no NaN: fill is the identity
                    if valid is None:
                        out_cols[v] = vc
                        dts[v] = np.dtype(np.int64)
                    else:
                        out_cols[v] = lib.fixup_empty(lib.cast_f64(vc),
                                                      valid)
                        dts[v] = np.dtype(np.float64)
                    continue
                if how == "bfill":
                    ridx, rperm, rhead = rstate
                    filled = self._seg_ffill_col(lib.gather(vc, ridx),
                                                 rperm, rhead, n)
                    res = lib.gather(filled, ridx)
                else:
                    res = self._seg_ffill_col(vc, perm, head, n)
                if valid is not None:
                    res = lib.fixup_empty(res, valid)
                out_cols[v] = res
                dts[v] = np.dtype(np.float64)
            part = HipDataframePartition(DeviceBlock(out_cols, n))
            return HipDataframe([part], pandas.RangeIndex(n), val_names,
                                [n], pandas.Series(dts))
        if how.startswith("b"):
            # broadcast aggregate (pandas gb.transform('sum'|'mean'|...)):
            # per-run aggregate = segmented-scan value at the run's LAST
            # row (NaN pre-filled with the identity so the end value is
            # the true aggregate), gathered back per row id
            agg = how[1:]
            op = {"sum": lib.AGG_SUM, "mean": lib.AGG_SUM,
                  "count": lib.AGG_SUM, "min": lib.AGG_MIN,
                  "max": lib.AGG_MAX}[agg]
            ident = {"min": float("inf"), "max": float("-inf")}.get(agg,
                                                                    0.0)
            plan = lib.filter_plan(head)
            ng = plan.n_kept
            hp = lib.filter_iota(plan, 0)
            rid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(head), 1)
            ends = lib.concat([lib.col_slice(hp, 1, ng - 1),
                               self._const_i64(n)]) if ng > 1 \
                else self._const_i64(n)
            ends_m1 = lib.map_scalar(lib.MAP_SUB, ends, 1)  # run last row
            for v in val_names:
                vc = concat_col(v)
                src_int = vc.dtype_code == lib.HF_INT64
                sv = lib.gather(vc, perm)
                if src_int:
                    m = None
                    dense = sv
                else:
                    m = lib.compare_scalar(lib.CMP_NOTNA, sv, 0.0)
                    dense = lib.map_scalar(lib.MAP_FILLNA, sv, ident)
                def per_run_count():
                    mm = m
                    if mm is None:  # int column: every row counts
                        mm = lib.alloc(n, lib.HF_INT64)
                        lib.fill_i64(mm.dptr(), 1, n)
                    segc = lib.seg_cumsum(mm, head, lib.AGG_SUM)
                    return lib.gather(lib.gather(segc, ends_m1), rid)

                if agg == "count":
                    br = per_run_count()
                else:
                    seg_a = lib.seg_cumsum(dense, head, op)
                    br = lib.gather(lib.gather(seg_a, ends_m1), rid)
                    if agg == "mean":
                        br = lib.binary(lib.BIN_DIV, lib.cast_f64(br),
                                        lib.cast_f64(per_run_count()))
                    elif agg in ("min", "max") and m is not None:
                        # all-NaN group: min/max is NaN, not ±inf
                        br = lib.fixup_empty(
                            lib.cast_f64(br),
                            lib.compare_scalar(lib.CMP_GE,
                                               per_run_count(), 1.0))
                res = lib.scatter(br, perm)
                if valid is not None:
                    res = lib.fixup_empty(lib.cast_f64(res), valid)
                out_cols[v] = res
                is_int = (valid is None
                          and (agg == "count"
                               or (src_int
                                   and agg in ("sum", "min", "max"))))
                dts[v] = np.dtype(np.int64 if is_int else np.float64)
            part = HipDataframePartition(DeviceBlock(out_cols, n))
            return HipDataframe([part], pandas.RangeIndex(n), val_names,
                                [n], pandas.Series(dts))
        agg_op = {"cumsum": lib.AGG_SUM, "cummin": lib.AGG_MIN,
                  "cummax": lib.AGG_MAX, "cumprod": lib.AGG_PROD}[how]
        for v in val_names:
            vc = concat_col(v)
            src_int = vc.dtype_code == lib.HF_INT64
            if src_int and valid is not None:
                vc = lib.cast_f64(vc)
            sv = lib.gather(vc, perm)
            seg = lib.seg_cumsum(sv, head, agg_op)
            res = lib.scatter(seg, perm)
            if valid is not None:
                res = lib.fixup_empty(res, valid)
            out_cols[v] = res
            dts[v] = np.dtype(np.int64 if (src_int and valid is None)
                              else np.float64)
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        return HipDataframe([part], pandas.RangeIndex(n), val_names, [n],
                            pandas.Series(dts))

    @staticmethod
    def _seg_ffill_col(vc, perm, head, n):
        """ffill one f64 column within key runs: segmented MAX over
        (valid ? sorted position : −1), gather the source rows, NaN
        where no prior valid exists, inverse-scatter to original order
        (the pinned prototype, test_host_logic.py)."""
        sv = lib.gather(vc, perm)
        m = lib.compare_scalar(lib.CMP_NOTNA, sv, 0.0)
        pos = HipDataframe._iota(n)
        posv = lib.binary(lib.BIN_ADD,
                          lib.binary(lib.BIN_MUL, pos, m),
                          lib.map_scalar(lib.MAP_SUB, m, 1))
        segmax = lib.seg_cumsum(posv, head, lib.AGG_MAX)
        ok = lib.compare_scalar(lib.CMP_GE, segmax, 0.0)
        cidx = lib.map_scalar(lib.MAP_MAX, segmax, 0)
        filled = lib.fixup_empty(lib.gather(sv, cidx), ok)
        return lib.scatter(filled, perm)

    def _with_const_key(self) -> "HipDataframe":
        """Copy of this frame with a zeros KEYCOL — frame-level transforms
        (rank, ffill/bfill) ride the groupby machinery over ONE group."""
        parts = []
        for p in self._partitions:
            b = p.block()
            z = lib.alloc(b.length, lib.HF_INT64)
            lib.fill_i64(z.dptr(), 0, b.length)
            cols = dict(b.columns)
            cols[self.KEYCOL] = z
            parts.append(HipDataframePartition(
                DeviceBlock(cols, b.length, b.cats)))
        dtypes = pandas.concat([self.dtypes, pandas.Series(
            {self.KEYCOL: np.dtype(np.int64)})])
        return HipDataframe(parts, self._index,
                            list(self.columns) + [self.KEYCOL],
                            self._row_lengths, dtypes)

    def rolling_agg(self, window: int, min_periods, op: str
                    ) -> "HipDataframe":
        """pandas rolling(window, min_periods).sum/mean/count/min/max —
        no new kernels (the pinned prototype, test_host_logic.py):
        sum/mean/count ride windowed differences of NaN-zero-filled
        prefix sums; min/max ride the van Herk/Gil-Werman two-scan (per
        w-aligned tile, prefix extreme = segmented scan with tile-start
        heads; suffix extreme = the same scan over reversed rows) with
        window extreme = comb(suffix[lo], prefix[i]).  Gates: count
        emits when the window holds >= min_periods ROWS; the other aggs
        when it holds >= min_periods OBSERVATIONS (non-NaN) — measured
        pandas 2.3.3 behavior.  Results float64 (pandas)."""
        w = int(window)
        minp = w if min_periods is None else int(min_periods)
        if w < 1 or minp < 0 or minp > w:
            raise lib.HfError("rolling: need 1 <= min_periods <= window")
        if op not in ("sum", "mean", "count", "min", "max"):
            raise lib.HfError(f"rolling.{op} not supported")
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        if blk_cats:
            raise lib.HfError("rolling over string columns")
        if self._dt_cols():
            raise lib.HfError("rolling over datetime columns (pandas is "
                              "numeric-only here too)")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError("rolling: only RangeIndex frames this round")
        n = len(self)
        names = list(self.columns)
        if n == 0:
            part = HipDataframePartition(DeviceBlock(
                {c: lib.alloc(0, lib.HF_FLOAT64) for c in names}, 0))
            return HipDataframe([part], pandas.RangeIndex(0), names, [0],
                                pandas.Series({c: np.dtype(np.float64)
                                               for c in names}))

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        pos = self._iota(n)
        lo = lib.map_scalar(lib.MAP_MAX,
                            lib.map_scalar(lib.MAP_SUB, pos, w - 1), 0)
        pidx = lib.map_scalar(lib.MAP_MAX,
                              lib.map_scalar(lib.MAP_SUB, lo, 1), 0)
        has_prev = lib.compare_scalar(
            lib.CMP_GE, lib.map_scalar(lib.MAP_SUB, pos, w - 1), 1.0)
        avail_ok = lib.compare_scalar(
            lib.CMP_GE,
            lib.map_scalar(lib.MAP_ADD,
                           lib.binary(lib.BIN_SUB, pos, lo), 1),
            float(minp))
        full = lib.compare_scalar(lib.CMP_GE, pos, float(w - 1))
        pad = -(-n // w) * w
        rev = tile_head = None
        if op in ("min", "max"):
            rev = lib.put(np.arange(pad - 1, -1, -1, dtype=np.int64))
            ones = lib.alloc(pad, lib.HF_INT64)
            lib.fill_i64(ones.dptr(), 1, pad)
            ipad = lib.filter_iota(lib.filter_plan(ones), 0)
            tid = lib.map_scalar(
                lib.MAP_CAST_I64,
                lib.map_scalar(lib.MAP_DIV, ipad, float(w)), 0)
            tile_head = lib.compare_scalar(
                lib.CMP_EQ,
                lib.binary(lib.BIN_SUB, ipad,
                           lib.map_scalar(lib.MAP_MUL, tid, w)), 0.0)
        out_cols = {}
        for c in names:
            vc = lib.cast_f64(concat_col(c))
            m = lib.compare_scalar(lib.CMP_NOTNA, vc, 0.0)
            ccnt = lib.cumsum(m)
            wcnt = lib.binary(
                lib.BIN_SUB, ccnt,
                lib.binary(lib.BIN_MUL, lib.gather(ccnt, pidx), has_prev))
            obs_ok = lib.compare_scalar(lib.CMP_GE, wcnt, float(minp))
            if op == "count":
                res = lib.fixup_empty(lib.cast_f64(wcnt), avail_ok)
            elif op in ("sum", "mean"):
                zf = lib.map_scalar(lib.MAP_FILLNA, vc, 0.0)
                cz = lib.cumsum(zf)
                wsum = lib.binary(
                    lib.BIN_SUB, cz,
                    lib.binary(lib.BIN_MUL, lib.gather(cz, pidx),
                               lib.cast_f64(has_prev)))
                if op == "mean":
                    wsum = lib.binary(lib.BIN_DIV, wsum,
                                      lib.cast_f64(wcnt))
                res = lib.fixup_empty(wsum, obs_ok)
            else:
                ident = float("inf") if op == "min" else float("-inf")
                agg_op = lib.AGG_MIN if op == "min" else lib.AGG_MAX
                bop = lib.BIN_MIN if op == "min" else lib.BIN_MAX
                vfill = lib.map_scalar(lib.MAP_FILLNA, vc, ident)
                if pad > n:
                    tailc = lib.alloc(pad - n, lib.HF_FLOAT64)
                    lib.fill_f64(tailc.dptr(), ident, pad - n)
                    vpad = lib.concat([vfill, tailc])
                else:
                    vpad = vfill
                pref = lib.seg_cumsum(vpad, tile_head, agg_op)
                suff = lib.gather(
                    lib.seg_cumsum(lib.gather(vpad, rev), tile_head,
                                   agg_op), rev)
                # head windows (< w rows) must use the prefix alone:
                # NaN-select the suffix leg there (fmin/fmax skip NaN)
                suff_sel = lib.fixup_empty(lib.gather(suff, lo), full)
                ans = lib.binary(bop, suff_sel,
                                 lib.col_slice(pref, 0, n))
                # min/max need >= 1 observation even at min_periods=0,
                # else the +/-inf scan identity leaks (pandas: NaN)
                mm_ok = obs_ok if minp >= 1 else lib.compare_scalar(
                    lib.CMP_GE, wcnt, 1.0)
                res = lib.fixup_empty(ans, mm_ok)
            out_cols[c] = res
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        return HipDataframe([part], pandas.RangeIndex(n), names, [n],
                            pandas.Series({c: np.dtype(np.float64)
                                           for c in names}))

    def expanding_agg(self, min_periods, op: str) -> "HipDataframe":
        """pandas expanding(min_periods).sum/mean/count/min/max — the
        prefix scans directly (identity-filled so running extremes pass
        NaN rows through), gated like rolling: count on available rows,
        the rest on non-NaN observations."""
        minp = 1 if min_periods is None else int(min_periods)
        if minp < 0:
            raise lib.HfError("expanding: min_periods >= 0")
        if self._dt_cols():
            raise lib.HfError("expanding over datetime columns (pandas "
                              "is numeric-only here too)")
        if op not in ("sum", "mean", "count", "min", "max"):
            raise lib.HfError(f"expanding.{op} not supported")
        if (self._partitions and self._partitions[0].block().cats):
            raise lib.HfError("expanding over string columns")
        n = len(self)
        names = list(self.columns)

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        if n == 0:
            part = HipDataframePartition(DeviceBlock(
                {c: lib.alloc(0, lib.HF_FLOAT64) for c in names}, 0))
            return HipDataframe([part], pandas.RangeIndex(0), names, [0],
                                pandas.Series({c: np.dtype(np.float64)
                                               for c in names}))
        pos = self._iota(n)
        avail_ok = lib.compare_scalar(
            lib.CMP_GE, lib.map_scalar(lib.MAP_ADD, pos, 1), float(minp))
        out_cols = {}
        for c in names:
            vc = lib.cast_f64(concat_col(c))
            m = lib.compare_scalar(lib.CMP_NOTNA, vc, 0.0)
            ccnt = lib.cumsum(m)
            obs_ok = lib.compare_scalar(lib.CMP_GE, ccnt, float(minp))
            if op == "count":
                res = lib.fixup_empty(lib.cast_f64(ccnt), avail_ok)
            elif op in ("sum", "mean"):
                cz = lib.cumsum(lib.map_scalar(lib.MAP_FILLNA, vc, 0.0))
                if op == "mean":
                    cz = lib.binary(lib.BIN_DIV, cz, lib.cast_f64(ccnt))
                res = lib.fixup_empty(cz, obs_ok)
            else:
                ident = float("inf") if op == "min" else float("-inf")
                agg_op = lib.AGG_MIN if op == "min" else lib.AGG_MAX
                run = lib.cumsum(
                    lib.map_scalar(lib.MAP_FILLNA, vc, ident), agg_op)
                # min/max need >= 1 observation even at min_periods=0
                mm_ok = obs_ok if minp >= 1 else lib.compare_scalar(
                    lib.CMP_GE, ccnt, 1.0)
                res = lib.fixup_empty(run, mm_ok)
            out_cols[c] = res
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        return HipDataframe([part], self._index, names, [n],
                            pandas.Series({c: np.dtype(np.float64)
                                           for c in names}))

    def dt_field(self, field: str) -> "HipDataframe":
        """Series.dt.<field> over datetime64[ns] typed columns: exact
        int64 calendar math on the ns view (Howard Hinnant's civil-from-
        days algorithm composed from MAP_IDIV/IMOD + compares — f64 would
        round modern-era ns values).  Reference surface: pandas
        Series.dt accessors (the reference delegates to pandas)."""
        fields = ("year", "month", "day", "hour", "minute", "second",
                  "dayofweek")
        if field not in fields:
            raise lib.HfError(f"dt.{field} not supported ({fields})")
        name = self.columns[0]
        if not (isinstance(self.dtypes[name], np.dtype)
                and np.issubdtype(self.dtypes[name], np.datetime64)):
            raise lib.HfError(
                f"dt accessor on non-datetime column {name!r}")

        def concat_col():
            cs = [p.block().columns[name] for p in self._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        ns = concat_col()
        n = ns.length
        DAY = 86_400 * 10**9

        def idiv(c, k):
            return lib.map_scalar(lib.MAP_IDIV, c, k)

        def imod(c, k):
            return lib.map_scalar(lib.MAP_IMOD, c, k)

        def addc(c, k):
            return lib.map_scalar(lib.MAP_ADD, c, k)

        def mulc(c, k):
            return lib.map_scalar(lib.MAP_MUL, c, k)

        def sub(a, b):
            return lib.binary(lib.BIN_SUB, a, b)

        if field == "hour":
            out = idiv(imod(ns, DAY), 3_600 * 10**9)
        elif field == "minute":
            out = idiv(imod(ns, 3_600 * 10**9), 60 * 10**9)
        elif field == "second":
            out = idiv(imod(ns, 60 * 10**9), 10**9)
        elif field == "dayofweek":
            out = imod(addc(idiv(ns, DAY), 3), 7)  # 1970-01-01 is a Thu
        else:
            days = idiv(ns, DAY)
            z = addc(days, 719_468)
            era = idiv(z, 146_097)
            doe = sub(z, mulc(era, 146_097))           # [0, 146096]
            # yoe = (doe - doe/1460 + doe/36524 - doe/146096) / 365
            yoe = idiv(sub(lib.binary(lib.BIN_ADD,
                                      sub(doe, idiv(doe, 1460)),
                                      idiv(doe, 36_524)),
                           idiv(doe, 146_096)), 365)
            doy = sub(doe, sub(lib.binary(lib.BIN_ADD, mulc(yoe, 365),
                                          idiv(yoe, 4)),
                               idiv(yoe, 100)))
            mp = idiv(addc(mulc(doy, 5), 2), 153)
            if field == "day":
                out = addc(sub(doy, idiv(addc(mulc(mp, 153), 2), 5)), 1)
            else:
                # m = mp + 3 - 12*(mp >= 10)
                ge10 = lib.compare_scalar(lib.CMP_GE, mp, 10.0)
                m = sub(addc(mp, 3), mulc(ge10, 12))
                if field == "month":
                    out = m
                else:  # year = yoe + era*400 + (m <= 2)
                    le2 = lib.compare_scalar(lib.CMP_LE, m, 2.0)
                    out = lib.binary(
                        lib.BIN_ADD,
                        lib.binary(lib.BIN_ADD, yoe, mulc(era, 400)), le2)
        if n and lib.reduce(ns).imn == INAT:
            # NaT rows -> NaN, float64 result (the pandas dtype rule when
            # NaT is present)
            notnat = lib.compare_scalar(lib.CMP_NE, ns, float(INAT))
            out = lib.fixup_empty(lib.cast_f64(out), notnat)
            part = HipDataframePartition(DeviceBlock({name: out}, n))
            return HipDataframe([part], self._index, [name], [n],
                                pandas.Series({name: np.dtype(np.float64)}))
        part = HipDataframePartition(DeviceBlock({name: out}, n))
        # pandas dt fields are int32
        return HipDataframe([part], self._index, [name], [n],
                            pandas.Series({name: np.dtype(np.int32)}))

    def rank_rows(self, ascending: bool = True, method: str = "average",
                  na_option: str = "keep") -> "HipDataframe":
        """Frame-level pandas rank(axis=0) over one constant-key group."""
        return self._with_const_key().groupby_transform(
            self.KEYCOL, "rank", ascending=ascending, method=method,
            na_option=na_option)

    def fill_rows(self, how: str) -> "HipDataframe":
        """Frame-level pandas ffill/bfill over one constant-key group."""
        return self._with_const_key().groupby_transform(self.KEYCOL, how)

    def groupby_prod(self, by) -> "HipDataframe":
        """groupby().prod(): sort by key, segmented PRODUCT scan
        (HF_AGG_PROD), take each run's value at its last non-NaN row
        (pandas skipna; all-NaN groups give 1.0 — min_count=0).  int64
        columns stay int64 with pandas' wrapping product."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            key = by if isinstance(by, str) else \
                (by[0] if len(by) == 1 else None)
            if key is None:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_prod(self.KEYCOL)
                res._index = decode(np.asarray(res.index).astype(np.int64))
                return res
            shuf = self._shuffle_frame_by_key(key)
            with dist_mod.local_mode():
                res = shuf.groupby_prod(key)
            return self._replicate_result_frame(res)
        if isinstance(by, (list, tuple)):
            if len(by) == 1:
                by = by[0]
            else:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_prod(self.KEYCOL)
                res._index = decode(lib.get(res._index.col))
                return res
        val_names = [c for c in self.columns if c != by]
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        bad = [v for v in val_names if v in blk_cats]
        if bad:
            raise lib.HfError(f"groupby prod over string column(s) {bad}")

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        n = len(self)
        res0 = self.groupby_size(by)
        ngv = len(res0)
        if n == 0 or ngv == 0:
            part = HipDataframePartition(DeviceBlock(
                {v: lib.alloc(0, lib.HF_FLOAT64) for v in val_names}, 0))
            return HipDataframe([part], res0._index, val_names, [0],
                                pandas.Series({v: np.dtype(np.float64)
                                               for v in val_names}))
        ekey, _ = self._effective_sort_key(concat_col(by), by in blk_cats,
                                           True)
        perm = self._compose_sort_perm([(ekey, True)])
        khead = self._run_head_col(lib.gather(ekey, perm), n)
        kplan = lib.filter_plan(khead)
        ng = kplan.n_kept
        hp = lib.filter_iota(kplan, 0)
        ends = lib.map_scalar(
            lib.MAP_SUB,
            lib.concat([lib.col_slice(hp, 1, ng - 1),
                        self._const_i64(n)]) if ng > 1
            else self._const_i64(n), 1)
        out_cols, dts = {}, {}
        for v in val_names:
            vc = concat_col(v)
            sv = lib.gather(vc, perm)
            seg = lib.seg_cumsum(sv, khead, lib.AGG_PROD)
            if vc.dtype_code == lib.HF_INT64:
                res = lib.col_slice(lib.gather(seg, ends), 0, ngv)
                dts[v] = np.dtype(np.int64)
            else:
                m = lib.compare_scalar(lib.CMP_NOTNA, sv, 0.0)
                segc = lib.seg_cumsum(m, khead, lib.AGG_SUM)
                cnt = lib.gather(segc, ends)
                pos = self._iota(n)
                posv = lib.binary(
                    lib.BIN_ADD, lib.binary(lib.BIN_MUL, pos, m),
                    lib.map_scalar(lib.MAP_SUB, m, 1))
                segmax = lib.seg_cumsum(posv, khead, lib.AGG_MAX)
                lastp = lib.map_scalar(lib.MAP_MAX,
                                       lib.gather(segmax, ends), 0)
                resf = lib.fixup_empty(lib.gather(seg, lastp), cnt)
                resf = lib.map_scalar(lib.MAP_FILLNA, resf, 1.0)
                res = lib.col_slice(resf, 0, ngv)
                dts[v] = np.dtype(np.float64)
            out_cols[v] = res
        part = HipDataframePartition(DeviceBlock(out_cols, ngv))
        return HipDataframe([part], res0._index, val_names, [ngv],
                            pandas.Series(dts))

    def groupby_idxminmax(self, by, maximum: bool) -> "HipDataframe":
        """groupby.idxmax/idxmin: per group and value column, the ORIGINAL
        row position (RangeIndex label) of the first occurrence of the
        extreme; all-NaN groups report NaN (float64 column, pandas 2.3).
        Composition: stable sort by (key, value) — the extreme's tie block
        start IS the first original occurrence; non-NaN counts locate the
        last non-NaN row (values sort NaN-last within each key run)."""
        from .. import distributed as dist_mod
        if dist_mod.is_active():
            key = by if isinstance(by, str) else \
                (by[0] if len(by) == 1 else None)
            if key is None:
                cf, decode = self._combined_key_frame(list(by))
                keep = [c for c in cf.columns if c not in by]
                res = cf.take_columns(keep).groupby_idxminmax(
                    self.KEYCOL, maximum)
                res._index = decode(np.asarray(res.index).astype(np.int64))
                return res
            # shuffle WITH global positions: local row order after the
            # exchange is global-position-ascending, so the local pick is
            # the right ROW; its label maps through the position column
            shuf, pos_col = self._shuffle_frame_by_key(key, with_pos=True)
            with dist_mod.local_mode():
                res = shuf.groupby_idxminmax(key, maximum)
            blk = res._partitions[0].block()
            cols2 = {}
            dts2 = {}
            for nm, col in blk.columns.items():
                if col.dtype_code == lib.HF_FLOAT64:
                    ok = lib.compare_scalar(lib.CMP_NOTNA, col, 0.0)
                    idx_i = lib.map_scalar(
                        lib.MAP_CAST_I64,
                        lib.map_scalar(lib.MAP_FILLNA, col, 0.0), 0)
                    mapped = lib.gather(pos_col, idx_i)
                    cols2[nm] = lib.fixup_empty(lib.cast_f64(mapped), ok)
                    dts2[nm] = np.dtype(np.float64)
                else:
                    cols2[nm] = lib.gather(pos_col, col)
                    dts2[nm] = np.dtype(np.int64)
            res = HipDataframe(
                [HipDataframePartition(DeviceBlock(cols2, blk.length))],
                res._index, list(res.columns), [blk.length],
                pandas.Series(dts2))
            return self._replicate_result_frame(res)
        by_list = [by] if isinstance(by, str) else list(by)
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        val_names = [c for c in self.columns if c not in by_list]
        for v in val_names:
            if v in blk_cats:
                raise lib.HfError("idxmax/idxmin over string columns")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError("groupby idxmax/idxmin: only RangeIndex "
                              "frames this round")

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        n = len(self)
        res0 = self.groupby_size(by)      # group-key index, NaN dropped
        ngv = len(res0)
        out_cols, dts = {}, {}
        if n == 0 or ngv == 0:
            part = HipDataframePartition(DeviceBlock(
                {v: lib.alloc(0, lib.HF_INT64) for v in val_names}, 0))
            return HipDataframe([part], res0._index, val_names, [0],
                                pandas.Series({v: np.dtype(np.int64)
                                               for v in val_names}))
        posmap = None
        if len(by_list) > 1:
            # the trailing-sentinel slicing below assumes NaN-key rows
            # sort last, which only holds for one key column — multi-key
            # NaN keys FILTER FIRST instead (the pinned prototype,
            # test_host_logic.py:330), with original positions restored
            # through the kept-row map afterwards
            nan_masks = []
            for b in by_list:
                c = concat_col(b)
                if not c.length:
                    continue
                if b in blk_cats:
                    if lib.reduce(c).imn < 0:
                        nan_masks.append(
                            lib.compare_scalar(lib.CMP_GE, c, 0.0))
                elif (c.dtype_code == lib.HF_FLOAT64
                      and lib.reduce(c).count != c.length):
                    nan_masks.append(
                        lib.compare_scalar(lib.CMP_NOTNA, c, 0.0))
            if nan_masks:
                acc = nan_masks[0]
                for m2 in nan_masks[1:]:
                    acc = lib.binary(lib.BIN_MUL, acc, m2)
                plan = lib.filter_plan(acc)
                posmap = lib.filter_iota(plan, 0)
                fcols = {name: lib.filter_apply(plan, concat_col(name))
                         for name in list(by_list) + list(val_names)}
                n = plan.n_kept

                def concat_col(name, _f=fcols):  # noqa: F811
                    return _f[name]
        eff_keys = [self._effective_sort_key(concat_col(b), b in blk_cats,
                                             True)
                    for b in by_list]
        for v in val_names:
            vc = concat_col(v)
            eff_v = self._effective_sort_key(vc, False, True)
            perm = self._compose_sort_perm(eff_keys + [eff_v])
            khead = None
            for ekc, _ in eff_keys:
                h = self._run_head_col(lib.gather(ekc, perm), n)
                khead = h if khead is None else lib.binary(lib.BIN_ADD,
                                                           khead, h)
            if len(eff_keys) > 1:
                khead = lib.compare_scalar(lib.CMP_GE, khead, 1.0)
            kplan = lib.filter_plan(khead)
            ng = kplan.n_kept
            hp = lib.filter_iota(kplan, 0)
            # non-NaN count per run (values sort NaN-last within the run)
            sv = lib.gather(vc, perm)
            if vc.dtype_code == lib.HF_FLOAT64:
                m = lib.compare_scalar(lib.CMP_NOTNA, sv, 0.0)
            else:
                m = lib.alloc(n, lib.HF_INT64)
                lib.fill_i64(m.dptr(), 1, n)
            segc = lib.seg_cumsum(m, khead, lib.AGG_SUM)
            ends = lib.concat([lib.col_slice(hp, 1, ng - 1),
                               self._const_i64(n)]) if ng > 1 \
                else self._const_i64(n)
            cnt = lib.gather(segc, lib.map_scalar(lib.MAP_SUB, ends, 1))
            if maximum:
                # last non-NaN row of the run, then its tie-block start
                pos = lib.binary(
                    lib.BIN_ADD, hp,
                    lib.map_scalar(lib.MAP_MAX,
                                   lib.map_scalar(lib.MAP_SUB, cnt, 1),
                                   0))
                vhead = self._run_head_col(lib.gather(eff_v[0], perm), n)
                thead = lib.compare_scalar(
                    lib.CMP_GE, lib.binary(lib.BIN_ADD, khead, vhead),
                    1.0)
                trid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(thead), 1)
                tplan = lib.filter_plan(thead)
                t_hp = lib.filter_iota(tplan, 0)
                tstart_row = lib.gather(t_hp, trid)   # [n]
                ts = lib.gather(tstart_row, pos)      # [ng]
            else:
                ts = hp  # ascending sort: the min's first occurrence
            orig = lib.gather(perm, ts)               # [ng]
            if posmap is not None:
                orig = lib.gather(posmap, orig)  # back to original rows
            # keep only the valid (non-NaN-key) groups: the sentinel runs
            # sort last, so they are the trailing ng - ngv runs
            orig = lib.col_slice(orig, 0, ngv)
            cntv = lib.col_slice(cnt, 0, ngv)
            n_empty = ngv - int(lib.reduce(
                lib.compare_scalar(lib.CMP_GE, lib.cast_f64(cntv),
                                   1.0)).isum)
            if n_empty:
                orig = lib.fixup_empty(
                    lib.cast_f64(orig),
                    lib.compare_scalar(lib.CMP_GE, lib.cast_f64(cntv),
                                       1.0))
                dts[v] = np.dtype(np.float64)
            else:
                dts[v] = np.dtype(np.int64)
            out_cols[v] = orig
        part = HipDataframePartition(DeviceBlock(out_cols, ngv))
        return HipDataframe([part], res0._index, val_names, [ngv],
                            pandas.Series(dts))

    def _groupby_rank(self, by_list, val_names, eff_keys, valid, n,
                      concat_col, ascending, method, na_option="keep"):
        """rank within groups (pandas DataFrameGroupBy.rank): per value
        column, sort by (keys…, value) with the effective-key transform,
        1-based position within the key run, tie runs collapsed per
        `method` ('average' -> first + (len-1)/2, 'min' -> first,
        'first' -> the position itself).  na_option: 'keep' (NaN -> NaN),
        'top'/'bottom' (NaN values rank lowest/highest — the NaN sentinel
        sorts first/last and forms ONE tie run, so the normal tie
        arithmetic produces pandas' shared NaN ranks)."""
        out_cols = {}
        for v in val_names:
            vc = concat_col(v)
            eff_v = self._effective_sort_key(vc, False, ascending,
                                             na_first=(na_option == "top"))
            perm = self._compose_sort_perm(eff_keys + [eff_v])
            khead = None
            for ekc, _ in eff_keys:
                h = self._run_head_col(lib.gather(ekc, perm), n)
                khead = h if khead is None else lib.binary(lib.BIN_ADD,
                                                           khead, h)
            if len(eff_keys) > 1:
                khead = lib.compare_scalar(lib.CMP_GE, khead, 1.0)
            kplan = lib.filter_plan(khead)
            hp = lib.filter_iota(kplan, 0)
            rid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(khead), 1)
            start = lib.gather(hp, rid)
            pos1 = lib.map_scalar(
                lib.MAP_ADD,
                lib.binary(lib.BIN_SUB, self._iota(n), start), 1)
            if method == "first":
                rank_s = lib.cast_f64(pos1)
            else:
                vhead = self._run_head_col(lib.gather(eff_v[0], perm), n)
                thead = lib.compare_scalar(
                    lib.CMP_GE, lib.binary(lib.BIN_ADD, khead, vhead), 1.0)
                tplan = lib.filter_plan(thead)
                nt = tplan.n_kept
                tstarts = lib.filter_iota(tplan, 0)
                first_pos = lib.filter_apply(tplan, pos1)
                if method == "average":
                    nxt = lib.concat([lib.col_slice(tstarts, 1, nt - 1),
                                      self._const_i64(n)]) if nt > 1 \
                        else self._const_i64(n)
                    tlen = lib.binary(lib.BIN_SUB, nxt, tstarts)
                    avg = lib.binary(
                        lib.BIN_ADD, lib.cast_f64(first_pos),
                        lib.map_scalar(
                            lib.MAP_DIV,
                            lib.cast_f64(lib.map_scalar(lib.MAP_SUB,
                                                        tlen, 1)), 2.0))
                else:
                    avg = lib.cast_f64(first_pos)
                trid = lib.map_scalar(lib.MAP_SUB, lib.cumsum(thead), 1)
                rank_s = lib.gather(avg, trid)
            res = lib.scatter(rank_s, perm)
            if na_option == "keep" and vc.dtype_code == lib.HF_FLOAT64:
                notna_v = lib.compare_scalar(lib.CMP_NOTNA, vc, 0.0)
                res = lib.fixup_empty(res, notna_v)
            if valid is not None:
                res = lib.fixup_empty(res, valid)
            out_cols[v] = res
        part = HipDataframePartition(DeviceBlock(out_cols, n))
        dts = pandas.Series({v: np.dtype(np.float64) for v in val_names})
        return HipDataframe([part], pandas.RangeIndex(n), val_names, [n],
                            dts)

    def where_rows(self, mask_frame: "HipDataframe",
                   other=None) -> "HipDataframe":
        """pandas where(cond, other): keep values where the row mask is
        true, else `other` (NaN default).  Exact NaN bookkeeping: an
        original NaN under a TRUE cond stays NaN even when `other` fills
        the false rows."""
        mcols = []
        for p in mask_frame._partitions:
            b = p.block()
            mcols.append(b.columns[list(b.columns)[0]])
        m = mcols[0] if len(mcols) == 1 else lib.concat(mcols)
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        n = len(self)
        if m.length != n:
            raise lib.HfError("where/mask: condition length mismatch")
        # pandas only upcasts int64 when a fill actually happens: an
        # all-true cond keeps the column untouched (one cached reduce)
        if other is not None and isinstance(other, float) \
                and np.isnan(other):
            other = None  # NaN fill == the default fill
        all_true = n == 0 or lib.reduce(m).imn >= 1
        dtc = self._dt_cols()
        if dtc and other is not None and not all_true:
            import datetime as _dtm
            if not isinstance(other, (pandas.Timestamp, np.datetime64,
                                      _dtm.datetime)):
                raise lib.HfError(
                    "where/mask over datetime columns: the fill must be "
                    "a Timestamp (or None -> NaT)")
        out_cols, dts, cats = {}, {}, {}
        for c in self.columns:
            col = concat_col(c)
            if c in dtc and not all_true:
                # datetime: false rows fill NaT (or the Timestamp's ns);
                # dtype kept — an int64 blend, no f64 round trip
                fill = (INAT if other is None
                        else int(pandas.Timestamp(other).value))
                inv = lib.map_scalar(lib.MAP_RSUB, m, 1)
                out_cols[c] = lib.binary(
                    lib.BIN_ADD, lib.binary(lib.BIN_MUL, col, m),
                    lib.map_scalar(lib.MAP_MUL, inv, fill))
                dts[c] = self.dtypes[c]
                continue
            if c in dtc and other is not None:
                # all-true: values untouched, dtype kept
                out_cols[c] = col
                dts[c] = self.dtypes[c]
                continue
            if all_true:
                out_cols[c] = col
                dts[c] = self.dtypes[c]
                if c in blk_cats:
                    cats[c] = blk_cats[c]
                continue
            if c in blk_cats:
                if other is not None and not isinstance(other, str):
                    raise lib.HfError("where/mask over string columns: "
                                      "the fill value must be a string")
                ccats = blk_cats[c]
                if other is None:
                    # codes: (code+1)*m - 1 -> untouched codes / −1 (NaN)
                    t = lib.map_scalar(
                        lib.MAP_SUB,
                        lib.binary(lib.BIN_MUL,
                                   lib.map_scalar(lib.MAP_ADD, col, 1),
                                   m), 1)
                else:
                    # string fill: union the fill value into the
                    # dictionary (host), recode, then blend codes:
                    # (code+1)*m + (fc+1)*(1-m) - 1
                    ucats = union_cats(ccats, pandas.Index([other]))
                    if not ucats.equals(ccats):
                        col = recode_dict_col(col, ccats, ucats)
                        ccats = ucats
                    fc = int(ccats.get_loc(other))
                    keep = lib.binary(
                        lib.BIN_MUL,
                        lib.map_scalar(lib.MAP_ADD, col, 1), m)
                    fill = lib.map_scalar(
                        lib.MAP_MUL, lib.map_scalar(lib.MAP_RSUB, m, 1),
                        fc + 1)
                    t = lib.map_scalar(
                        lib.MAP_SUB, lib.binary(lib.BIN_ADD, keep, fill),
                        1)
                out_cols[c] = t
                dts[c] = self.dtypes[c]
                cats[c] = ccats
            elif (col.dtype_code == lib.HF_INT64 and other is not None
                    and isinstance(other, (int, np.integer))):
                t1 = lib.binary(lib.BIN_MUL, col, m)
                t2 = lib.map_scalar(
                    lib.MAP_MUL, lib.map_scalar(lib.MAP_RSUB, m, 1),
                    int(other))
                out_cols[c] = lib.binary(lib.BIN_ADD, t1, t2)
                dts[c] = np.dtype(np.int64)
            else:
                if other is not None and not isinstance(
                        other, (int, float, np.integer, np.floating)):
                    raise lib.HfError(
                        f"where/mask: fill {type(other).__name__} over "
                        f"numeric column {c!r} (pandas would upcast to "
                        "object)")
                cf = lib.cast_f64(col)
                t = lib.fixup_empty(cf, m)
                if other is not None:
                    isna = lib.map_scalar(
                        lib.MAP_RSUB,
                        lib.compare_scalar(lib.CMP_NOTNA, cf, 0.0), 1)
                    nan_keep = lib.binary(lib.BIN_MUL, isna, m)
                    t = lib.map_scalar(lib.MAP_FILLNA, t, float(other))
                    t = lib.fixup_empty(
                        t, lib.map_scalar(lib.MAP_RSUB, nan_keep, 1))
                out_cols[c] = t
                dts[c] = np.dtype(np.float64)
        part = HipDataframePartition(DeviceBlock(out_cols, n, cats))
        return HipDataframe([part], self._index, list(self.columns), [n],
                            pandas.Series(dts))

    def round_cols(self, decimals: int = 0) -> "HipDataframe":
        """pandas round(decimals): half-even via rint(x*10^d)/10^d (the
        numpy scaling rule); int columns unchanged."""
        scale = float(10.0 ** int(decimals))

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            out = {}
            for name, col in block.columns.items():
                if name in block.cats:
                    out[name] = col
                elif col.dtype_code == lib.HF_FLOAT64:
                    out[name] = lib.map_scalar(lib.MAP_ROUND, col, scale)
                elif decimals < 0:
                    # pandas rounds int columns too for negative decimals
                    # (to tens/hundreds/…), keeping int64
                    out[name] = lib.map_scalar(
                        lib.MAP_CAST_I64,
                        lib.map_scalar(lib.MAP_ROUND, lib.cast_f64(col),
                                       scale), 0)
                else:
                    out[name] = col
            return DeviceBlock(out, block.length, block.cats)
        return self.map(block_fn)

    def hconcat(self, others: list) -> "HipDataframe":
        """Horizontal compose of single-partition frames with identical row
        count and index (the device form of the reference's axis=1 concat
        over aligned frames — used by agg-list/dict composition)."""
        frames = [self] + list(others)
        n = len(self)
        cols, dtypes, cats = {}, {}, {}
        for f in frames:
            if len(f) != n or len(f._partitions) != 1:
                raise lib.HfError("hconcat: frames must be aligned "
                                  "single-partition results")
            block = f._partitions[0].block()
            for name, c in block.columns.items():
                if name in cols:
                    raise lib.HfError(f"hconcat: duplicate column {name!r}")
                cols[name] = c
                dtypes[name] = f.dtypes[name]
                if name in block.cats:
                    cats[name] = block.cats[name]
        part = HipDataframePartition(DeviceBlock(cols, n, cats))
        return HipDataframe([part], self._index, list(cols), [n],
                            pandas.Series(dtypes))

    def alias_column(self, name: str, new_name: str) -> "HipDataframe":
        """Append ``new_name`` as a second reference to column ``name`` —
        zero-copy (ColumnRef is refcounted, so both names share the device
        buffer).  Used by the left_on/right_on merge rewrite."""
        if new_name in self.columns:
            raise lib.HfError(f"alias_column: {new_name!r} exists")
        parts = []
        for p, ln in zip(self._partitions, self._row_lengths):
            b = p.block()
            cols = dict(b.columns)
            cols[new_name] = b.columns[name]
            cats = dict(b.cats)
            if name in cats:
                cats[new_name] = cats[name]
            parts.append(HipDataframePartition(DeviceBlock(cols, ln, cats)))
        dtypes = self.dtypes.copy()
        dtypes[new_name] = self.dtypes[name]
        return HipDataframe(parts, self._index,
                            list(self.columns) + [new_name],
                            self._row_lengths, dtypes)

    def set_column(self, name: str, other: "HipDataframe") -> "HipDataframe":
        """Replace-or-append column ``name`` with ``other``'s single column
        (POSITIONAL assignment — the device form of the reference's
        `__setitem__`/`insert` path, dataframe.py:2575 setitem_builder;
        index alignment beyond equal length is the caller's concern).
        ``other`` re-slices device-side to this frame's partitioning."""
        if len(other) != len(self):
            raise lib.HfError(
                f"set_column: length mismatch ({len(other)} vs {len(self)})")
        if len(other.columns) != 1:
            raise lib.HfError("set_column: value must be a single column")
        src = other.columns[0]
        other = other.repartition_like(self._row_lengths)
        parts = []
        for p, q, ln in zip(self._partitions, other._partitions,
                            self._row_lengths):
            b, ob = p.block(), q.block()
            cols = dict(b.columns)
            cols[name] = ob.columns[src]
            cats = dict(b.cats)
            cats.pop(name, None)
            if src in ob.cats:
                cats[name] = ob.cats[src]
            parts.append(HipDataframePartition(DeviceBlock(cols, ln, cats)))
        columns = list(self.columns)
        if name not in columns:
            columns = columns + [name]
        dtypes = self.dtypes.copy()
        dtypes[name] = other.dtypes[src]
        return HipDataframe(parts, self._index, columns,
                            self._row_lengths, dtypes)

    def set_scalar_column(self, name: str, value) -> "HipDataframe":
        """Broadcast-scalar column assignment: df[name] = scalar.  Fills
        device-side (no host array); strings become a one-entry
        dictionary column; bools/ints land as int64 (our bool carrier),
        floats/NaN/None as float64."""
        parts, cats_add = [], None
        if isinstance(value, str):
            cats_add = pandas.Index([value])

        def mk(ln):
            if isinstance(value, str):
                c = lib.alloc(ln, lib.HF_INT64)
                lib.fill_i64(c.dptr(), 0, ln)
                return c, np.dtype(object)
            if value is None or (isinstance(value, float)
                                 and np.isnan(value)):
                c = lib.alloc(ln, lib.HF_FLOAT64)
                lib.fill_f64(c.dptr(), float("nan"), ln)
                return c, np.dtype(np.float64)
            if isinstance(value, (bool, np.bool_, int, np.integer)):
                c = lib.alloc(ln, lib.HF_INT64)
                lib.fill_i64(c.dptr(), int(value), ln)
                return c, np.dtype(np.int64)
            c = lib.alloc(ln, lib.HF_FLOAT64)
            lib.fill_f64(c.dptr(), float(value), ln)
            return c, np.dtype(np.float64)

        if isinstance(value, str):
            dt = np.dtype(object)
        elif isinstance(value, (bool, np.bool_, int, np.integer)):
            dt = np.dtype(np.int64)
        else:
            dt = np.dtype(np.float64)
        for p, ln in zip(self._partitions, self._row_lengths):
            b = p.block()
            cols = dict(b.columns)
            cols[name], dt = mk(ln)
            cats = dict(b.cats)
            cats.pop(name, None)
            if cats_add is not None:
                cats[name] = cats_add
            parts.append(HipDataframePartition(DeviceBlock(cols, ln, cats)))
        columns = list(self.columns)
        if name not in columns:
            columns = columns + [name]
        dtypes = self.dtypes.copy()
        dtypes[name] = dt
        return HipDataframe(parts, self._index, columns,
                            self._row_lengths, dtypes)

    def reverse_rows(self) -> "HipDataframe":
        """Row reversal, device-side: reversed iota (cumsum of ones,
        RSUB n) + one gather per column.  Used by duplicated(keep=
        'last') — pandas computes keep='last' as keep='first' over the
        reversed rows."""
        from ..distributed import is_active, world_size
        if is_active() and world_size() > 1:
            raise lib.HfError("duplicated(keep='last'/False) at world>1 "
                              "is a later round")
        n = len(self)

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        ones = lib.alloc(n, lib.HF_INT64)
        if n:
            lib.fill_i64(ones.dptr(), 1, n)
        perm = lib.map_scalar(lib.MAP_RSUB, lib.cumsum(ones, lib.AGG_SUM),
                              n)  # n-1, n-2, …, 0
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        cols = {c: lib.gather(concat_col(c), perm) for c in self.columns}
        idx = pandas.Index(np.asarray(self.index)[::-1])
        part = HipDataframePartition(DeviceBlock(cols, n, dict(blk_cats)))
        return HipDataframe([part], idx, list(self.columns), [n],
                            self.dtypes.copy())

    def sample_rows(self, n: int, seed: int) -> "HipDataframe":
        """Uniform sample WITHOUT replacement, fully device-side: one
        splitmix64 uniform key per row (hf_fill_randf64), the n smallest
        keys win (hf_sort_perm + gather) — the draw never leaves HBM;
        only the n selected index labels come host-side.  Row order is
        the random draw order (pandas sample also permutes)."""
        total = len(self)
        if not 0 <= n <= total:
            raise lib.HfError(f"sample: n={n} out of range 0..{total}")
        from ..distributed import is_active, world_size
        if is_active() and world_size() > 1:
            raise lib.HfError("sample at world>1 is a later round")

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        keys = lib.ordered_i64(lib.fill_randf64(total, seed))
        perm = lib.col_slice(lib.sort_perm(keys), 0, n)
        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        out_cols = {c: lib.gather(concat_col(c), perm)
                    for c in self.columns}
        idx = pandas.Index(np.asarray(self.index))[lib.get(perm)]
        part = HipDataframePartition(
            DeviceBlock(out_cols, n, dict(blk_cats)))
        return HipDataframe([part], idx, list(self.columns), [n],
                            self.dtypes.copy())

    # ---- broadcast inner join (MergeImpl.row_axis_merge device form,
    #      merge.py:104-178: combine() the right frame once, probe per left
    #      partition; pandas suffix rules "_x"/"_y" on collisions) ----
    def cross_join(self, other: "HipDataframe") -> "HipDataframe":
        """pandas merge(how='cross'): the nl x nr cartesian product —
        device gather through hf_cross_idx row indices; '_x'/'_y' suffixes
        on every colliding column name (no key column to exempt).
        Reference: MergeImpl/pandas cross merge (merge.py)."""
        from ..distributed import is_active
        if is_active():
            raise lib.HfError("distributed cross merge is a later round")
        nl, nr = len(self), len(other)
        if nl * max(nr, 1) > (1 << 31):
            raise lib.HfError("cross merge result exceeds 2^31 rows")

        def concat_col(frame, name):
            cols = [p.block().columns[name] for p in frame._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        lcats = (self._partitions[0].block().cats
                 if self._partitions else {})
        rcats = (other._partitions[0].block().cats
                 if other._partitions else {})
        common = set(self.columns) & set(other.columns)
        lout = {n2: (n2 + "_x" if n2 in common else n2)
                for n2 in self.columns}
        rout = {n2: (n2 + "_y" if n2 in common else n2)
                for n2 in other.columns}
        n = nl * nr
        if nr == 0 or nl == 0:
            cols = {}
            dts = {}
            cats = {}
            for c in self.columns:
                code = (lib.HF_FLOAT64 if self.dtypes[c] == np.float64
                        else lib.HF_INT64)
                cols[lout[c]] = lib.alloc(0, code)
                dts[lout[c]] = self.dtypes[c]
                if c in lcats:
                    cats[lout[c]] = lcats[c]
            for c in other.columns:
                code = (lib.HF_FLOAT64 if other.dtypes[c] == np.float64
                        else lib.HF_INT64)
                cols[rout[c]] = lib.alloc(0, code)
                dts[rout[c]] = other.dtypes[c]
                if c in rcats:
                    cats[rout[c]] = rcats[c]
            part = HipDataframePartition(DeviceBlock(cols, 0, cats))
            names = list(cols)
            return HipDataframe([part], pandas.RangeIndex(0), names, [0],
                                pandas.Series(dts))
        lidx, ridx = lib.cross_idx(nl, nr)
        cols, cats = {}, {}
        for c in self.columns:
            cols[lout[c]] = lib.gather(concat_col(self, c), lidx)
            if c in lcats:
                cats[lout[c]] = lcats[c]
        for c in other.columns:
            cols[rout[c]] = lib.gather(concat_col(other, c), ridx)
            if c in rcats:
                cats[rout[c]] = rcats[c]
        names = [lout[c] for c in self.columns] + \
            [rout[c] for c in other.columns]
        dts = pandas.Series(
            {**{lout[c]: self.dtypes[c] for c in self.columns},
             **{rout[c]: other.dtypes[c] for c in other.columns}})
        part = HipDataframePartition(DeviceBlock(cols, n, cats))
        return HipDataframe([part], pandas.RangeIndex(n), names, [n], dts)

    def merge_multi(self, other: "HipDataframe", on: list,
                    how: str = "inner") -> "HipDataframe":
        """Multi-key merge: fold the key columns of BOTH sides into ONE
        int64 key with SHARED mins/strides (global over the union, so the
        fold is consistent cross-frame; float keys ride the ordered
        canonical-NaN transform first — NaN tuples match like pandas;
        string keys recode into the left dictionary).  The right frame
        drops its key columns (pandas on=[...] emits one set, from the
        left), then the single-key engine runs on the fold and the
        combined column drops from the result."""
        from .. import distributed as dist_mod
        KEY = "\x00mkey\x00"
        lcats = (self._partitions[0].block().cats
                 if self._partitions else {})
        rcats = (other._partitions[0].block().cats
                 if other._partitions else {})

        def concat_col(frame, name):
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        mins, spans, enc_l, enc_r = [], [], [], []
        for b in on:
            if b not in self.columns or b not in other.columns:
                raise lib.HfError(f"merge: key column {b!r} missing")
            if (b in lcats) != (b in rcats):
                raise lib.HfError("merge: key column is a string on one "
                                  "side only")
            lc, rc = concat_col(self, b), concat_col(other, b)
            if b in lcats:
                if not rcats[b].equals(lcats[b]):
                    rc = recode_dict_col(rc, rcats[b], lcats[b],
                                         missing=-2)
                # shift codes so -2/-1 fold like ordinary keys
                lo = -2
                lc2, rc2 = lc, rc
            elif (self.dtypes[b] == np.dtype(np.float64)
                  or other.dtypes[b] == np.dtype(np.float64)):
                lc2 = lib.ordered_i64(lib.cast_f64(lc))
                rc2 = lib.ordered_i64(lib.cast_f64(rc))
                lo = None
            else:
                lc2, rc2 = lc, rc
                lo = None
            mn = mx = None
            for c in (lc2, rc2):
                if c.length:
                    r = lib.reduce(c)
                    mn = r.imn if mn is None else min(mn, r.imn)
                    mx = r.imx if mx is None else max(mx, r.imx)
            if dist_mod.is_active():
                mn, mx = dist_mod.allreduce_minmax(mn, mx)
            if mn is None:
                mn, mx = 0, 0
            if lo is not None:
                mn = min(mn, lo)
            if mx - mn + 1 > (1 << 31) and not dist_mod.is_active():
                # wide span (float ordered keys span ~2^63): densify both
                # sides through the sorted distinct UNION, fold on codes
                both = lib.concat([lc2, rc2]) if rc2.length else lc2
                perm = lib.sort_perm(both)
                uniq, _u, _c, n_u = lib.groupby_sorted(
                    lib.gather(both, perm), [], lib.AGG_SUM, False)
                lc2 = lib.search_sorted(lc2, uniq)
                rc2 = lib.search_sorted(rc2, uniq)
                mn, mx = 0, max(int(n_u) - 1, 0)
            mins.append(mn)
            spans.append(mx - mn + 1)
            enc_l.append(lc2)
            enc_r.append(rc2)
        total = 1
        for sp in spans:
            total *= sp
            if total > (1 << 62):
                raise lib.HfError(
                    "merge: combined key range of "
                    f"{on} exceeds 2^62 (hashed multi-key merge is a "
                    "later round)")
        strides = [1] * len(on)
        for i in range(len(on) - 2, -1, -1):
            strides[i] = strides[i + 1] * spans[i + 1]

        def fold(cols):
            comb = None
            for c, mn, st in zip(cols, mins, strides):
                t = lib.map_scalar(lib.MAP_SUB, c, mn)
                if st != 1:
                    t = lib.map_scalar(lib.MAP_MUL, t, st)
                comb = t if comb is None else lib.binary(lib.BIN_ADD,
                                                         comb, t)
            return comb

        def with_key(frame, comb, drop_keys):
            names = [c for c in frame.columns
                     if not (drop_keys and c in on)]
            cats0 = (frame._partitions[0].block().cats
                     if frame._partitions else {})
            cols = {c: concat_col(frame, c) for c in names}
            cols[KEY] = comb
            n = comb.length
            dts = {c: frame.dtypes[c] for c in names}
            dts[KEY] = np.dtype(np.int64)
            return HipDataframe(
                [HipDataframePartition(DeviceBlock(
                    cols, n, {c: v for c, v in cats0.items()
                              if c in names}))],
                pandas.RangeIndex(n), names + [KEY], [n],
                pandas.Series(dts))

        L = with_key(self, fold(enc_l), drop_keys=False)
        R = with_key(other, fold(enc_r), drop_keys=True)
        res = L.broadcast_join(R, KEY, how)
        keep = [c for c in res.columns if c != KEY]
        out = res.take_columns(keep)
        return out

    def _binned_merge(self, other: "HipDataframe", on: str, how: str,
                      key_f64: bool, n_bins: int = 0) -> "HipDataframe":
        """Range-binned (co-shuffled) merge for giant right tables — the
        device form of the reference's range_partitioning_merge
        (storage_formats/pandas/merge.py:39 -> dataframe.py:4087): both
        sides are binned by splitters drawn from the right side's sorted
        DISTINCT keys (hf_shuffle_dest), each bin runs the existing dense
        CSR broadcast join, and pandas' left-major match order is restored
        by one stable sort over a hidden global-left-row column (the
        composition pinned in tests/test_host_logic.py:251).  Removes the
        2^27-distinct-right-keys cap; inner/left on non-dictionary keys
        (right joins arrive here pre-swapped as left).  At world>1 the
        splitters are made identical on every rank (sample all-gather), so
        each bin's broadcast join gathers exactly the global right rows of
        its key range."""
        from .. import distributed as dist_mod

        def concat_col(frame, name):
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        def enc_key(col):
            return lib.ordered_i64(lib.cast_f64(col)) if key_f64 else col

        lk = enc_key(concat_col(self, on))
        rk = enc_key(concat_col(other, on))
        # splitters: quantiles of the right side's sorted distinct keys
        if rk.length:
            perm = lib.sort_perm(rk)
            uniq, _u1, _u2, n_uniq = lib.groupby_sorted(
                lib.gather(rk, perm), [], lib.AGG_SUM, False)
        else:
            uniq, n_uniq = lib.put(np.empty(0, dtype=np.int64)), 0
        n_uniq_glob = int(n_uniq)
        if dist_mod.is_active():
            n_uniq_glob = sum(dist_mod.allgather_lengths(int(n_uniq)))
        B = n_bins or max(2, -(-n_uniq_glob // (1 << 26)))
        S = min(int(n_uniq), 4096)
        if S:
            sidx = np.linspace(0, int(n_uniq) - 1, S).astype(np.int64)
            sample = lib.get(lib.gather(uniq, lib.put(sidx)))
        else:
            sample = np.empty(0, dtype=np.int64)
        if dist_mod.is_active():
            # identical splitters on every rank (each bin's broadcast join
            # is a collective)
            sample = np.sort(dist_mod.allgather_arrays([sample])[0])
        if sample.size == 0:
            splitters = np.empty(0, dtype=np.int64)
            B = 1
        else:
            qs = [(i * sample.size) // B for i in range(1, B)]
            splitters = np.unique(sample[qs])
            B = splitters.size + 1
        ld = lib.shuffle_dest(lk, splitters)
        rd = lib.shuffle_dest(rk, splitters)

        lcats = (self._partitions[0].block().cats
                 if self._partitions else {})
        rcats = (other._partitions[0].block().cats
                 if other._partitions else {})
        LROW = "__hf_lrow__"
        lcols = {c: concat_col(self, c) for c in self.columns}
        rcols = {c: concat_col(other, c) for c in other.columns}
        nl = lk.length
        lrow = lib.filter_iota(lib.filter_plan(
            lib.compare_scalar(lib.CMP_GE, ld, 0.0)), 0) if nl else \
            lib.put(np.empty(0, dtype=np.int64))

        bin_frames = []
        for b in range(B):
            lplan = lib.filter_plan(
                lib.compare_scalar(lib.CMP_EQ, ld, float(b)))
            rplan = lib.filter_plan(
                lib.compare_scalar(lib.CMP_EQ, rd, float(b)))
            any_rows = lplan.n_kept or (dist_mod.is_active()
                                        and rplan.n_kept)
            if not any_rows and not dist_mod.is_active():
                continue
            lbin = {c: lib.filter_apply(lplan, lcols[c])
                    for c in self.columns}
            lbin[LROW] = lib.filter_apply(lplan, lrow)
            rbin = {c: lib.filter_apply(rplan, rcols[c])
                    for c in other.columns}
            Lb = HipDataframe(
                [HipDataframePartition(
                    DeviceBlock(lbin, lplan.n_kept, dict(lcats)))],
                pandas.RangeIndex(lplan.n_kept),
                list(self.columns) + [LROW], [lplan.n_kept],
                pandas.Series({**{c: self.dtypes[c] for c in self.columns},
                               LROW: np.dtype(np.int64)}))
            Rb = HipDataframe(
                [HipDataframePartition(
                    DeviceBlock(rbin, rplan.n_kept, dict(rcats)))],
                pandas.RangeIndex(rplan.n_kept),
                list(other.columns), [rplan.n_kept],
                pandas.Series({c: other.dtypes[c] for c in other.columns}))
            bin_frames.append(Lb.broadcast_join(Rb, on, how, _no_bin=True))
        if not bin_frames:
            empty = self.take_row_range(0, 0)
            return empty.broadcast_join(other.take_row_range(0, 0), on, how)
        # unify per-column dtypes across bins (a bin with NaN fills promoted
        # int64 right columns to float64 — the pandas rule is global)
        out_columns = list(bin_frames[0].columns)
        final_dt = {}
        for c in out_columns:
            dts = [f.dtypes[c] for f in bin_frames]
            final_dt[c] = (np.dtype(np.float64)
                           if any(d == np.dtype(np.float64) for d in dts)
                           else dts[0])
        cat_map = {}
        for f in bin_frames:
            cat_map.update(f._partitions[0].block().cats)
        merged = {}
        for c in out_columns:
            pieces = []
            for f in bin_frames:
                col = f._partitions[0].block().columns[c]
                if (final_dt[c] == np.dtype(np.float64)
                        and col.dtype_code == lib.HF_INT64
                        and c not in cat_map):
                    col = lib.cast_f64(col)
                pieces.append(col)
            merged[c] = pieces[0] if len(pieces) == 1 else lib.concat(pieces)
        order = lib.sort_perm(merged[LROW])
        n_out = merged[LROW].length
        out_cols = {c: lib.gather(merged[c], order) for c in out_columns
                    if c != LROW}
        out_columns = [c for c in out_columns if c != LROW]
        part = HipDataframePartition(
            DeviceBlock(out_cols, n_out,
                        {c: v for c, v in cat_map.items() if c != LROW}))
        return HipDataframe(
            [part], pandas.RangeIndex(n_out), out_columns, [n_out],
            pandas.Series({c: final_dt[c] for c in out_columns}))

    def broadcast_join(self, other: "HipDataframe", on: str,
                       how: str = "inner",
                       _no_bin: bool = False) -> "HipDataframe":
        if on not in self.columns or on not in other.columns:
            raise lib.HfError(f"merge: key column {on!r} missing")
        if how not in ("inner", "left", "outer"):
            raise lib.HfError(
                f"merge how={how!r} not implemented (inner/left/outer/"
                "right this round)")
        fill_dtc = (other._dt_cols() if how == "left"
                    else (self._dt_cols() | other._dt_cols()
                          if how == "outer" else set()))
        if fill_dtc - {on} or (how == "outer" and on in fill_dtc):
            raise lib.HfError(
                "merge: unmatched rows would need NaT fills in datetime "
                f"column(s) {sorted(fill_dtc)} — a later round (inner "
                "joins support datetime)")
        left_names = [c for c in self.columns if c != on]
        right_names = [c for c in other.columns if c != on]
        common = set(left_names) & set(right_names)
        lout = {n: (n + "_x" if n in common else n) for n in left_names}
        rout = {n: (n + "_y" if n in common else n) for n in right_names}

        # materialize right device-side (combine(), dataframe.py:2918)
        def concat_col(frame, name):
            cols = [p.block().columns[name] for p in frame._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        lcats = (self._partitions[0].block().cats
                 if self._partitions else {})
        rcats = (other._partitions[0].block().cats
                 if other._partitions else {})
        key_cats = None
        if (on in lcats) != (on in rcats):
            raise lib.HfError("merge: key column is a string on one side "
                              "only")
        rkeys = concat_col(other, on)
        # float keys ride the order-preserving bit transform: all NaNs
        # canonicalize to ONE ordered value, so NaN==NaN matches — exactly
        # pandas merge semantics for NaN keys.  An int64/float64 mixed key
        # promotes to float64 first (pandas rule).
        key_f64 = on not in lcats and (
            self.dtypes[on] == np.dtype(np.float64)
            or other.dtypes[on] == np.dtype(np.float64))

        def enc_key(col):
            return lib.ordered_i64(lib.cast_f64(col)) if key_f64 else col

        # test hook: MODIN_AMD_MERGE_BINS=B forces the co-shuffled binned
        # merge at any size (parity tests exercise the giant-right path on
        # small data)
        force_bins = int(os.environ.get("MODIN_AMD_MERGE_BINS", "0"))
        if (force_bins > 0 and not _no_bin and on not in lcats
                and how in ("inner", "left")):
            return self._binned_merge(other, on, how, key_f64,
                                      n_bins=force_bins)

        rkeys = enc_key(rkeys)
        if on in lcats:
            # dictionary keys: join in the LEFT dictionary's code space —
            # recode the right key codes; NaN keys are code −1 on BOTH
            # sides, so they match each other (pandas NaN==NaN in merges);
            # right categories absent from the left dictionary map to −2
            # (recode_dict_col) so they never collide with NaN
            key_cats = lcats[on]
            if not rcats[on].equals(key_cats):
                rkeys = recode_dict_col(rkeys, rcats[on], key_cats,
                                        missing=-2)
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
            gathered = dist_mod.allgather_arrays(
                [lib.get(rkeys)] + [lib.get(c) for c in rvals])
            rkeys = lib.put(gathered[0])
            rvals = [lib.put(a) for a in gathered[1:]]
        if rkeys.length:
            r = lib.reduce(rkeys)
            kmin, n_slots = r.imn, r.imx - r.imn + 1
        else:
            kmin, n_slots = 0, 1
        uniq = None  # set on the unbounded-span path (codes join)
        if n_slots > (1 << 27):
            # unbounded key span: densify through the sorted DISTINCT right
            # keys (hf_search_sorted binary search per row; the CSR join
            # then runs in code space [0, n_uniq)).  Covers any int64 span
            # with <= 2^27 distinct right keys.
            perm = lib.sort_perm(rkeys)
            skeys = lib.gather(rkeys, perm)
            uniq, _su, _cu, n_uniq = lib.groupby_sorted(
                skeys, [], lib.AGG_SUM, False)
            if n_uniq > (1 << 27):
                if key_cats is None and how in ("inner", "left"):
                    # giant right: range-binned (co-shuffled) merge
                    return self._binned_merge(other, on, how, key_f64)
                raise lib.HfError(
                    "merge: more than 2^27 distinct right keys with "
                    f"how={how!r}/string keys (binned merge covers "
                    "inner/left on non-dictionary keys)")
            rkeys = lib.search_sorted(rkeys, uniq)
            kmin, n_slots = 0, max(n_uniq, 1)
        # cache the build side on the right frame: its columns are immutable,
        # so repeated merges with the same right frame skip hist/scan/fill
        # (the lazy-metadata pattern again; a real broadcast join caches its
        # build side)
        cache_key = (on, tuple(right_names), kmin, n_slots,
                     uniq is not None, key_f64)
        cached = getattr(other, "_join_build_cache", None)
        if key_cats is not None:
            # dictionary keys: the build lives in the LEFT frame's code
            # space, which varies per left frame — don't cache on the right
            j = lib.join_build(rkeys, rvals, kmin, n_slots)
        elif cached is not None and cached[0] == cache_key:
            j, uniq = cached[1], cached[2]
        else:
            j = lib.join_build(rkeys, rvals, kmin, n_slots)
            other._join_build_cache = (cache_key, j, uniq)

        runiq = uniq  # probe-space distinct right keys (left-join test)
        if how in ("left", "outer") and runiq is None:
            if rkeys.length:
                rperm = lib.sort_perm(rkeys)
                runiq, _ru, _rc2, _rn = lib.groupby_sorted(
                    lib.gather(rkeys, rperm), [], lib.AGG_SUM, False)
            else:
                runiq = lib.put(np.empty(0, dtype=np.int64))
        # pandas dtype rule: a left join that introduces NaN rights turns
        # int64 right columns into float64 — decided GLOBALLY first
        total_unmatched = 0
        un_plans = []
        if how in ("left", "outer"):
            for p in self._partitions:
                block = p.block()
                lk = enc_key(block.columns[on])
                if uniq is not None:
                    lk = lib.search_sorted(lk, uniq)
                    m = lib.compare_scalar(lib.CMP_EQ, lk, -1.0)
                else:
                    code = lib.search_sorted(lk, runiq)
                    m = lib.compare_scalar(lib.CMP_EQ, code, -1.0)
                plan = lib.filter_plan(m)
                un_plans.append((plan, lk))
                total_unmatched += plan.n_kept

        out_parts, lengths = [], []
        for pi, p in enumerate(self._partitions):
            block = p.block()
            lkeys = enc_key(block.columns[on])
            if lkeys.dtype_code != lib.HF_INT64:
                raise lib.HfError("merge: key column must be int64")
            if uniq is not None:  # code space: unmatched lefts become -1
                lkeys = un_plans[pi][1] if how in ("left", "outer") \
                    else lib.search_sorted(lkeys, uniq)
            keys_c, lidx, rcols, nout = lib.join_probe(j, lkeys)
            if uniq is not None:  # decode output codes back to key values
                keys_c = lib.gather(uniq, keys_c)
            if how in ("left", "outer"):
                plan = un_plans[pi][0]
                n_un = plan.n_kept
                if n_un:
                    ulidx = lib.filter_iota(plan, 0)
                    lidx_all = lib.concat([lidx, ulidx])
                    order = lib.sort_perm(lidx_all)
                    lidx_s = lib.gather(lidx_all, order)
                    cols = {}
                    for name in self.columns:
                        key = on if name == on else lout[name]
                        cols[key] = lib.gather(block.columns[name], lidx_s)
                    for i, rn in enumerate(right_names):
                        rc = rcols[i]
                        if rn in rcats:  # dict payload: NaN is code −1
                            nanfill = lib.alloc(n_un, lib.HF_INT64)
                            lib.fill_i64(nanfill.dptr(), -1, n_un)
                        else:
                            if rc.dtype_code == lib.HF_INT64:
                                rc = lib.cast_f64(rc)
                            nanfill = lib.alloc(n_un, lib.HF_FLOAT64)
                            lib.fill_f64(nanfill.dptr(), float("nan"),
                                         n_un)
                        cols[rout[rn]] = lib.gather(
                            lib.concat([rc, nanfill]), order)
                    nout2 = nout + n_un
                    out_parts.append(HipDataframePartition(
                        DeviceBlock(cols, nout2)))
                    lengths.append(nout2)
                    continue
                # no unmatched rows in this partition: fall through, but
                # keep the global dtype decision consistent
                if total_unmatched:
                    rcols = [lib.cast_f64(rc)
                             if (rn not in rcats
                                 and rc.dtype_code == lib.HF_INT64)
                             else rc
                             for rn, rc in zip(right_names, rcols)]
            cols = {}
            for name in self.columns:  # left column order, key in place
                if name == on:
                    cols[on] = lib.ordered_i64(keys_c, inverse=True) \
                        if key_f64 else keys_c
                else:
                    cols[lout[name]] = lib.gather(block.columns[name], lidx)
            for i, rn in enumerate(right_names):
                cols[rout[rn]] = rcols[i]
            out_parts.append(HipDataframePartition(DeviceBlock(cols, nout)))
            lengths.append(nout)
        n_run = 0
        if how == "outer":
            # unmatched RIGHT rows: keys absent from the left key set
            luniq = None
            lk_all = enc_key(concat_col(self, on))
            if uniq is not None:
                lk_all = lib.search_sorted(lk_all, uniq)
            if lk_all.length:
                lperm = lib.sort_perm(lk_all)
                luniq, _l1, _l2, _l3 = lib.groupby_sorted(
                    lib.gather(lk_all, lperm), [], lib.AGG_SUM, False)
            else:
                luniq = lib.put(np.empty(0, dtype=np.int64))
            rk_probe = rkeys  # possibly code space already
            rm = lib.compare_scalar(lib.CMP_EQ,
                                    lib.search_sorted(rk_probe, luniq),
                                    -1.0)
            rplan = lib.filter_plan(rm)
            n_run = rplan.n_kept
            if n_run:
                cols = {}
                rk_out = lib.filter_apply(rplan, rkeys)
                if uniq is not None:
                    rk_out = lib.gather(uniq, rk_out)
                if key_f64:
                    rk_out = lib.ordered_i64(rk_out, inverse=True)
                for name in self.columns:
                    key = on if name == on else lout[name]
                    if name == on:
                        cols[key] = rk_out
                    elif name in lcats:
                        c2 = lib.alloc(n_run, lib.HF_INT64)
                        lib.fill_i64(c2.dptr(), -1, n_run)
                        cols[key] = c2
                    else:
                        c2 = lib.alloc(n_run, lib.HF_FLOAT64)
                        lib.fill_f64(c2.dptr(), float("nan"), n_run)
                        cols[key] = c2
                for i, rn in enumerate(right_names):
                    rc = lib.filter_apply(rplan, rvals[i])
                    if rn not in rcats and rc.dtype_code == lib.HF_INT64 \
                            and total_unmatched:
                        rc = lib.cast_f64(rc)
                    cols[rout[rn]] = rc
                out_parts.append(HipDataframePartition(
                    DeviceBlock(cols, n_run)))
                lengths.append(n_run)
        out_columns = ([on if c == on else lout[c] for c in self.columns]
                       + [rout[c] for c in right_names])
        dtypes = {}
        for c in self.columns:
            d = self.dtypes[c]
            if c == on and key_f64:
                d = np.dtype(np.float64)
            if (how == "outer" and n_run and c != on and c not in lcats
                    and d == np.dtype(np.int64)):
                d = np.dtype(np.float64)
            dtypes[on if c == on else lout[c]] = d
        for c in right_names:
            d = other.dtypes[c]
            if (how in ("left", "outer") and total_unmatched
                    and c not in rcats and d == np.dtype(np.int64)):
                d = np.dtype(np.float64)
            dtypes[rout[c]] = d
        out_cats = {}
        if key_cats is not None:
            out_cats[on] = key_cats
        for c in self.columns:
            if c != on and c in lcats:
                out_cats[lout[c]] = lcats[c]
        for c in right_names:
            if c in rcats:
                out_cats[rout[c]] = rcats[c]
        if out_cats:
            for part in out_parts:
                part._block.cats = dict(out_cats)
        # align device dtypes to the declared result dtypes: partitions
        # with no NaN fills still hold int64 where the frame-wide rule
        # promoted to float64 (left/outer joins) — cast so device-side
        # concat/sort stay type-uniform
        for part in out_parts:
            block = part._block
            changed = False
            cols2 = dict(block.columns)
            for cn, c in cols2.items():
                if (dtypes.get(cn) == np.dtype(np.float64)
                        and c.dtype_code == lib.HF_INT64):
                    cols2[cn] = lib.cast_f64(c)
                    changed = True
            if changed:
                part._block = DeviceBlock(cols2, block.length, block.cats)
        total = sum(lengths)
        res = HipDataframe(out_parts, pandas.RangeIndex(total), out_columns,
                           lengths, pandas.Series(dtypes))
        if how == "outer":
            # pandas sorts outer-join keys; stable sort keeps the
            # within-key left-order/right-match expansion order
            res = res.sort_rows(on, True)
            res._index = pandas.RangeIndex(total)
        return res

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
            out_parts.append(HipDataframePartition(
                DeviceBlock(cols, plan.n_kept, block.cats)))
            lengths.append(plan.n_kept)
            base += length
        idx_col = idx_cols[0] if len(idx_cols) == 1 else lib.concat(idx_cols)
        return HipDataframe(out_parts, DeviceIndex(idx_col, name=None),
                            self.columns, lengths, self.dtypes)

    # ---- comparison map (mask column) ----
    def compare_scalar(self, op_code: int, scalar) -> "HipDataframe":
        dtc = self._dt_cols()

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            out = {}
            for name, col in block.columns.items():
                if name in block.cats:
                    out[name] = _compare_dict_col(op_code, col,
                                                  block.cats[name], scalar)
                else:
                    if isinstance(scalar, str):
                        raise lib.HfError(
                            f"comparing numeric column {name!r} to a "
                            "string scalar")
                    m = lib.compare_scalar(op_code, col, float(scalar))
                    if name in dtc:
                        # pandas NaT semantics: every ordered compare is
                        # False on NaT rows (NE stays True — IEEE-NaN
                        # style); NOTNA is exactly (!= iNaT)
                        notnat = lib.compare_scalar(lib.CMP_NE, col,
                                                    float(INAT))
                        if op_code == lib.CMP_NOTNA:
                            m = notnat
                        elif op_code != lib.CMP_NE:
                            m = lib.binary(lib.BIN_MUL, m, notnat)
                    out[name] = m
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
                    DeviceBlock(cols, hi - lo, block.cats)))
                lengths.append(hi - lo)
            off += ln
        if not out_parts:
            block = self._partitions[0].block()
            cols = {n: lib.col_slice(c, 0, 0)
                    for n, c in block.columns.items()}
            out_parts = [HipDataframePartition(
                DeviceBlock(cols, 0, block.cats))]
            lengths = [0]
        idx = self.index[start:stop]
        return HipDataframe(out_parts, idx, self.columns, lengths, self.dtypes)

    def take_rows(self, positions: np.ndarray) -> "HipDataframe":
        """Positional row selection (iloc list/array form): one device
        gather per column over the concatenated frame; the result keeps
        the selected original index labels."""
        pos = np.asarray(positions, dtype=np.int64)
        n = len(self)
        if pos.size and (pos.min() < -n or pos.max() >= n):
            raise lib.HfError("iloc: position out of bounds")
        pos = np.where(pos < 0, pos + n, pos)

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        gidx = lib.put(pos)
        cols = {c: lib.gather(concat_col(c), gidx) for c in self.columns}
        part = HipDataframePartition(
            DeviceBlock(cols, int(pos.size), dict(blk_cats)))
        idx = self.index[pos] if pos.size else self.index[:0]
        return HipDataframe([part], idx, self.columns, [int(pos.size)],
                            self.dtypes)

    # ---- astype over all columns ----
    def astype_all(self, dtype) -> "HipDataframe":
        dt = np.dtype(dtype)
        blk_cats0 = (self._partitions[0].block().cats
                     if self._partitions else {})
        if blk_cats0:
            if dt == np.dtype(object) or dt.kind in ("U", "S"):
                return self  # string -> str is the identity
            if dt not in (np.dtype(np.float64), np.dtype(np.int64)):
                raise lib.HfError(f"astype({dt}) on string columns is a "
                                  "later round")
            # numeric parse of the HOST DICTIONARY + one device gather
            # (pandas astype semantics: unparseable or NaN->int raise)
            def parse_block(block: DeviceBlock) -> DeviceBlock:
                out = {}
                for n, c in block.columns.items():
                    if n not in block.cats:
                        out[n] = lib.map_scalar(
                            lib.MAP_CAST_F64 if dt == np.dtype(np.float64)
                            else lib.MAP_CAST_I64, c, 0)
                        continue
                    cats = block.cats[n].to_numpy(dtype=object)
                    try:
                        parsed = [float(x) for x in cats]
                    except (TypeError, ValueError) as e:
                        raise lib.HfError(
                            f"astype({dt}): column {n!r}: {e}")
                    if dt == np.dtype(np.int64):
                        if any(p != int(p) for p in parsed):
                            raise lib.HfError(
                                f"astype(int64): column {n!r} holds "
                                "non-integral strings")
                        lut = np.empty(len(cats) + 1, dtype=np.int64)
                        lut[1:] = [int(p) for p in parsed]
                        lut[0] = 0
                        if lib.reduce(c).imn < 0:
                            raise lib.HfError(
                                "astype(int64): NaN strings cannot "
                                "convert (pandas raises too)")
                    else:
                        lut = np.empty(len(cats) + 1, dtype=np.float64)
                        lut[0] = np.nan
                        lut[1:] = parsed
                    shifted = lib.map_scalar(lib.MAP_ADD, c, 1)
                    out[n] = lib.gather(lib.put(lut), shifted)
                return DeviceBlock(out, block.length)
            return self.map(parse_block)
        if np.issubdtype(dt, np.datetime64):
            # int64 ns view -> datetime tag (device data unchanged; float
            # sources cast to int64 ns first, pandas' rule)
            def to_dt(block: DeviceBlock) -> DeviceBlock:
                return DeviceBlock(
                    {n: (c if c.dtype_code == lib.HF_INT64
                         else lib.map_scalar(lib.MAP_CAST_I64, c, 0))
                     for n, c in block.columns.items()}, block.length)
            out = self.map(to_dt)
            out.dtypes = pandas.Series(
                {c: np.dtype("datetime64[ns]") for c in self.columns})
            return out
        if any(isinstance(d, np.dtype) and np.issubdtype(d, np.datetime64)
               for d in self.dtypes) and dt == np.dtype(np.int64):
            # datetime -> int64: drop the tag (device already int64 ns);
            # pandas raises on NaT -> int64 and so do we
            self._guard_nat("astype(int64)", self._dt_cols())
            out = HipDataframe(self._partitions, self._index, self.columns,
                               self._row_lengths,
                               pandas.Series({c: np.dtype(np.int64)
                                              for c in self.columns}))
            return out
        code = {np.dtype(np.float64): lib.MAP_CAST_F64,
                np.dtype(np.int64): lib.MAP_CAST_I64}.get(dt)
        if code is None:
            raise lib.HfError(f"astype to {dt} is a later round")

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
        # unify per-column dictionaries across frames (host union + device
        # LUT recode where they differ)
        all_cats = [f._partitions[0].block().cats if f._partitions else {}
                    for f in frames]
        dict_names = sorted({n for c in all_cats for n in c})
        for n in dict_names:
            if not all(n in c for c in all_cats):
                raise lib.HfError(
                    f"concat: column {n!r} is a string in some frames only")
        parts = []
        if dict_names:
            merged = {n: all_cats[0][n] for n in dict_names}
            for c in all_cats[1:]:
                for n in dict_names:
                    merged[n] = union_cats(merged[n], c[n])
            for f, fcats in zip(frames, all_cats):
                for p in f._partitions:
                    block = p.block()
                    newcols = dict(block.columns)
                    for n in dict_names:
                        if not fcats[n].equals(merged[n]):
                            newcols[n] = recode_dict_col(
                                block.columns[n], fcats[n], merged[n])
                    parts.append(HipDataframePartition(
                        DeviceBlock(newcols, block.length, dict(merged))))
        else:
            parts = [p for f in frames for p in f._partitions]
        from .. import distributed as dist_mod
        if dist_mod.is_active() and dist_mod.world_size() > 1:
            blk_cats = (parts[0].block().cats if parts else {})
            return self._concat_rows_exchange(frames, parts, cols, blk_cats)
        lengths = [ln for f in frames for ln in f._row_lengths]
        idx = pandas.Index(np.concatenate([np.asarray(f.index) for f in frames]))
        return HipDataframe(parts, idx, cols, lengths, self.dtypes)

    def _concat_rows_exchange(self, frames, parts, cols, blk_cats):
        """World>1 concat: re-shard so the GLOBAL row order equals pandas
        concat order — all of frame 0 (ranks in order), then frame 1, … —
        instead of the rank-local interleave (the round-1 known deviation).
        Local rows are already ascending in global concat position, so the
        per-destination spans are contiguous interval overlaps (host
        arithmetic, no per-row planning) and ride exchange_column; the
        received source-major pieces are restored to global order with one
        device gather per column."""
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        r = dist_mod.rank()
        F = len(frames)
        lens = [dist_mod.allgather_lengths(len(f)) for f in frames]
        FB = np.cumsum([0] + [sum(L) for L in lens])
        N = int(FB[-1])
        runs = [(int(FB[fi]) + sum(lens[fi][:r]), lens[fi][r])
                for fi in range(F)]
        per = -(-N // P) if N else 0
        T = [min(i * per, N) for i in range(P + 1)]
        send_counts = [0] * P
        for g0, L in runs:
            for d in range(P):
                lo, hi = max(g0, T[d]), min(g0 + L, T[d + 1])
                if hi > lo:
                    send_counts[d] += hi - lo
        local_cols = {}
        for m in cols:
            cs = [p.block().columns[m] for p in parts]
            local_cols[m] = cs[0] if len(cs) == 1 else lib.concat(cs)
        recv = {m: dist_mod.exchange_column(local_cols[m], send_counts)
                for m in cols}
        idx_arrays = [np.asarray(f.index) for f in frames]
        if any(a.dtype == object for a in idx_arrays):
            raise lib.HfError("distributed concat: non-numeric index is a "
                              "later round")
        as_f64 = any(a.dtype.kind == "f" for a in idx_arrays)
        idx_cat = np.concatenate(
            [a.astype(np.float64 if as_f64 else np.int64)
             for a in idx_arrays]) if idx_arrays else np.empty(0, np.int64)
        ridx = dist_mod.exchange_column(lib.put(idx_cat), send_counts)
        # received layout: src-major, frames in order within src; each piece
        # is the (src, frame) shard ∩ my target interval
        pieces = []
        off = 0
        for src in range(P):
            for fi in range(F):
                g0 = int(FB[fi]) + sum(lens[fi][:src])
                L = lens[fi][src]
                lo, hi = max(g0, T[r]), min(g0 + L, T[r + 1])
                if hi > lo:
                    pieces.append((lo, off, hi - lo))
                    off += hi - lo
        my_n = off
        pieces.sort()
        gather_idx = (np.concatenate(
            [np.arange(o, o + L, dtype=np.int64) for (_, o, L) in pieces])
            if pieces else np.empty(0, dtype=np.int64))
        gidx = lib.put(gather_idx)
        out_cols = {m: lib.gather(recv[m], gidx) for m in cols}
        idx_vals = lib.get(lib.gather(ridx, gidx))
        part = HipDataframePartition(DeviceBlock(out_cols, my_n, blk_cats))
        return HipDataframe([part], pandas.Index(idx_vals), cols, [my_n],
                            self.dtypes)

    # ---- sort (PandasDataframe.sort_by device form, dataframe.py:2742;
    #      SURVEY §8f.2): stable radix permutation + column gathers ----
    @staticmethod
    def _effective_sort_key(col, is_dict, ascending, na_first=False,
                            is_dt=False):
        """(key col, ascending) -> (int64 key, ascending') whose stable
        ASCENDING' radix sort realizes pandas order.  NaNs — dictionary
        code −1 or float NaN — sort LAST for both directions by default
        (na_position='last'): the NaN rows map to a sentinel above every
        valid key and valid keys keep (asc) or negate (desc) their order,
        so the pass always runs ascending when an adjustment is needed.
        na_first=True (na_position='first') mirrors the sentinel BELOW
        every valid key instead.  Float keys first ride the
        order-preserving f64->i64 bit transform (hf_ordered_i64)."""
        BIG = 1 << 62
        if col.dtype_code == lib.HF_FLOAT64:
            r = lib.reduce(col) if col.length else None
            okey = lib.ordered_i64(col)
            if r is None or r.count == col.length:  # no NaN
                return okey, ascending
            # NaN sentinel must clear ordered(±inf) ~ ±0x7FF0... ~ 9.22e18
            # (2^62 would land INSIDE the ordered bits of floats >= 2.0)
            NANKEY = ((1 << 63) - 1) * (-1 if na_first else 1)
            notna = lib.compare_scalar(lib.CMP_NOTNA, col, 0.0)
            isna_big = lib.map_scalar(
                lib.MAP_MUL, lib.map_scalar(lib.MAP_RSUB, notna, 1),
                NANKEY)
            base = okey if ascending else lib.map_scalar(lib.MAP_NEG,
                                                         okey, 0)
            return lib.binary(lib.BIN_ADD,
                              lib.binary(lib.BIN_MUL, base, notna),
                              isna_big), True
        if is_dt and col.length and lib.reduce(col).imn == INAT:
            # datetime key with NaT: same sentinel scheme as float NaN —
            # NaT rows jump above (na last) / below (na first) every
            # valid ns value, valid keys keep or negate their order
            NANKEY = ((1 << 63) - 1) * (-1 if na_first else 1)
            notnat = lib.compare_scalar(lib.CMP_NE, col, float(INAT))
            nat_big = lib.map_scalar(
                lib.MAP_MUL, lib.map_scalar(lib.MAP_RSUB, notnat, 1),
                NANKEY)
            base = col if ascending else lib.map_scalar(lib.MAP_NEG,
                                                        col, 0)
            return lib.binary(lib.BIN_ADD,
                              lib.binary(lib.BIN_MUL, base, notnat),
                              nat_big), True
        if not is_dict or not col.length or lib.reduce(col).imn >= 0:
            return col, ascending
        if ascending and na_first:
            return col, ascending  # code −1 already sorts first
        m = lib.compare_scalar(lib.CMP_EQ, col, -1.0)
        if ascending:
            t = lib.map_scalar(lib.MAP_MUL, m, BIG + 1)
            return lib.binary(lib.BIN_ADD, col, t), True
        neg = lib.map_scalar(lib.MAP_NEG, col, 0)
        # descending: NaN code −1 negates to +1; push it far above (last)
        # or far below (first) the negated valid codes
        t = lib.map_scalar(lib.MAP_MUL, m, -BIG if na_first else BIG - 1)
        return lib.binary(lib.BIN_ADD, neg, t), True

    @staticmethod
    def _compose_sort_perm(eff_keys):
        """Stable multi-key permutation: LSD over the key list (sort by the
        LAST key first; stability carries earlier keys' order through) —
        the device form of pandas lexsort semantics."""
        perm = None
        for ekc, ea in reversed(eff_keys):
            if perm is None:
                perm = lib.sort_perm(ekc, ea)
            else:
                gk = lib.gather(ekc, perm)
                p2 = lib.sort_perm(gk, ea)
                perm = lib.gather(perm, p2)
        return perm

    def sort_rows(self, by, ascending=True,
                  na_position: str = "last") -> "HipDataframe":
        if na_position not in ("last", "first"):
            raise lib.HfError("sort_values: na_position must be 'last' or "
                              "'first'")
        na_first = na_position == "first"
        by_list = [by] if isinstance(by, str) else list(by)
        if isinstance(ascending, (bool, np.bool_, int)):
            asc_list = [bool(ascending)] * len(by_list)
        else:
            asc_list = [bool(a) for a in ascending]
        if len(asc_list) != len(by_list) or not by_list:
            raise lib.HfError("sort_values: by/ascending length mismatch")
        for b in by_list:
            if b not in self.columns:
                raise lib.HfError(f"sort_values: column {b!r} missing")
        if not isinstance(self._index, pandas.RangeIndex) or \
                self._index.start != 0 or self._index.step != 1:
            raise lib.HfError(
                "sort_values: only RangeIndex frames this round")

        def concat_col(name):
            cols = [p.block().columns[name] for p in self._partitions]
            return cols[0] if len(cols) == 1 else lib.concat(cols)

        blk_cats = (self._partitions[0].block().cats
                    if self._partitions else {})
        from ..distributed import is_active
        if is_active():
            return self._sort_rows_distributed(by_list, asc_list, blk_cats,
                                               concat_col, na_first)
        cache = {}

        def cat_col(name):
            if name not in cache:
                cache[name] = concat_col(name)
            return cache[name]

        dtc = self._dt_cols()
        eff = [self._effective_sort_key(cat_col(b), b in blk_cats, a,
                                        na_first, is_dt=b in dtc)
               for b, a in zip(by_list, asc_list)]
        perm = self._compose_sort_perm(eff)
        cols = {name: lib.gather(cat_col(name), perm)
                for name in self.columns}
        n = perm.length
        part = HipDataframePartition(DeviceBlock(cols, n, blk_cats))
        return HipDataframe([part], DeviceIndex(perm, name=None),
                            self.columns, [n], self.dtypes)

    def _sort_rows_distributed(self, by_list, asc_list, blk_cats,
                               concat_col, na_first=False):
        """Distributed sort_values: the range-partitioning shuffle + local
        stable sort (SURVEY §8f.2 "reuses the shuffle"; reference
        sort_by -> _apply_func_to_range_partitioning, dataframe.py:2742).
        Rows shuffle by the PRIMARY effective key (equal primaries share a
        rank, so secondary keys stay a local matter), each rank then runs
        the same multi-key stable sort the single-rank path uses.
        Stability: exchange output is source-rank (= global-position)
        ordered and the per-dest filter preserves order.  Result: rank r
        holds globally-sorted slice r; the index carries the original
        global positions."""
        import numpy as np
        from .. import distributed as dist_mod
        P = dist_mod.world_size()
        dtc = self._dt_cols()
        k0 = concat_col(by_list[0])
        ek0, ea0 = self._effective_sort_key(k0, by_list[0] in blk_cats,
                                            asc_list[0], na_first,
                                            is_dt=by_list[0] in dtc)
        n = ek0.length
        S = min(n, 4096)
        if S:
            sidx = np.linspace(0, n - 1, S).astype(np.int64)
            sample = lib.get(lib.gather(ek0, lib.put(sidx)))
        else:
            sample = np.empty(0, dtype=np.int64)
        splitters = dist_mod.sample_splitters(sample)
        dest = lib.shuffle_dest(ek0, splitters)
        if not ea0:  # rank 0 takes the LARGEST primary-key range
            dest = lib.map_scalar(lib.MAP_RSUB, dest, P - 1)
        base = dist_mod.global_row_base(n)
        names = list(self.columns)
        send_cols = {m: [] for m in names}
        send_pos, send_counts = [], []
        cols_cat = {m: concat_col(m) for m in names}
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
        eff = [self._effective_sort_key(recv[b], b in blk_cats, a,
                                        na_first, is_dt=b in dtc)
               for b, a in zip(by_list, asc_list)]
        perm = self._compose_sort_perm(eff)
        out_cols = {m: lib.gather(recv[m], perm) for m in names}
        pos_sorted = lib.gather(rpos, perm)
        ln = perm.length
        part = HipDataframePartition(DeviceBlock(out_cols, ln, blk_cats))
        return HipDataframe([part], DeviceIndex(pos_sorted, name=None),
                            names, [ln], self.dtypes)

    # ---- dropna mask: AND of per-column notna (pandas dropna(how="any")) ----
    def notna_all_mask(self) -> "HipDataframe":
        dtc = self._dt_cols()

        def block_fn(block: DeviceBlock) -> DeviceBlock:
            acc = None
            for name, col in block.columns.items():
                if name in block.cats:  # dict-encoded: NaN is code −1
                    m = lib.compare_scalar(lib.CMP_NE, col, -1.0)
                elif name in dtc:  # datetime: NaT is iNaT
                    m = lib.compare_scalar(lib.CMP_NE, col, float(INAT))
                else:
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
    """Derive result dtypes from the first partition's (drained) block.
    Dictionary-encoded (string) columns hold int64 codes on device but
    are object frames API-side."""
    block = parts[0].block()
    return pandas.Series({
        name: (np.dtype(object) if name in block.cats else col.np_dtype)
        for name, col in block.columns.items()})


