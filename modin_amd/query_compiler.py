"""HipQueryCompiler — L2 of the stack, reusing the reference's method names.

Mirrors ``modin/core/storage_formats/pandas/query_compiler.py``
(``PandasQueryCompiler`` :279,297 — wraps one ``_modin_frame``): every hot op
is bound to an operator template exactly as the reference binds them:
  sum   = TreeReduce.register(...)        (query_compiler.py:984)
  add   = Binary.register(...)            (:535)
  fillna = Map.register(...)              (:2710)
  groupby_sum = GroupByReduce agg table   (:3747 via
                storage_formats/pandas/groupby.py:75 build_qc_method)
The registered kernels are hipframe device ops, not pandas callables.
"""

from __future__ import annotations

import numpy as np
import pandas

from .algebra import Binary, GroupByReduce, Map, TreeReduce
from .core import lib
from .core.dataframe import HipDataframe


class HipQueryCompiler:
    def __init__(self, modin_frame: HipDataframe):
        self._modin_frame = modin_frame

    @property
    def __constructor__(self):
        return type(self)

    # ---- ingestion ----
    @classmethod
    def from_pandas(cls, df: pandas.DataFrame) -> "HipQueryCompiler":
        return cls(HipDataframe.from_pandas(df))

    def to_pandas(self) -> pandas.DataFrame:
        return self._modin_frame.to_pandas()

    @property
    def columns(self):
        return self._modin_frame.columns

    @property
    def index(self):
        return self._modin_frame.index

    @property
    def dtypes(self):
        return self._modin_frame.dtypes

    def __len__(self):
        return len(self._modin_frame)

    # ---- Map ops (query_compiler.py:2036 abs, :2710 fillna) ----
    abs = Map.register(lib.MAP_ABS)
    neg = Map.register(lib.MAP_NEG)
    fillna = Map.register(lib.MAP_FILLNA, f64_only=True)

    # ---- Binary ops (query_compiler.py:535 add & friends) ----
    add = Binary.register(lib.MAP_ADD, lib.BIN_ADD)
    sub = Binary.register(lib.MAP_SUB, lib.BIN_SUB)
    rsub = Binary.register(lib.MAP_RSUB, lib.BIN_SUB)
    mul = Binary.register(lib.MAP_MUL, lib.BIN_MUL)
    truediv = Binary.register(lib.MAP_DIV, lib.BIN_DIV)
    rtruediv = Binary.register(lib.MAP_RDIV, lib.BIN_DIV)
    # scalar-only int ops (HF_MAP_IDIV/IMOD: Python floor/mod semantics)
    floordiv_int = Map.register(lib.MAP_IDIV)
    mod_int = Map.register(lib.MAP_IMOD)
    sqrt = Map.register(lib.MAP_SQRT, f64_only=True)

    # ---- TreeReduce (query_compiler.py:984 sum etc.) ----
    sum = TreeReduce.register("sum")
    count = TreeReduce.register("count")
    mean = TreeReduce.register("mean")
    min = TreeReduce.register("min")
    max = TreeReduce.register("max")

    # ---- GroupByReduce (query_compiler.py:3747 groupby_sum) ----
    groupby_sum = GroupByReduce.register("sum")
    groupby_count = GroupByReduce.register("count")
    groupby_mean = GroupByReduce.register("mean")
    groupby_min = GroupByReduce.register("min")
    groupby_max = GroupByReduce.register("max")

    def var(self, ddof: int = 1):
        return self._moment(ddof, sqrt_=False)

    def std(self, ddof: int = 1):
        return self._moment(ddof, sqrt_=True)

    def _moment(self, ddof, sqrt_):
        """Column var/std: Σx and Σx² partials from two tree-reduce passes
        (the squared pass reuses the same single-scan reduce kernel),
        composed on host — pandas nanvar, ddof default 1."""
        import math
        frame = self._modin_frame
        p1 = frame.tree_reduce(frame.columns)

        def square(block):
            from modin_amd.core.partition import DeviceBlock
            out = {}
            for name, col in block.columns.items():
                f = lib.cast_f64(col)
                out[name] = lib.binary(lib.BIN_MUL, f, f)
            return DeviceBlock(out, block.length)

        sq = frame.map(square)
        p2 = sq.tree_reduce(frame.columns)
        vals = []
        for name in frame.columns:
            n = p1[name]["count"]
            if n - ddof <= 0:
                vals.append(float("nan"))
                continue
            v = (p2[name]["sum"] - p1[name]["sum"] ** 2 / n) / (n - ddof)
            v = max(v, 0.0)
            vals.append(math.sqrt(v) if sqrt_ else v)
        return pandas.Series(vals, index=pandas.Index(list(frame.columns)),
                             dtype=np.float64)

    def corr(self) -> pandas.DataFrame:
        return self._pairwise_moments(corr=True)

    def cov(self, ddof: int = 1) -> pandas.DataFrame:
        return self._pairwise_moments(corr=False, ddof=ddof)

    def _pairwise_moments(self, corr: bool, ddof: int = 1):
        """Pearson corr / cov over PAIRWISE-COMPLETE rows (pandas rule):
        per pair, z = x + 0*y propagates NaN from either side and the
        NaN-skipping f64 reduce then yields the masked n/Σ/Σ² in one
        pass each — no mask materialization, 6 device passes per pair,
        k×k scalars combined on host; world>1 SUM-all-reduces the six
        scalars per pair (moments are linear over shards)."""
        import math
        frame = self._modin_frame
        blk_cats = (frame._partitions[0].block().cats
                    if frame._partitions else {})
        names = list(frame.columns)
        bad = [c for c in names
               if c in blk_cats or not (
                   isinstance(frame.dtypes[c], np.dtype)
                   and frame.dtypes[c] in (np.dtype(np.int64),
                                           np.dtype(np.float64)))]
        if bad:
            raise lib.HfError(f"corr/cov: non-numeric columns {bad} — "
                              "select numeric columns")
        from . import distributed as dist_mod

        def concat_col(name):
            cs = [p.block().columns[name] for p in frame._partitions]
            c = cs[0] if len(cs) == 1 else lib.concat(cs)
            return lib.cast_f64(c)

        cols = {c: concat_col(c) for c in names}
        k = len(names)
        # shard-local masked moments per (i, j) pair — sums are linear,
        # so world>1 just SUM-all-reduces the 6 scalars per pair
        moments = {}
        for i in range(k):
            for j in range(i, k):
                x, y = cols[names[i]], cols[names[j]]
                zx = lib.binary(lib.BIN_ADD, x,
                                lib.map_scalar(lib.MAP_MUL, y, 0.0))
                zy = lib.binary(lib.BIN_ADD, y,
                                lib.map_scalar(lib.MAP_MUL, x, 0.0))
                rx, ry = lib.reduce(zx), lib.reduce(zy)
                moments[(i, j)] = [
                    float(rx.count), rx.sum, ry.sum,
                    lib.reduce(lib.binary(lib.BIN_MUL, zx, zy)).sum,
                    lib.reduce(lib.binary(lib.BIN_MUL, zx, zx)).sum,
                    lib.reduce(lib.binary(lib.BIN_MUL, zy, zy)).sum]
        if dist_mod.is_active() and dist_mod.world_size() > 1:
            keys_ = sorted(moments)
            flat = [v for p in keys_ for v in moments[p]]
            flat = dist_mod.allreduce_scalars(flat)
            for a, p in enumerate(keys_):
                moments[p] = flat[a * 6:(a + 1) * 6]
        out = np.full((k, k), np.nan)
        for i in range(k):
            for j in range(i, k):
                n, sx, sy, sxy, sxx, syy = moments[(i, j)]
                if corr:
                    if n < 2:
                        continue
                    den = ((sxx - sx ** 2 / n)
                           * (syy - sy ** 2 / n))
                    if den <= 0:
                        continue
                    v = (sxy - sx * sy / n) / math.sqrt(den)
                    v = max(-1.0, min(1.0, v))
                else:
                    if n - ddof <= 0:
                        continue
                    v = (sxy - sx * sy / n) / (n - ddof)
                out[i, j] = out[j, i] = v
        idx = pandas.Index(names)
        return pandas.DataFrame(out, index=idx, columns=idx)

    def _tag_key_index(self, res: "HipQueryCompiler", by):
        """Groupby by a datetime64 key: the result index materializes back
        to the tagged dtype (the int64-ns typed-column layer)."""
        key = by if isinstance(by, str) else (
            by[0] if isinstance(by, (list, tuple)) and len(by) == 1 else None)
        if key is not None:
            dt = dict(self._modin_frame.dtypes).get(key)
            if (dt is not None and isinstance(dt, np.dtype)
                    and np.issubdtype(dt, np.datetime64)):
                res._modin_frame._index_dtype = dt
        return res

    def groupby_var(self, by: str, ddof: int = 1) -> "HipQueryCompiler":
        return self._tag_key_index(self.__constructor__(
            self._modin_frame.groupby_var(by, ddof, sqrt=False)), by)

    def groupby_std(self, by: str, ddof: int = 1) -> "HipQueryCompiler":
        return self._tag_key_index(self.__constructor__(
            self._modin_frame.groupby_var(by, ddof, sqrt=True)), by)

    def median(self):
        vals = self._modin_frame.median_columns()
        return pandas.Series(vals, dtype=np.float64)

    def clip(self, lower=None, upper=None) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.clip_columns(lower, upper))

    def reduce_axis1(self, op: str) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.reduce_axis1(op))

    def cumsum(self) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.cumsum_rows(lib.AGG_SUM))

    def cummin(self) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.cumsum_rows(lib.AGG_MIN))

    def cummax(self) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.cumsum_rows(lib.AGG_MAX))

    def cumprod(self) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.cumsum_rows(lib.AGG_PROD))

    def shift(self, periods: int) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.shift_rows(periods))

    def diff(self, periods: int) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.diff_rows(periods))

    def idxmax(self):
        vals = self._modin_frame.idx_extreme(True)
        return pandas.Series(vals)

    def idxmin(self):
        vals = self._modin_frame.idx_extreme(False)
        return pandas.Series(vals)

    def quantile(self, qs):
        vals = self._modin_frame.quantile_columns(list(qs))
        return pandas.DataFrame(vals, index=pandas.Index(list(qs)))

    def groupby_tail_agg(self, by, fn_name: str, dropna: bool = True,
                         **kw) -> "HipQueryCompiler":
        """dropna=False for the non-reduce aggs (var/std/median/quantile/
        nunique/size/first/last/idxmax): NaN float keys are encoded to the
        order-preserving int64 form whose canonical-NaN sentinel sorts
        LAST (exactly where pandas puts the NaN group), the agg runs with
        ordinary int64 keys, and the result index decodes back to float
        (sentinel -> NaN).  Single-key; string keys stay loud."""
        fn = getattr(self, fn_name)
        if dropna:
            return fn(by, **kw)
        frame = self._modin_frame
        keys = [by] if isinstance(by, str) else list(by)
        cats = (frame._partitions[0].block().cats
                if frame._partitions else {})
        if any(b in cats for b in keys):
            raise lib.HfError(
                f"groupby(dropna=False).{fn_name} with string keys is a "
                "later round")
        f64keys = [b for b in keys
                   if frame.dtypes[b] == np.dtype(np.float64)]
        if not f64keys:
            return fn(by, **kw)  # int64 keys carry no NaN
        if len(keys) > 1:
            raise lib.HfError(
                f"groupby(dropna=False).{fn_name} with multiple float "
                "keys is a later round")
        res = self.__constructor__(
            frame.encode_keys_ordered(f64keys)).__getattribute__(
                fn_name)(by, **kw)
        rframe = res._modin_frame
        ivals = np.asarray(rframe.index)
        rframe._index = pandas.Index(
            lib.ordered_to_f64_np(ivals.astype(np.int64)), name=keys[0])
        return res

    def groupby_median(self, by) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.groupby_median(by))

    def groupby_quantile(self, by, q: float) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.groupby_quantile(by, q))

    def groupby_prod(self, by) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.groupby_prod(by))

    def groupby_first(self, by) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.groupby_firstlast(by, last=False))

    def groupby_last(self, by) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.groupby_firstlast(by, last=True))

    def groupby_nunique(self, by) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.groupby_nunique(by))

    def groupby_transform(self, by, how: str, ascending: bool = True,
                          method: str = "average", periods: int = 1,
                          dropna: bool = True,
                          na_option: str = "keep") -> "HipQueryCompiler":
        """Same-length transforms in original row order (pandas
        DataFrameGroupBy.cumsum/cummin/cummax/cumprod/cumcount/rank/
        ngroup/shift/diff; reference routes these through
        modin/pandas/groupby.py -> qc groupby methods)."""
        return self.__constructor__(self._modin_frame.groupby_transform(
            by, how, ascending=ascending, method=method, periods=periods,
            dropna=dropna, na_option=na_option))

    def groupby_idxmax(self, by) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.groupby_idxminmax(by, maximum=True))

    def groupby_idxmin(self, by) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.groupby_idxminmax(by, maximum=False))

    def str_op(self, op: str, pat: str = None,
               na=None, repl: str = None, regex: bool = False,
               width: int = None) -> "HipQueryCompiler":
        """Series.str.<op> over a dictionary column: the transform runs on
        the HOST DICTIONARY (O(#categories)) and reaches the rows with one
        device gather — no string ever touches the GPU (SURVEY §8f.3
        design).  ops: len, lower, upper, strip/lstrip/rstrip, title,
        capitalize, zfill, replace (literal or regex),
        contains (literal or regex)/startswith/endswith/match/fullmatch
        (na=True/False/None->NaN)."""
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import DeviceBlock, \
            HipDataframePartition
        import pandas as pd
        frame = self._modin_frame
        name = frame.columns[0]
        blk_cats = (frame._partitions[0].block().cats
                    if frame._partitions else {})
        if name not in blk_cats:
            raise lib.HfError(f"str accessor on non-string column {name!r}")
        cats = blk_cats[name]

        def concat_col():
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        codes = concat_col()
        n = codes.length
        shifted = lib.map_scalar(lib.MAP_ADD, codes, 1)  # NaN(-1) -> slot 0
        if op in ("lower", "upper", "strip", "lstrip", "rstrip", "title",
                  "capitalize", "replace", "zfill"):
            import re
            if op == "replace":
                if regex:
                    rx = re.compile(pat)
                    fn = lambda c: rx.sub(repl, c)  # noqa: E731
                else:
                    fn = lambda c: c.replace(pat, repl)  # noqa: E731
            elif op == "zfill":
                fn = lambda c: c.zfill(int(width))  # noqa: E731
            else:
                fn = lambda c: getattr(c, op)()  # noqa: E731
            vals = [fn(c) for c in cats.to_numpy(dtype=object)]
            new_codes, new_cats = pd.factorize(pd.Series(vals), sort=True)
            lut = np.empty(len(cats) + 1, dtype=np.int64)
            lut[0] = -1
            lut[1:] = new_codes
            out = lib.gather(lib.put(lut), shifted)
            blk = DeviceBlock({name: out}, n, {name: pd.Index(new_cats)})
            dts = pd.Series({name: np.dtype(object)})
        elif op == "len":
            lut = np.empty(len(cats) + 1, dtype=np.float64)
            lut[0] = np.nan
            lut[1:] = [len(c) for c in cats.to_numpy(dtype=object)]
            out = lib.gather(lib.put(lut), shifted)
            blk = DeviceBlock({name: out}, n)
            dts = pd.Series({name: np.dtype(np.float64)})
        elif op in ("contains", "startswith", "endswith", "match",
                    "fullmatch"):
            import re
            if op == "contains" and regex:
                rx = re.compile(pat)
                test = lambda c: rx.search(c) is not None  # noqa: E731
            elif op == "match":
                rx = re.compile(pat)
                test = lambda c: rx.match(c) is not None  # noqa: E731
            elif op == "fullmatch":
                rx = re.compile(pat)
                test = (lambda c:  # noqa: E731
                        rx.fullmatch(c) is not None)
            else:
                test = {"contains": lambda c: pat in c,
                        "startswith": lambda c: c.startswith(pat),
                        "endswith": lambda c: c.endswith(pat)}[op]
            hit = [bool(test(c)) for c in cats.to_numpy(dtype=object)]
            if na is None:
                # pandas returns object [True, False, NaN]; this backend
                # returns float64 [1.0, 0.0, NaN] (documented deviation —
                # pass na=True/False for a clean bool column)
                lut = np.empty(len(cats) + 1, dtype=np.float64)
                lut[0] = np.nan
                lut[1:] = np.asarray(hit, dtype=np.float64)
                dts = pd.Series({name: np.dtype(np.float64)})
            else:
                lut = np.empty(len(cats) + 1, dtype=np.int64)
                lut[0] = 1 if na else 0
                lut[1:] = np.asarray(hit, dtype=np.int64)
                dts = pd.Series({name: np.dtype(bool)})
            out = lib.gather(lib.put(lut), shifted)
            blk = DeviceBlock({name: out}, n)
        else:
            raise lib.HfError(f"str.{op} not supported")
        res = HipDataframe([HipDataframePartition(blk)], frame._index,
                           [name], [n], dts)
        return self.__constructor__(res)

    def map_dict(self, mapping: dict,
                 keep_missing: bool) -> "HipQueryCompiler":
        """Series.map(dict) (keep_missing=False: unmapped -> NaN, pandas
        map rule) and Series.replace(dict) (keep_missing=True: unmapped
        keep their value).  Dictionary columns remap on the HOST
        DICTIONARY (O(#categories)) + one device gather; int64 columns
        match exactly via hf_search_sorted over the sorted key set + LUT
        gather.  float64 source columns are a later round (loud)."""
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import DeviceBlock, \
            HipDataframePartition
        import pandas as pd
        frame = self._modin_frame
        name = frame.columns[0]
        blk_cats = (frame._partitions[0].block().cats
                    if frame._partitions else {})

        def concat_col():
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        n = len(frame)
        _MISS = object()
        if name in blk_cats:
            cats = blk_cats[name].to_numpy(dtype=object)
            vals = [mapping.get(c, _MISS) for c in cats]
            if keep_missing:
                vals = [c if v is _MISS else v
                        for c, v in zip(cats, vals)]
            codes = concat_col()
            shifted = lib.map_scalar(lib.MAP_ADD, codes, 1)
            if all(v is _MISS or isinstance(v, str) for v in vals):
                vv = pd.Series([None if v is _MISS else v for v in vals],
                               dtype=object)
                new_codes, new_cats = pd.factorize(vv, sort=True)
                lut = np.empty(len(cats) + 1, dtype=np.int64)
                lut[0] = -1
                lut[1:] = new_codes
                out = lib.gather(lib.put(lut), shifted)
                blk = DeviceBlock({name: out}, n,
                                  {name: pd.Index(new_cats)})
                dts = pd.Series({name: np.dtype(object)})
            elif all(v is _MISS
                     or isinstance(v, (int, float, np.integer, np.floating))
                     for v in vals):
                lut = np.empty(len(cats) + 1, dtype=np.float64)
                lut[0] = np.nan
                lut[1:] = [np.nan if v is _MISS else float(v)
                           for v in vals]
                out = lib.gather(lib.put(lut), shifted)
                blk = DeviceBlock({name: out}, n)
                dts = pd.Series({name: np.dtype(np.float64)})
            else:
                raise lib.HfError(
                    "map/replace over a string column needs all-string "
                    "or all-numeric mapped values")
            res = HipDataframe([HipDataframePartition(blk)], frame._index,
                               [name], [n], dts)
            return self.__constructor__(res)
        src_dt = frame.dtypes[name]
        src_is_dt = (isinstance(src_dt, np.dtype)
                     and np.issubdtype(src_dt, np.datetime64))
        if src_dt != np.dtype(np.int64) and not src_is_dt:
            raise lib.HfError("map/replace(dict) over float64 columns is "
                              "a later round (int64/string sources only)")
        if not all(isinstance(k, (int, np.integer)) for k in mapping):
            raise lib.HfError("map/replace(dict) on an int64 column needs "
                              "int keys")
        ks = np.array(sorted(mapping), dtype=np.int64)
        vs = [mapping[int(k)] for k in ks]
        if any(isinstance(v, float) and np.isnan(v) for v in vs) \
                and keep_missing:
            raise lib.HfError("replace(dict) with NaN values is a later "
                              "round")
        int_vals = all(isinstance(v, (bool, int, np.integer)) for v in vs)
        col = concat_col()
        pos = lib.search_sorted(col, lib.put(ks))
        shifted = lib.map_scalar(lib.MAP_ADD, pos, 1)
        if src_is_dt and not (keep_missing and int_vals):
            raise lib.HfError("map over datetime columns is a later "
                              "round (replace with int ns values works)")
        if keep_missing:
            m = lib.compare_scalar(lib.CMP_GE, pos, 0)
            inv = lib.map_scalar(lib.MAP_RSUB, m, 1)
            if int_vals:
                lut = np.r_[np.int64(0), np.array(vs, dtype=np.int64)]
                mapped = lib.gather(lib.put(lut), shifted)
                out = lib.binary(lib.BIN_ADD,
                                 lib.binary(lib.BIN_MUL, mapped, m),
                                 lib.binary(lib.BIN_MUL, col, inv))
                dt = src_dt if src_is_dt else np.dtype(np.int64)
            else:
                lut = np.r_[np.float64(0.0),
                            np.array(vs, dtype=np.float64)]
                mapped = lib.gather(lib.put(lut), shifted)
                out = lib.binary(
                    lib.BIN_ADD,
                    lib.binary(lib.BIN_MUL, mapped, lib.cast_f64(m)),
                    lib.binary(lib.BIN_MUL, lib.cast_f64(col),
                               lib.cast_f64(inv)))
                dt = np.dtype(np.float64)
        else:
            covered = n == 0 or lib.reduce(pos).imn >= 0
            if covered and int_vals:
                lut = np.r_[np.int64(0), np.array(vs, dtype=np.int64)]
                out = lib.gather(lib.put(lut), shifted)
                dt = np.dtype(np.int64)
            else:
                lut = np.empty(len(ks) + 1, dtype=np.float64)
                lut[0] = np.nan
                lut[1:] = np.array(vs, dtype=np.float64)
                out = lib.gather(lib.put(lut), shifted)
                dt = np.dtype(np.float64)
        blk = DeviceBlock({name: out}, n)
        res = HipDataframe([HipDataframePartition(blk)], frame._index,
                           [name], [n], pd.Series({name: dt}))
        return self.__constructor__(res)

    def cut_codes(self, edges, right: bool = True,
                  as_codes: bool = False,
                  label_edges=None) -> "HipQueryCompiler":
        """pd.cut engine: device range binning — ONE hf_shuffle_dest
        pass over the ordered-transformed edges (#{edge < x} in the
        order-isomorphic int64 space), codes −1 for NaN/out-of-range.
        as_codes=False -> dictionary column over an IntervalIndex (the
        categorical form); as_codes=True -> float64 codes (pandas
        labels=False, NaN for unbinned)."""
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import DeviceBlock, \
            HipDataframePartition
        import pandas as pd
        frame = self._modin_frame
        name = frame.columns[0]
        blk_cats = (frame._partitions[0].block().cats
                    if frame._partitions else {})
        dt = frame.dtypes[name]
        if name in blk_cats or (isinstance(dt, np.dtype)
                                and np.issubdtype(dt, np.datetime64)):
            raise lib.HfError("cut: numeric columns only this round")
        edges = np.asarray(edges, dtype=np.float64)
        if len(edges) < 2 or not (np.diff(edges) > 0).all():
            raise lib.HfError("cut: edges must be increasing (duplicate "
                              "quantile edges: qcut duplicates='drop')")
        nbins = len(edges) - 1

        def concat_col():
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        col = concat_col()
        n = col.length
        ox = lib.ordered_i64(lib.cast_f64(col))
        oe = lib.get(lib.ordered_i64(lib.put(edges)))
        dest = lib.shuffle_dest(ox, oe + 1 if right else oe)
        c = lib.map_scalar(lib.MAP_SUB, dest, 1)
        ok = lib.binary(lib.BIN_MUL,
                        lib.compare_scalar(lib.CMP_GE, c, 0.0),
                        lib.compare_scalar(lib.CMP_LE, c,
                                           float(nbins - 1)))
        codes = lib.map_scalar(
            lib.MAP_SUB,
            lib.binary(lib.BIN_MUL,
                       lib.map_scalar(lib.MAP_ADD, c, 1), ok), 1)
        if as_codes:
            # pandas labels=False: float64 with NaN for unbinned rows
            out = lib.fixup_empty(lib.cast_f64(codes), ok)
            blk = DeviceBlock({name: out}, n)
            dts = pd.Series({name: np.dtype(np.float64)})
            cats = {}
        else:
            cats_idx = pd.IntervalIndex.from_breaks(
                edges if label_edges is None else label_edges,
                closed="right" if right else "left")
            blk = DeviceBlock({name: codes}, n,
                              {name: pd.Index(cats_idx)})
            dts = pd.Series({name: np.dtype(object)})
            cats = {name: pd.Index(cats_idx)}
        res = HipDataframe([HipDataframePartition(blk)], frame._index,
                           [name], [n], dts)
        return self.__constructor__(res)

    def dt_floor(self, unit_ns: int) -> "HipQueryCompiler":
        """Series.dt.floor / normalize: ns − (ns mod unit) — HF_MAP_IMOD
        carries Python's sign rule, so pre-1970 values floor correctly;
        NaT rows keep iNaT through an arithmetic blend."""
        from modin_amd.core.dataframe import HipDataframe, INAT
        from modin_amd.core.partition import DeviceBlock, \
            HipDataframePartition
        frame = self._modin_frame
        name = frame.columns[0]
        dt = frame.dtypes[name]
        if not (isinstance(dt, np.dtype) and np.issubdtype(
                dt, np.datetime64)):
            raise lib.HfError("dt.floor on a non-datetime column")

        def concat_col():
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        ns = concat_col()
        n = ns.length
        r = lib.binary(lib.BIN_SUB, ns,
                       lib.map_scalar(lib.MAP_IMOD, ns, int(unit_ns)))
        if n and lib.reduce(ns).imn == INAT:
            m = lib.compare_scalar(lib.CMP_NE, ns, float(INAT))
            inv = lib.map_scalar(lib.MAP_RSUB, m, 1)
            nat = lib.map_scalar(lib.MAP_MUL, inv, INAT)
            r = lib.binary(lib.BIN_ADD,
                           lib.binary(lib.BIN_MUL, r, m), nat)
        blk = DeviceBlock({name: r}, n)
        res = HipDataframe([HipDataframePartition(blk)], frame._index,
                           [name], [n], pandas.Series({name: dt}))
        return self.__constructor__(res)

    def to_datetime_from_strings(self, format=None,
                                 errors: str = "raise"
                                 ) -> "HipQueryCompiler":
        """pandas.to_datetime over a string (dictionary) column: the
        parse runs ONCE PER CATEGORY on the host (pandas' own parser —
        exact format semantics), then one device LUT gather maps every
        row's code to its int64 ns; string NaN (code −1) -> NaT.  No
        per-row Python object is ever created."""
        from modin_amd.core.dataframe import HipDataframe, INAT
        from modin_amd.core.partition import DeviceBlock, \
            HipDataframePartition
        import pandas as pd
        frame = self._modin_frame
        name = frame.columns[0]
        dt = frame.dtypes[name]
        if isinstance(dt, np.dtype) and np.issubdtype(dt, np.datetime64):
            return self  # already datetime
        blk_cats = (frame._partitions[0].block().cats
                    if frame._partitions else {})
        if name not in blk_cats:
            raise lib.HfError("to_datetime: string (or datetime) Series "
                              "only — int64 ns columns use "
                              "astype('datetime64[ns]')")
        cats = blk_cats[name]
        parsed = pd.to_datetime(pd.Series(cats.to_numpy(dtype=object)),
                                format=format, errors=errors)
        lut = np.empty(len(cats) + 1, dtype=np.int64)
        lut[0] = INAT
        lut[1:] = parsed.to_numpy().astype("datetime64[ns]").view(np.int64)

        def concat_col():
            cs = [p.block().columns[name] for p in frame._partitions]
            return cs[0] if len(cs) == 1 else lib.concat(cs)

        codes = concat_col()
        n = codes.length
        shifted = lib.map_scalar(lib.MAP_ADD, codes, 1)
        out = lib.gather(lib.put(lut), shifted)
        blk = DeviceBlock({name: out}, n)
        res = HipDataframe([HipDataframePartition(blk)], frame._index,
                           [name], [n],
                           pandas.Series({name: np.dtype(
                               "datetime64[ns]")}))
        return self.__constructor__(res)

    def dt_field(self, field: str) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.dt_field(field))

    def rank(self, method: str = "average", ascending: bool = True,
             na_option: str = "keep") -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.rank_rows(
            ascending=ascending, method=method, na_option=na_option))

    def fillna_directional(self, how: str) -> "HipQueryCompiler":
        """pandas ffill/bfill (frame-level, one constant-key group)."""
        return self.__constructor__(self._modin_frame.fill_rows(how))

    def rolling_agg(self, window: int, min_periods,
                    op: str) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.rolling_agg(
            window, min_periods, op))

    def expanding_agg(self, min_periods, op: str) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.expanding_agg(
            min_periods, op))

    def fillna_dict(self, values: dict) -> "HipQueryCompiler":
        """pandas fillna({column: scalar}): per-column fill values;
        unlisted columns pass through."""
        from modin_amd.core.partition import DeviceBlock
        frame = self._modin_frame
        for c in values:
            if c not in frame.columns:
                raise lib.HfError(f"fillna: column {c!r} missing")

        def block_fn(block):
            out = {}
            for name, col in block.columns.items():
                # int64 columns are NaN-free: pandas leaves them (and
                # their dtype) untouched — skip MAP_FILLNA's f64 promotion
                if (name in values and name not in block.cats
                        and col.dtype_code != lib.HF_INT64):
                    out[name] = lib.map_scalar(lib.MAP_FILLNA, col,
                                               values[name])
                else:
                    out[name] = col
            return DeviceBlock(out, block.length, block.cats)
        return self.__constructor__(frame.map(block_fn))

    def duplicated(self, subset=None, keep="first") -> "HipQueryCompiler":
        """Row-duplicate mask (pandas duplicated): cumcount over ALL
        subset columns with dropna=False (NaN==NaN, the canonical-NaN
        effective key) > 0.  keep='last' runs keep='first' over the
        device-reversed rows (pandas' own reduction); keep=False ORs the
        two masks."""
        if keep == "last":
            rev = self._modin_frame.reverse_rows()
            rev._index = pandas.RangeIndex(len(rev))  # positional pass
            d = self.__constructor__(rev).duplicated(subset)
            m = d._modin_frame.reverse_rows()
            m._index = pandas.RangeIndex(len(m))
            return self.__constructor__(m)
        if keep is False:
            f = self.duplicated(subset)._modin_frame
            l_ = self.duplicated(subset, keep="last")._modin_frame

            def orzip(lb, rb):
                from modin_amd.core.partition import DeviceBlock
                (ln, lc), = lb.columns.items()
                (rc,) = rb.columns.values()
                return DeviceBlock(
                    {ln: lib.binary(lib.BIN_MAX, lc, rc)}, lb.length)

            return self.__constructor__(f.n_ary_op(orzip, l_))
        if keep != "first":
            raise lib.HfError(f"duplicated: bad keep {keep!r}")
        by = (list(self._modin_frame.columns) if subset is None
              else ([subset] if isinstance(subset, str) else list(subset)))
        cc = self._modin_frame.groupby_transform(by, "cumcount",
                                                 dropna=False)
        return self.__constructor__(cc.compare_scalar(lib.CMP_GE, 1.0))

    def drop_duplicates(self, subset=None,
                        keep="first") -> "HipQueryCompiler":
        dup = self.duplicated(subset, keep=keep)
        inv = dup._modin_frame.compare_scalar(lib.CMP_EQ, 0.0)
        # the transform result is one partition; re-slice the mask to the
        # frame's partition lengths so filter_rows stays co-partitioned
        frame = self._modin_frame
        mask2 = inv.repartition_like(frame._row_lengths)
        return self.__constructor__(frame.filter_rows(mask2))

    def where_mask(self, cond: "HipQueryCompiler",
                   other=None) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.where_rows(
            cond._modin_frame, other))

    def round(self, decimals: int = 0) -> "HipQueryCompiler":  # noqa: A003
        return self.__constructor__(
            self._modin_frame.round_cols(decimals))

    def groupby_size(self, by: str) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.groupby_size(by))

    def hconcat(self, others: list) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.hconcat(
            [o._modin_frame for o in others]))

    def rename_columns(self, mapping: dict) -> "HipQueryCompiler":
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import DeviceBlock
        frame = self._modin_frame
        new_cols = [mapping.get(c, c) for c in frame.columns]

        def relabel(block):
            return DeviceBlock(
                {mapping.get(n, n): c for n, c in block.columns.items()},
                block.length,
                {mapping.get(n, n): c for n, c in block.cats.items()})
        parts = [p.add_to_apply_calls(relabel) for p in frame._partitions]
        dts = pandas.Series({mapping.get(n, n): d
                             for n, d in frame.dtypes.items()})
        return self.__constructor__(HipDataframe(
            parts, frame._index, new_cols, frame._row_lengths, dts))

    def distinct_stats(self):
        return self._modin_frame.distinct_stats(self._modin_frame.columns[0])

    def groupby_agg(self, by: str, agg: str,
                    dropna: bool = True) -> "HipQueryCompiler":
        fn = {
            "sum": type(self).groupby_sum,
            "count": type(self).groupby_count,
            "mean": type(self).groupby_mean,
            "min": type(self).groupby_min,
            "max": type(self).groupby_max,
            "var": type(self).groupby_var,
            "std": type(self).groupby_std,
            "median": type(self).groupby_median,
            "prod": type(self).groupby_prod,
            "first": type(self).groupby_first,
            "last": type(self).groupby_last,
        }.get(agg)
        if fn is None:
            raise lib.HfError(
                f"groupby agg {agg!r} not implemented on the HipNative backend"
            )
        if agg in ("sum", "count", "mean", "min", "max"):
            return self._tag_key_index(fn(self, by, dropna=dropna), by)
        if not dropna:
            name = {"var": "groupby_var", "std": "groupby_std",
                    "median": "groupby_median", "first": "groupby_first",
                    "last": "groupby_last",
                    "prod": "groupby_prod"}[agg]
            return self._tag_key_index(
                self.groupby_tail_agg(by, name, dropna=False), by)
        return self._tag_key_index(fn(self, by), by)

    # ---- comparisons (query_compiler gt/lt/eq bindings) -> int64 0/1 mask
    def _compare(self, op_code, other):
        import datetime
        if other is None or (isinstance(other, float) and np.isnan(other)) \
                or other is pandas.NaT:
            raise lib.HfError("comparisons against NaN/NaT: pandas "
                              "returns all-False — compare explicitly")
        if isinstance(other, (pandas.Timestamp, np.datetime64,
                              datetime.datetime)):
            # datetime operand vs the int64-ns typed column
            other = int(pandas.Timestamp(other).value)
        if not np.isscalar(other):
            raise lib.HfError("comparisons support scalars this round")
        return self.__constructor__(
            self._modin_frame.compare_scalar(op_code, other))

    def gt(self, other):
        return self._compare(lib.CMP_GT, other)

    def ge(self, other):
        return self._compare(lib.CMP_GE, other)

    def lt(self, other):
        return self._compare(lib.CMP_LT, other)

    def le(self, other):
        return self._compare(lib.CMP_LE, other)

    def eq(self, other):
        return self._compare(lib.CMP_EQ, other)

    def ne(self, other):
        return self._compare(lib.CMP_NE, other)

    def notna(self):
        return self._compare(lib.CMP_NOTNA, 0.0)

    def dropna_mask(self) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.notna_all_mask())

    # ---- boolean row mask (qc.getitem_array device form) ----
    def getitem_array(self, mask_qc: "HipQueryCompiler") -> "HipQueryCompiler":
        mask = mask_qc._modin_frame
        frame = self._modin_frame
        if list(mask._row_lengths) != list(frame._row_lengths):
            # single-partition masks (str/dt accessor outputs) re-slice to
            # the frame's partition boundaries
            mask = mask.repartition_like(frame._row_lengths)
        return self.__constructor__(frame.filter_rows(mask))

    def sort_index(self, ascending: bool = True) -> "HipQueryCompiler":
        """pandas sort_index (host argsort of the index labels — the index
        is host metadata — + one device gather per column)."""
        idx = self._modin_frame.index
        # Index.sort_values gives pandas' exact (stable, dup-safe) order
        # for both directions — index labels are host metadata
        _, order = idx.sort_values(return_indexer=True,
                                   ascending=ascending)
        return self.take_rows(order)

    def take_rows(self, positions) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.take_rows(positions))

    def take_row_range(self, start: int, stop: int) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.take_row_range(start, stop))

    def astype(self, dtype) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.astype_all(dtype))

    # ---- concat (reference qc.concat -> PartitionManager.concat :943) ----
    def concat(self, others: list) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.concat_rows(
            [o._modin_frame for o in others]))

    # ---- sort (reference qc.sort_rows_by_column_values) ----
    def sort_rows_by_column_values(self, by: str,
                                   ascending: bool = True,
                                   na_position: str = "last"
                                   ) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.sort_rows(by, ascending,
                                        na_position=na_position))

    # ---- merge (query_compiler merge -> MergeImpl.row_axis_merge,
    #      storage_formats/pandas/merge.py:104) ----
    def merge(self, right: "HipQueryCompiler", on: str = None,
              how: str = "inner", left_on=None,
              right_on=None) -> "HipQueryCompiler":
        if left_on is not None or right_on is not None:
            # pandas left_on/right_on: both key columns survive in the
            # output (merge.py keep_keys path).  Rewritten as an `on`
            # merge over a zero-copy ALIAS of each key under one hidden
            # name: the original key columns then ride the join as plain
            # payload (suffixes, NaN fill for unmatched rows — exactly
            # the pandas column semantics), and the hidden key is dropped.
            if on is not None:
                raise lib.HfError("merge: 'on' excludes left_on/right_on")
            if left_on is None or right_on is None:
                raise lib.HfError("merge: left_on and right_on must be "
                                  "given together")
            if isinstance(left_on, (list, tuple)):
                if len(left_on) != 1:
                    raise lib.HfError("merge: multi-key left_on/right_on "
                                      "is a later round")
                left_on = left_on[0]
            if isinstance(right_on, (list, tuple)):
                if len(right_on) != 1:
                    raise lib.HfError("merge: multi-key left_on/right_on "
                                      "is a later round")
                right_on = right_on[0]
            if left_on == right_on:
                # pandas collapses same-named keys into the `on` form
                return self.merge(right, on=left_on, how=how)
            tmpk = "\x00lrk\x00"
            l2 = self.__constructor__(
                self._modin_frame.alias_column(left_on, tmpk))
            r2 = self.__constructor__(
                right._modin_frame.alias_column(right_on, tmpk))
            merged = l2.merge(r2, on=tmpk, how=how)
            return merged.getitem_column_array(
                [c for c in merged.columns if c != tmpk])
        if how == "cross":
            if on is not None:
                raise lib.HfError("merge: how='cross' forbids 'on'")
            return self.__constructor__(
                self._modin_frame.cross_join(right._modin_frame))
        if on is None:
            raise lib.HfError("merge: 'on' required (except how='cross')")
        if isinstance(on, (list, tuple)):
            if len(on) == 1:
                on = on[0]
            elif how not in ("inner", "left", "right"):
                raise lib.HfError(
                    f"multi-key merge how={how!r} is a later round "
                    "(inner/left/right)")
            elif how != "right":
                return self.__constructor__(
                    self._modin_frame.merge_multi(right._modin_frame,
                                                  list(on), how))
        if how == "right":
            # pandas right join == swapped left join with the suffix roles
            # flipped back and columns restored to left-then-right order
            # (single- and multi-key: `keyset` carries both forms)
            keyset = {on} if isinstance(on, str) else set(on)
            swapped = right.merge(self, on=on, how="left")
            frame = swapped._modin_frame
            lcols = [c for c in self.columns if c not in keyset]
            rcols = [c for c in right.columns if c not in keyset]
            common = set(lcols) & set(rcols)
            # in the swapped join, OUR columns got "_y" and right's "_x"
            ren = {}
            for c in lcols:
                if c in common:
                    ren[c + "_y"] = c + "_x"
            for c in rcols:
                if c in common:
                    ren[c + "_x"] = c + "_y"
            # two-step rename through temporaries to avoid collisions
            tmp = {k: ("\x00tmp\x00" + k) for k in ren}
            out = swapped.rename_columns(tmp) if ren else swapped
            if ren:
                out = out.rename_columns(
                    {("\x00tmp\x00" + k): v for k, v in ren.items()})
            # pandas puts the keys at their left-frame positions; ours
            # keeps the left column order with the keys in place
            left_order = [c if c in keyset else
                          (c + "_x" if c in common else c)
                          for c in self.columns]
            order = left_order + [(c + "_y" if c in common else c)
                                  for c in rcols]
            return out.getitem_column_array(order)
        return self.__constructor__(
            self._modin_frame.broadcast_join(right._modin_frame, on, how)
        )

    # ---- column assignment (reference qc.setitem / insert,
    #      storage_formats/pandas/query_compiler.py setitem_builder) ----
    def write_column(self, name: str,
                     value: "HipQueryCompiler") -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.set_column(name, value._modin_frame))

    def write_scalar_column(self, name: str, value) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.set_scalar_column(name, value))

    def sample_rows(self, n: int, seed: int) -> "HipQueryCompiler":
        return self.__constructor__(
            self._modin_frame.sample_rows(int(n), int(seed)))

    # ---- projection ----
    def getitem_column_array(self, names) -> "HipQueryCompiler":
        return self.__constructor__(self._modin_frame.take_columns(list(names)))
