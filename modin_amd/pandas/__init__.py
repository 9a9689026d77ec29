"""modin_amd.pandas — the drop-in user API (reference L1) for the hot path.

Mirrors ``modin/pandas``'s DataFrame/Series/GroupBy surface for the operators
the HipNative backend accelerates (SURVEY.md §8a + §8f, see README/DESIGN
for the full coverage list): construction/from_pandas/read_parquet,
elementwise arithmetic and comparisons, map ops (fillna/abs/round/clip/
where/mask), reductions (sum/mean/count/min/max/var/std/median/quantile/
idxmax/idxmin, axis 0 and 1), sort_values (multi-key, na_position),
filter/dropna/duplicated/drop_duplicates/nlargest, merge
(inner/left/right/outer/cross, on/left_on+right_on, int64/float64/string
keys incl. NaN matching), concat, column assignment
(setitem/insert/assign), map/replace(dict), melt/pivot_table, sample,
corr/cov (incl. world>1), to_datetime + NaT semantics (DESIGN.md NaT
scope), floordiv/mod, astype(dict), read_csv/to_csv, concat column
alignment, query, cut/qcut, get_dummies, sample, set_index,
filter/select_dtypes, dt.floor/normalize, named aggregation,
and the groupby family: the reduce aggs + agg forms
(modin/pandas/dataframe.py:2188 sum; modin/pandas/groupby.py:1330
DataFrameGroupBy.sum -> _wrap_aggregation), size/nunique/first/last/
median/quantile/idxmax/idxmin, dropna=False, and the same-length
transforms (cum*/cumcount/ngroup/rank/shift/diff/transform(agg)).
Anything beyond this surface raises loudly (no silent pandas fallback —
DESIGN.md "No CPU fallback").

Reductions return pandas.Series (the reference's API layer also lowers
1×N reduce results into Series via ``_reduce_dimension``).
"""

from __future__ import annotations

import numpy as np
import pandas

from ..core import lib
from ..core.lib import HfError as HfErrorProxy
from ..query_compiler import HipQueryCompiler

__all__ = ["DataFrame", "Series", "concat", "merge", "from_pandas"]


def read_parquet(path, columns=None):
    """Columnar parquet ingestion straight to device:
    pyarrow -> numpy views / dictionary parts -> hf_put.  Matches
    pandas.read_parquet output (strings as object, nullable ints as
    float64, RangeIndex)."""
    from ..io import read_parquet as _rp
    return DataFrame(query_compiler=_rp(path, columns=columns))


def read_csv(path, columns=None, **csv_kwargs):
    """CSV ingestion straight to device via pyarrow.csv — the reference's
    headline read_csv op, columnar, no per-row pandas materialization."""
    from ..io import read_csv as _rc
    return DataFrame(query_compiler=_rc(path, columns=columns,
                                        **csv_kwargs))


def concat(objs, ignore_index: bool = False, axis=0):
    """pandas.concat.  axis=1 composes columns positionally through the
    zero-copy set_column path (equal lengths enforced device-side).
    axis=0: mismatched columns align with NaN fills
    (pandas outer-join rule: first frame's columns, then new names in
    appearance order; an int64 column missing anywhere promotes to
    float64 — the pandas dtype rule).  Missing datetime columns raise
    (NaT is a later round)."""
    objs = list(objs)
    if not objs:
        raise HfErrorProxy("concat of empty list")
    if axis in (1, "columns"):
        frames = [o.to_frame() if isinstance(o, Series) else o
                  for o in objs]
        out = frames[0].copy()
        for o in frames[1:]:
            for c in o.columns:
                if c in list(out.columns):
                    raise HfErrorProxy(
                        f"concat(axis=1): duplicate column {c!r}")
                out[c] = o[c]
        return out
    union = []
    for o in objs:
        for c in o.columns:
            if c not in union:
                union.append(c)
    if any(list(o.columns) != union for o in objs):
        dts = {}
        for o in objs:
            for c in o.columns:
                dts.setdefault(c, o.dtypes[c])
        aligned = []
        for o in objs:
            qc = o._query_compiler
            have = set(o.columns)
            for c in union:
                if c in have:
                    continue
                dt = dts[c]
                if isinstance(dt, np.dtype) and np.issubdtype(
                        dt, np.datetime64):
                    raise HfErrorProxy(
                        f"concat: datetime column {c!r} missing in one "
                        "frame (NaT fill is a later round)")
                if dt == np.dtype(object):
                    qc = qc.write_column(
                        c, HipQueryCompiler.from_pandas(
                            pandas.DataFrame(
                                {c: pandas.Series(
                                    [None] * len(o), dtype=object)})))
                else:
                    qc = qc.write_scalar_column(c, np.nan)
            # a NaN-filled int64 column promotes EVERY frame's copy
            for c in union:
                if (c in have and dts[c] == np.dtype(np.int64)
                        and any(c not in set(x.columns) for x in objs)):
                    qc = qc.write_column(
                        c, qc.getitem_column_array([c]).astype(
                            np.float64))
            aligned.append(qc.getitem_column_array(union))
        out = DataFrame(query_compiler=aligned[0].concat(aligned[1:]))
    else:
        out = DataFrame(query_compiler=objs[0]._query_compiler.concat(
            [o._query_compiler for o in objs[1:]]))
    if ignore_index:
        out._query_compiler._modin_frame._index = pandas.RangeIndex(
            len(out))
    return out


def merge(left: "DataFrame", right: "DataFrame", **kwargs) -> "DataFrame":
    """Module-level pandas.merge form."""
    return left.merge(right, **kwargs)


def to_datetime(arg, format=None, errors: str = "raise"):  # noqa: A002
    """pandas.to_datetime over a string Series: parses the HOST
    DICTIONARY once per distinct value + one device gather (string NaN
    -> NaT); datetime Series pass through."""
    if not isinstance(arg, Series):
        raise HfErrorProxy("to_datetime takes a Series")
    return Series(
        query_compiler=arg._query_compiler.to_datetime_from_strings(
            format=format, errors=errors), name=arg.name)


def _round_frac(x, precision):
    # pandas core/reshape/tile.py _round_frac: round to `precision`
    # SIGNIFICANT fractional digits for sub-1 magnitudes
    if not np.isfinite(x) or x == 0:
        return x
    frac, whole = np.modf(x)
    if whole == 0:
        digits = -int(np.floor(np.log10(abs(frac)))) - 1 + precision
    else:
        digits = precision
    return np.around(x, digits)


def _infer_precision(base, edges):
    # pandas tile.py _infer_precision: smallest precision keeping the
    # rounded breaks unique
    for precision in range(base, 20):
        lv = np.asarray([_round_frac(b, precision) for b in edges])
        if np.unique(lv).size == edges.size:
            return precision
    return base


def cut(x: "Series", bins, right: bool = True, labels=None) -> "Series":
    """pandas.cut: device range binning (one hf_shuffle_dest pass in the
    order-isomorphic int64 space).  labels=None -> Interval values as a
    dictionary column (groupby/value_counts behave observed=True —
    empty bins don't appear; documented deviation from pandas'
    Categorical); labels=False -> float64 bin codes (NaN unbinned),
    exactly pandas."""
    if not isinstance(x, Series):
        raise HfErrorProxy("cut takes a Series")
    if labels not in (None, False):
        raise HfErrorProxy("cut: labels=None|False this round")
    if isinstance(bins, (int, np.integer)):
        mn = x.min()
        mx = x.max()
        if np.isnan(mn):
            raise HfErrorProxy("cut: all-NaN input")
        if mn == mx:  # pandas widens a zero-range by 0.1%
            mn = mn - 0.001 * abs(mn) if mn != 0 else mn - 0.001
            mx = mx + 0.001 * abs(mx) if mx != 0 else mx + 0.001
            edges = np.linspace(mn, mx, int(bins) + 1)
        else:
            edges = np.linspace(mn, mx, int(bins) + 1)
            adj = (mx - mn) * 0.001  # pandas includes the boundary point
            if right:
                edges[0] -= adj
            else:
                edges[-1] += adj
        # pandas labels round the breaks to an inferred precision while
        # the BINNING itself uses the unrounded edges
        prec = _infer_precision(3, edges)
        lab = np.asarray([_round_frac(b, prec) for b in edges])
        return Series(query_compiler=x._query_compiler.cut_codes(
            edges, right=right, as_codes=labels is False,
            label_edges=lab), name=x.name)
    edges = np.asarray(bins, dtype=np.float64)
    return Series(query_compiler=x._query_compiler.cut_codes(
        edges, right=right, as_codes=labels is False), name=x.name)


def qcut(x: "Series", q, labels=None,
         duplicates: str = "raise") -> "Series":
    """pandas.qcut: quantile edges from the device sorted-column
    quantiles, then the cut engine.  labels=False matches pandas
    exactly; labels=None yields Interval values from the raw quantile
    edges (pandas' display rounding is not replicated)."""
    if not isinstance(x, Series):
        raise HfErrorProxy("qcut takes a Series")
    qs = (np.linspace(0, 1, int(q) + 1) if isinstance(q, (int, np.integer))
          else np.asarray(q, dtype=np.float64))
    edges = np.asarray(x.quantile(list(qs)))
    if duplicates == "drop":
        edges = np.unique(edges)
    elif (np.diff(edges) <= 0).any():
        raise HfErrorProxy("qcut: duplicate bin edges (pass "
                           "duplicates='drop')")
    edges = edges.copy()
    # pandas qcut includes the minimum: widen the first edge a hair
    edges[0] -= (abs(edges[0]) * 0.001 if edges[0] != 0 else 0.001)
    return Series(query_compiler=x._query_compiler.cut_codes(
        edges, right=True, as_codes=labels is False), name=x.name)


def get_dummies(data: "Series", prefix: str = None) -> "DataFrame":
    """pandas.get_dummies over a string Series: one 0/1 device column
    per category (EQ masks over the dictionary codes; NaN rows are 0 in
    every column, the pandas dummy_na=False default).  Columns land as
    int64 0/1 (this backend's bool carrier — pandas emits bool)."""
    if not isinstance(data, Series):
        raise HfErrorProxy("get_dummies takes a Series")
    qc = data._query_compiler
    frame = qc._modin_frame
    name = frame.columns[0]
    blk_cats = (frame._partitions[0].block().cats
                if frame._partitions else {})
    if name not in blk_cats:
        raise HfErrorProxy("get_dummies: string Series only this round")
    cats = list(blk_cats[name].to_numpy(dtype=object))
    if len(cats) > 64:
        raise HfErrorProxy("get_dummies: > 64 categories")
    names = [c if prefix is None else f"{prefix}_{c}" for c in cats]
    acc = qc.eq(cats[0]).rename_columns({name: names[0]})
    for c, out_name in zip(cats[1:], names[1:]):
        acc = acc.write_column(out_name, qc.eq(c))
    return DataFrame(query_compiler=acc)


def isna(obj):
    """Module-level pandas.isna over this backend's Series."""
    if isinstance(obj, Series):
        return obj.isna()
    return pandas.isna(obj)


def notna(obj):
    if isinstance(obj, Series):
        return obj.notna()
    return pandas.notna(obj)


def unique(s: "Series"):
    if not isinstance(s, Series):
        raise HfErrorProxy("unique takes a Series")
    return s.unique()


def from_pandas(df: pandas.DataFrame) -> "DataFrame":
    return DataFrame(query_compiler=HipQueryCompiler.from_pandas(df))


class _HipPandasBase:
    _query_compiler: HipQueryCompiler

    # ---- reductions ----
    def _reduce(self, name, **kwargs):
        axis = kwargs.pop("axis", 0)
        numeric_only = kwargs.pop("numeric_only", False)
        qc = self._query_compiler
        if numeric_only:
            keep = [c for c in qc.columns
                    if isinstance(qc.dtypes[c], np.dtype)
                    and qc.dtypes[c] in (np.dtype(np.int64),
                                         np.dtype(np.float64))]
            qc = qc.getitem_column_array(keep)
        if axis in (1, "columns"):
            out = Series(query_compiler=qc.reduce_axis1(name), name=None)
            return out
        return self._lower(getattr(qc, name)(**kwargs))

    def sum(self, **kwargs):
        return self._reduce("sum", **kwargs)

    def mean(self, **kwargs):
        return self._reduce("mean", **kwargs)

    def count(self, **kwargs):
        return self._reduce("count", **kwargs)

    def min(self, **kwargs):
        return self._reduce("min", **kwargs)

    def max(self, **kwargs):
        return self._reduce("max", **kwargs)

    def clip(self, lower=None, upper=None):
        return self._rewrap(self._query_compiler.clip(lower, upper))

    def median(self, **kwargs):
        return self._lower(self._query_compiler.median())

    def isna(self):
        """Boolean NaN mask (inverse of notna)."""
        qc = self._query_compiler
        notna = qc.notna()
        one = type(notna).mul(notna, -1)
        inv = type(one).add(one, 1)  # 1 - notna
        out = self._rewrap(inv)
        out._bool_mask = True
        return out

    def notna(self):
        out = self._rewrap(self._query_compiler.notna())
        out._bool_mask = True
        return out

    def var(self, ddof: int = 1):
        return self._lower(self._query_compiler.var(ddof=ddof))

    def sem(self, ddof: int = 1):
        """pandas sem: std/sqrt(count), composed from the one-pass
        reduce partials."""
        qc = self._query_compiler
        std = qc.std(ddof=ddof)
        cnt = qc.count()
        return self._lower(std / np.sqrt(cnt.astype(float)))

    def rank(self, method: str = "average", ascending: bool = True,
             na_option: str = "keep"):
        """pandas rank(axis=0): always float64."""
        if na_option not in ("keep", "top", "bottom"):
            raise lib.HfError(f"rank: bad na_option {na_option!r}")
        return self._rewrap(self._query_compiler.rank(
            method=method, ascending=bool(ascending),
            na_option=na_option))

    def round(self, decimals: int = 0):  # noqa: A003
        """pandas round: half-even on float columns, ints unchanged."""
        return self._rewrap(self._query_compiler.round(int(decimals)))

    def where(self, cond, other=None):
        """pandas where(cond, other): cond is a boolean Series row mask;
        false rows become `other` (NaN default)."""
        if not isinstance(cond, Series):
            raise lib.HfError("where/mask take a boolean Series condition")
        return self._rewrap(self._query_compiler.where_mask(
            cond._query_compiler, other))

    def mask(self, cond, other=None):  # noqa: A003
        """pandas mask = where(~cond)."""
        if not isinstance(cond, Series):
            raise lib.HfError("where/mask take a boolean Series condition")
        return self.where(~cond, other)

    def duplicated(self, subset=None, keep="first"):
        """pandas duplicated(keep='first'|'last'|False): boolean Series;
        NaN keys compare equal (pandas semantics); 'last' rides the
        device row reversal."""
        qc = self._query_compiler.duplicated(subset, keep=keep)
        out = Series(query_compiler=qc, name=None)
        out._bool_mask = True
        return out

    def drop_duplicates(self, subset=None, keep="first"):
        """pandas drop_duplicates: original index labels of the kept
        rows (keep='first'|'last'|False)."""
        return self._rewrap(self._query_compiler.drop_duplicates(
            subset, keep=keep))

    def cumsum(self):
        return self._rewrap(self._query_compiler.cumsum())

    def cummin(self):
        return self._rewrap(self._query_compiler.cummin())

    def cummax(self):
        return self._rewrap(self._query_compiler.cummax())

    def cumprod(self):
        return self._rewrap(self._query_compiler.cumprod())

    def shift(self, periods: int = 1):
        return self._rewrap(self._query_compiler.shift(int(periods)))

    def diff(self, periods: int = 1):
        """x - x.shift(periods) (pandas diff)."""
        return self._rewrap(self._query_compiler.diff(int(periods)))

    def pct_change(self, periods: int = 1):
        """pandas pct_change(fill_method=None): x / x.shift(p) - 1."""
        return self / self.shift(periods) - 1

    def rolling(self, window: int, min_periods=None):
        """pandas rolling(window, min_periods): fixed forward-closed
        windows; .sum/.mean/.count/.min/.max."""
        return Rolling(self, int(window), min_periods)

    def expanding(self, min_periods: int = 1):
        """pandas expanding(min_periods); .sum/.mean/.count/.min/.max."""
        return Expanding(self, min_periods)

    def ffill(self):
        """pandas ffill (forward fill down the rows)."""
        return self._rewrap(self._query_compiler.fillna_directional(
            "ffill"))

    def bfill(self):
        """pandas bfill (backward fill down the rows)."""
        return self._rewrap(self._query_compiler.fillna_directional(
            "bfill"))

    def idxmax(self):
        return self._lower(self._query_compiler.idxmax())

    def idxmin(self):
        return self._lower(self._query_compiler.idxmin())

    def std(self, ddof: int = 1):
        return self._lower(self._query_compiler.std(ddof=ddof))

    # ---- arithmetic ----
    def __add__(self, other):
        return self._rewrap(type(self._query_compiler).add(self._query_compiler,
                                                           _unwrap(other)))

    def __radd__(self, other):
        return self.__add__(other)

    def __sub__(self, other):
        return self._rewrap(type(self._query_compiler).sub(self._query_compiler,
                                                           _unwrap(other)))

    def __rsub__(self, other):
        return self._rewrap(type(self._query_compiler).rsub(self._query_compiler,
                                                            _unwrap(other)))

    def __mul__(self, other):
        return self._rewrap(type(self._query_compiler).mul(self._query_compiler,
                                                           _unwrap(other)))

    def __rmul__(self, other):
        return self.__mul__(other)

    def __truediv__(self, other):
        return self._rewrap(type(self._query_compiler).truediv(self._query_compiler,
                                                               _unwrap(other)))

    def __rtruediv__(self, other):
        return self._rewrap(type(self._query_compiler).rtruediv(self._query_compiler,
                                                                _unwrap(other)))

    def _int_scalar_op(self, qc_name, other, opname):
        if not isinstance(other, (int, np.integer)) or other == 0:
            raise lib.HfError(f"{opname}: nonzero int scalar only "
                              "(float floor ops are a later round)")
        bad = [c for c, dt in self._query_compiler.dtypes.items()
               if dt != np.dtype(np.int64)]
        if bad:
            raise lib.HfError(f"{opname}: int64 columns only (got "
                              f"{bad})")
        return self._rewrap(getattr(type(self._query_compiler), qc_name)(
            self._query_compiler, int(other)))

    def __floordiv__(self, other):
        """// with an int scalar over int64 columns: HF_MAP_IDIV (Python
        floor semantics, exact toward -inf)."""
        return self._int_scalar_op("floordiv_int", other, "floordiv")

    def __mod__(self, other):
        """% with an int scalar over int64 columns: HF_MAP_IMOD (Python
        sign rule)."""
        return self._int_scalar_op("mod_int", other, "mod")

    def floordiv(self, other):
        return self.__floordiv__(other)

    def mod(self, other):
        return self.__mod__(other)

    def add(self, other):
        return self.__add__(other)

    def sub(self, other):
        return self.__sub__(other)

    def mul(self, other):
        return self.__mul__(other)

    def truediv(self, other):
        return self.__truediv__(other)

    def fillna(self, value):
        """pandas fillna: scalar, {column: scalar} dict, or a Timestamp
        for datetime Series (NaT -> value, exact int64 ns replace)."""
        import datetime as _dtm
        if isinstance(value, (pandas.Timestamp, np.datetime64,
                              _dtm.datetime)):
            qc = self._query_compiler
            if len(qc.columns) != 1:
                raise lib.HfError("fillna(Timestamp): Series only")
            from ..core.dataframe import INAT
            return self._rewrap(qc.map_dict(
                {int(INAT): int(pandas.Timestamp(value).value)},
                keep_missing=True))
        if isinstance(value, dict):
            return self._rewrap(
                self._query_compiler.fillna_dict(
                    {k: float(v) for k, v in value.items()}))
        return self._rewrap(type(self._query_compiler).fillna(self._query_compiler,
                                                              float(value)))

    # ---- comparisons -> boolean mask (int64 0/1 device column; to_pandas
    #      lowers to bool dtype) ----
    def _cmp(self, name, other):
        out = self._rewrap(getattr(self._query_compiler, name)(other))
        out._bool_mask = True
        return out

    def __gt__(self, other):
        return self._cmp("gt", other)

    def __ge__(self, other):
        return self._cmp("ge", other)

    def __lt__(self, other):
        return self._cmp("lt", other)

    def __le__(self, other):
        return self._cmp("le", other)

    def __eq__(self, other):  # noqa: A003 — pandas-style elementwise eq
        return self._cmp("eq", other)

    def __ne__(self, other):  # noqa: A003
        return self._cmp("ne", other)

    __hash__ = None  # elementwise __eq__ makes these unhashable, like pandas

    def notna(self):
        out = self._rewrap(self._query_compiler.notna())
        out._bool_mask = True
        return out

    def isna(self):
        return ~self.notna()

    # ---- boolean mask algebra (int64 0/1 masks) ----
    def __and__(self, other):
        out = self._rewrap(type(self._query_compiler).mul(
            self._query_compiler, _unwrap(other)))
        out._bool_mask = True
        return out

    def __or__(self, other):
        s = type(self._query_compiler).add(self._query_compiler, _unwrap(other))
        out = self._rewrap(s.ge(1))
        out._bool_mask = True
        return out

    def __invert__(self):
        out = self._rewrap(type(self._query_compiler).rsub(
            self._query_compiler, 1))
        out._bool_mask = True
        return out

    def dropna(self, how: str = "any", subset=None):
        """pandas dropna(axis=0): how='any' keeps rows with no NaN in
        the subset, how='all' drops rows where EVERY subset column is
        NaN (notna-OR mask)."""
        qc = self._query_compiler
        sel = qc
        if subset is not None:
            subset = [subset] if isinstance(subset, str) else list(subset)
            for c in subset:
                if c not in list(qc.columns):
                    raise lib.HfError(f"dropna: unknown column {c!r}")
            sel = qc.getitem_column_array(subset)
        if how == "any":
            mask_qc = sel.dropna_mask()
        elif how == "all":
            masks = sel.notna()  # per-column 0/1 frame
            acc = None
            for c in masks.columns:
                m = masks.getitem_column_array([c])
                if acc is None:
                    acc = m
                else:
                    t = type(acc).add(acc, m)
                    u = type(acc).mul(acc, m)
                    acc = type(t).sub(t, u)  # OR
            mask_qc = acc
        else:
            raise lib.HfError(f"dropna: bad how {how!r}")
        return self._rewrap(qc.getitem_array(mask_qc))

    def abs(self):
        return self._rewrap(type(self._query_compiler).abs(self._query_compiler,
                                                           None))

    def __getattr__(self, name):
        raise AttributeError(
            f"{type(self).__name__}.{name} is outside the HipNative hot-path "
            "scope (SURVEY.md §8) — not implemented in this round"
        )


def _unwrap(other):
    return other._query_compiler if isinstance(other, _HipPandasBase) else other


class _LocIndexer:
    """df.loc — the forms the hot path uses: boolean-mask rows (Series
    mask), optionally with a column list/name; ':' rows with columns;
    RangeIndex label slices (inclusive stop, pandas loc semantics)."""

    def __init__(self, df):
        self._df = df

    def __getitem__(self, key):
        rows, cols = key if isinstance(key, tuple) else (key, None)
        out = self._df
        if isinstance(rows, Series):
            out = out[rows]
        elif isinstance(rows, slice):
            if rows.start is None and rows.stop is None:
                pass
            else:
                idx = self._df.index
                if not isinstance(idx, pandas.RangeIndex) or \
                        idx.step != 1:
                    raise lib.HfError("loc: label slices need a RangeIndex")
                a = idx.start if rows.start is None else int(rows.start)
                b = (idx.stop - 1) if rows.stop is None else int(rows.stop)
                out = out._rewrap(out._query_compiler.take_row_range(
                    a - idx.start, b - idx.start + 1))
        else:
            raise lib.HfError(f"loc: unsupported row selector {type(rows)}")
        if cols is None:
            return out
        if isinstance(cols, str):
            return out[cols]
        return out[list(cols)]


class _ILocIndexer:
    """df.iloc — positional rows: integer slices (step 1/None) ride the
    device column slice, integer lists/arrays ride one device gather per
    column, a bare int returns the row as a pandas Series."""

    def __init__(self, df):
        self._df = df

    def __getitem__(self, key):
        import numpy as _np
        qc = self._df._query_compiler
        n = len(qc)
        if isinstance(key, slice):
            if key.step not in (None, 1):
                raise lib.HfError("iloc: slice step other than 1 is a "
                                  "later round")
            start, stop, _ = key.indices(n)
            return self._df._rewrap(qc.take_row_range(start, stop))
        if isinstance(key, (int, _np.integer)):
            pos = int(key) + (n if key < 0 else 0)
            row = DataFrame(
                query_compiler=qc.take_rows([pos])).to_pandas()
            if isinstance(self._df, Series):
                return row.iloc[0, 0]
            out = row.iloc[0]
            out.name = row.index[0]
            return out
        if isinstance(key, (list, _np.ndarray, pandas.Index)):
            return self._df._rewrap(qc.take_rows(_np.asarray(key)))
        raise lib.HfError(f"iloc: unsupported selector {type(key)}")


class DataFrame(_HipPandasBase):
    def __init__(self, data=None, query_compiler=None):
        if query_compiler is not None:
            self._query_compiler = query_compiler
            return
        if isinstance(data, pandas.DataFrame):
            pdf = data
        elif isinstance(data, dict):
            pdf = pandas.DataFrame(data)
        else:
            raise lib.HfError(
                "DataFrame accepts a dict of columns or a pandas.DataFrame"
            )
        self._query_compiler = HipQueryCompiler.from_pandas(pdf)

    def _rewrap(self, qc):
        return DataFrame(query_compiler=qc)

    def _lower(self, series):
        return series  # reductions come back as pandas.Series already

    @property
    def columns(self):
        return self._query_compiler.columns

    @property
    def index(self):
        return self._query_compiler.index

    @property
    def dtypes(self):
        return self._query_compiler.dtypes

    @property
    def shape(self):
        return (len(self._query_compiler), len(self.columns))

    def __len__(self):
        return len(self._query_compiler)

    def __getitem__(self, key):
        if isinstance(key, str):
            return Series(query_compiler=self._query_compiler.getitem_column_array(
                [key]), name=key)
        if isinstance(key, Series):  # boolean row mask: df[df.v > x]
            return DataFrame(
                query_compiler=self._query_compiler.getitem_array(
                    key._query_compiler)
            )
        if isinstance(key, (list, tuple, pandas.Index)):
            return DataFrame(
                query_compiler=self._query_compiler.getitem_column_array(list(key))
            )
        raise lib.HfError("only column selection / boolean masks are supported")

    def _value_qc(self, value):
        """Coerce a __setitem__/insert value to a 1-column qc (POSITIONAL
        assignment; equal length enforced device-side).  Host arrays /
        pandas Series upload through from_pandas (strings dictionary-
        encode, datetimes tag)."""
        if isinstance(value, Series):
            return value._query_compiler
        if isinstance(value, pandas.Series):
            value = value.reset_index(drop=True)
        elif isinstance(value, (list, np.ndarray, pandas.Index)):
            value = pandas.Series(np.asarray(value))
        else:
            raise lib.HfError(
                "setitem value must be a Series, array, list or scalar")
        return HipQueryCompiler.from_pandas(value.to_frame(name="\x00v\x00"))

    def __setitem__(self, key: str, value):
        """df[col] = value — replace-or-append, POSITIONAL (length must
        match; pandas' index-alignment beyond that is not replicated).
        Scalars broadcast device-side with no host array; a boolean-mask
        Series lands as int64 0/1 (our bool carrier)."""
        if not isinstance(key, str):
            raise lib.HfError("setitem key must be a column name")
        if (np.isscalar(value) or value is None
                or isinstance(value, (bool, np.bool_))):
            self._query_compiler = \
                self._query_compiler.write_scalar_column(key, value)
            return
        self._query_compiler = self._query_compiler.write_column(
            key, self._value_qc(value))

    def insert(self, loc: int, column: str, value):
        """pandas DataFrame.insert: new column at position ``loc``."""
        if column in list(self.columns):
            raise lib.HfError(f"insert: column {column!r} already exists")
        self[column] = value
        cols = [c for c in self.columns if c != column]
        cols.insert(int(loc), column)
        self._query_compiler = \
            self._query_compiler.getitem_column_array(cols)

    def assign(self, **kwargs) -> "DataFrame":
        """pandas DataFrame.assign over non-callable values."""
        out = DataFrame(query_compiler=self._query_compiler)
        for k, v in kwargs.items():
            if callable(v):
                v = v(out)
            out[k] = v
        return out

    @property
    def iloc(self):
        return _ILocIndexer(self)

    @property
    def loc(self):
        return _LocIndexer(self)

    def sort_index(self, ascending: bool = True):
        return self._rewrap(
            self._query_compiler.sort_index(bool(ascending)))

    def head(self, n: int = 5):
        total = len(self._query_compiler)
        stop = max(0, total + n) if n < 0 else n  # pandas head(-n)
        return DataFrame(
            query_compiler=self._query_compiler.take_row_range(0, stop))

    def tail(self, n: int = 5):
        total = len(self._query_compiler)
        start = min(total, -n) if n < 0 else total - n  # pandas tail(-n)
        return DataFrame(
            query_compiler=self._query_compiler.take_row_range(start,
                                                               total))

    def astype(self, dtype):
        """Scalar dtype (all columns) or {column: dtype} (per-column
        device casts through write_column)."""
        if isinstance(dtype, dict):
            qc = self._query_compiler
            for c, dt in dtype.items():
                if c not in list(self.columns):
                    raise lib.HfError(f"astype: unknown column {c!r}")
                qc = qc.write_column(
                    c, qc.getitem_column_array([c]).astype(dt))
            return DataFrame(query_compiler=qc)
        return DataFrame(query_compiler=self._query_compiler.astype(dtype))

    def quantile(self, q=0.5):
        """Per-column quantiles (linear interpolation, NaN skipped)."""
        qs = [q] if np.isscalar(q) else list(q)
        out = self._query_compiler.quantile(qs)
        if np.isscalar(q):
            return out.iloc[0]
        return out

    def describe(self):
        """count/mean/std/min/25%/50%/75%/max per numeric column (pandas
        describe), composed from the one-pass reduce and the sorted
        quantile machinery."""
        qc = self._query_compiler
        cnt = qc.count()
        mean = qc.mean()
        std = qc.std()
        mn = qc.min()
        mx = qc.max()
        qs = qc.quantile([0.25, 0.5, 0.75])
        rows = {"count": cnt.astype(float), "mean": mean, "std": std,
                "min": mn, "25%": qs.iloc[0], "50%": qs.iloc[1],
                "75%": qs.iloc[2], "max": mx}
        return pandas.DataFrame(rows).T[list(self.columns)]

    def drop(self, columns=None):
        """pandas DataFrame.drop(columns=...): metadata-only column
        removal (device blocks re-select lazily)."""
        if columns is None:
            raise lib.HfError("drop: only drop(columns=...) is supported")
        drop_set = {columns} if isinstance(columns, str) else set(columns)
        missing = drop_set - set(self.columns)
        if missing:
            raise lib.HfError(f"drop: columns not found: {sorted(missing)}")
        keep = [c for c in self.columns if c not in drop_set]
        return self[keep]

    def nunique(self):
        """Per-column distinct count (NaN excluded), as a pandas Series."""
        vals = {}
        for c in self.columns:
            vals[c] = Series(
                query_compiler=self._query_compiler.getitem_column_array(
                    [c]), name=c).nunique()
        return pandas.Series(vals, dtype=np.int64)

    def to_parquet(self, path):
        """Round-trip columnar write: device columns -> pyarrow -> parquet
        (strings rebuilt as dictionary arrays from the codes, so no per-row
        Python objects either direction)."""
        from ..io import write_parquet
        write_parquet(self._query_compiler, path)

    def to_csv(self, path):
        """Columnar CSV write through pyarrow.csv (the C++ writer), same
        device->arrow path as to_parquet."""
        from ..io import write_csv
        write_csv(self._query_compiler, path)

    def rename(self, columns: dict):
        qc = self._query_compiler
        frame = qc._modin_frame
        new_cols = [columns.get(c, c) for c in frame.columns]
        # metadata-only: device blocks are re-labelled lazily on access
        from ..core.dataframe import HipDataframe
        from ..core.partition import DeviceBlock, HipDataframePartition

        def relabel(block):
            return DeviceBlock(
                {columns.get(n, n): c for n, c in block.columns.items()},
                block.length,
                {columns.get(n, n): c for n, c in block.cats.items()})
        parts = [p.add_to_apply_calls(relabel) for p in frame._partitions]
        import pandas as _pd
        nf = HipDataframe(parts, frame._index, new_cols, frame._row_lengths,
                          _pd.Series({columns.get(n, n): d
                                      for n, d in frame.dtypes.items()}))
        return DataFrame(query_compiler=type(qc)(nf))

    def reset_index(self, drop: bool = False):
        qc = self._query_compiler
        frame = qc._modin_frame
        from ..core.dataframe import HipDataframe
        if not drop:
            # pandas: the old index becomes the leading column
            idx = self.index
            name = idx.name if idx.name is not None else "index"
            if name in list(self.columns):
                raise lib.HfError(
                    f"reset_index: column {name!r} already exists")
            out = DataFrame(query_compiler=qc)
            vals = np.asarray(idx)
            if vals.dtype == object:
                out[name] = pandas.Series(vals, dtype=object)
            else:
                out[name] = vals
            cols = [name] + [c for c in out.columns if c != name]
            qc2 = out._query_compiler.getitem_column_array(cols)
            qc2._modin_frame._index = pandas.RangeIndex(len(frame))
            return DataFrame(query_compiler=qc2)
        nf = HipDataframe(frame._partitions, pandas.RangeIndex(len(frame)),
                          frame.columns, frame._row_lengths, frame.dtypes)
        return DataFrame(query_compiler=type(qc)(nf))

    def set_index(self, keys: str, drop: bool = True):
        """pandas set_index(column): the column BECOMES the index as a
        lazy DeviceIndex (no host materialization until the index is
        read); dictionary columns materialize through their cats,
        datetime columns keep the dtype tag."""
        if not isinstance(keys, str) or keys not in list(self.columns):
            raise lib.HfError("set_index: one existing column name")
        from ..core.dataframe import DeviceIndex, HipDataframe
        frame = self._query_compiler._modin_frame
        cols = [p.block().columns[keys] for p in frame._partitions]
        col = cols[0] if len(cols) == 1 else lib.concat(cols)
        cats = (frame._partitions[0].block().cats.get(keys)
                if frame._partitions else None)
        didx = DeviceIndex(col, name=keys, cats=cats)
        names = ([c for c in frame.columns if c != keys] if drop
                 else list(frame.columns))
        nf = frame.take_columns(names)
        nf = HipDataframe(nf._partitions, didx, names, nf._row_lengths,
                          nf.dtypes)
        dt = frame.dtypes[keys]
        if isinstance(dt, np.dtype) and np.issubdtype(dt, np.datetime64):
            nf._index_dtype = dt
        return DataFrame(
            query_compiler=type(self._query_compiler)(nf))

    def sort_values(self, by: str, ascending: bool = True,
                    kind: str = "stable", na_position: str = "last"):
        """Always stable (equals pandas sort_values(kind='stable'), a
        stronger guarantee than the default quicksort).  na_position
        'last'/'first' as pandas."""
        return DataFrame(
            query_compiler=self._query_compiler.sort_rows_by_column_values(
                by, ascending, na_position=na_position))

    def merge(self, other: "DataFrame", on: str = None,
              how: str = "inner", left_on=None, right_on=None):
        """Merge on an int64/float64/string key column (modin/pandas API
        -> qc.merge -> broadcast-right device join); how='cross' takes no
        key (cartesian product); left_on/right_on keep both key columns
        (pandas keep-keys rule)."""
        return DataFrame(query_compiler=self._query_compiler.merge(
            other._query_compiler, on=on, how=how,
            left_on=left_on, right_on=right_on))

    def corr(self) -> pandas.DataFrame:
        """pandas corr (Pearson, pairwise-complete rows): masked moments
        from NaN-propagating device passes; k x k host combine."""
        return self._query_compiler.corr()

    def cov(self, ddof: int = 1) -> pandas.DataFrame:
        """pandas cov (pairwise-complete rows, ddof=1)."""
        return self._query_compiler.cov(ddof=ddof)

    def melt(self, id_vars=None, value_vars=None, var_name=None,
             value_name: str = "value") -> "DataFrame":
        """pandas melt: wide -> long, entirely on device — per value
        column a projection + zero-copy value alias + device-filled
        `variable` column, then one concat (dictionary union recodes the
        variable column's single-entry cats).  Mixed int64/float64 value
        columns promote to float64 (pandas concat rule); mixing numeric
        with string/datetime value columns is loud."""
        qc = self._query_compiler
        cols = list(self.columns)
        if id_vars is None:
            id_vars = []
        elif isinstance(id_vars, str):
            id_vars = [id_vars]
        else:
            id_vars = list(id_vars)
        if value_vars is None:
            value_vars = [c for c in cols if c not in id_vars]
        elif isinstance(value_vars, str):
            value_vars = [value_vars]
        else:
            value_vars = list(value_vars)
        var_name = var_name or "variable"
        if not value_vars:
            raise lib.HfError("melt: no value columns")
        for c in id_vars + value_vars:
            if c not in cols:
                raise lib.HfError(f"melt: unknown column {c!r}")
        if value_name in id_vars or var_name in id_vars \
                or var_name == value_name:
            raise lib.HfError("melt: var_name/value_name collide with "
                              "id_vars")
        dts = self.dtypes
        vset = {str(dts[v]) for v in value_vars}
        cast_f64 = False
        if len(vset) > 1:
            if vset <= {"int64", "float64"}:
                cast_f64 = True
            else:
                raise lib.HfError(
                    f"melt: value columns mix incompatible dtypes {vset}")
        pieces = []
        for v in value_vars:
            sub = qc.getitem_column_array(id_vars + [v])
            if cast_f64 and str(dts[v]) == "int64":
                sub = sub.write_column(
                    v, qc.getitem_column_array([v]).astype(np.float64))
            sub = sub.rename_columns({v: value_name})
            sub = sub.write_scalar_column(var_name, v)
            pieces.append(sub.getitem_column_array(
                id_vars + [var_name, value_name]))
        out = pieces[0].concat(pieces[1:]) if len(pieces) > 1 else pieces[0]
        out._modin_frame._index = pandas.RangeIndex(len(out))
        return DataFrame(query_compiler=out)

    def pivot_table(self, values=None, index=None, columns=None,
                    aggfunc: str = "mean", fill_value=None) -> "DataFrame":
        """pandas pivot_table over single index/columns/values names:
        the aggregation runs as a device multi-key groupby reduce; only
        the REDUCED ngroups-sized table is reshaped host-side
        (unstack)."""
        for arg, nm in ((values, "values"), (index, "index"),
                        (columns, "columns")):
            if not isinstance(arg, str) or arg not in list(self.columns):
                raise lib.HfError(
                    f"pivot_table: {nm} must name one column")
        sub = self[[index, columns, values]]
        red = sub.groupby([index, columns]).agg(aggfunc).to_pandas()
        wide = red[values].unstack(level=-1)
        if fill_value is not None:
            wide = wide.fillna(fill_value)
        return DataFrame(wide)

    def groupby(self, by, as_index: bool = True,
                dropna: bool = True) -> "DataFrameGroupBy":
        bys = list(by) if isinstance(by, (list, tuple)) else [by]
        for b in bys:
            if not isinstance(b, str) or b not in list(self.columns):
                raise lib.HfError(
                    "groupby(by=<column name> | [column names]) only")
        return DataFrameGroupBy(self, by, as_index=as_index, dropna=dropna)

    def filter(self, items=None, like: str = None,  # noqa: A003
               regex: str = None) -> "DataFrame":
        """pandas filter(axis=1): column selection by list, substring or
        regex (metadata-only)."""
        import re
        given = sum(x is not None for x in (items, like, regex))
        if given != 1:
            raise lib.HfError("filter: exactly one of items/like/regex")
        if items is not None:
            keep = [c for c in self.columns if c in set(items)]
        elif like is not None:
            keep = [c for c in self.columns if like in str(c)]
        else:
            rx = re.compile(regex)
            keep = [c for c in self.columns if rx.search(str(c))]
        return DataFrame(
            query_compiler=self._query_compiler.getitem_column_array(keep))

    def select_dtypes(self, include=None, exclude=None) -> "DataFrame":
        """pandas select_dtypes over this backend's dtype set (int64,
        float64, object/strings, datetime64[ns])."""
        def norm(spec):
            if spec is None:
                return None
            spec = [spec] if not isinstance(spec, (list, tuple)) else spec
            out = set()
            for x in spec:
                if x in (object, "object", str, "str"):
                    out.add("object")
                elif x in ("number", np.number):
                    out.update(("int64", "float64"))
                elif x in ("datetime", "datetime64", "datetime64[ns]",
                           np.datetime64):
                    out.add("datetime64[ns]")
                else:
                    out.add(str(np.dtype(x)))
            return out

        inc, exc = norm(include), norm(exclude)
        keep = []
        for c in self.columns:
            dt = str(self.dtypes[c])
            if inc is not None and dt not in inc:
                continue
            if exc is not None and dt in exc:
                continue
            keep.append(c)
        return DataFrame(
            query_compiler=self._query_compiler.getitem_column_array(keep))

    def query(self, expr: str) -> "DataFrame":
        """pandas query over the supported mask algebra: comparisons of
        columns against literals or other columns (col-col rides the
        subtract-compare composition), &/|/~ and and/or/not,
        parentheses.  Everything else raises loudly."""
        import ast
        try:
            tree = ast.parse(expr, mode="eval").body
        except SyntaxError as e:
            raise lib.HfError(f"query: cannot parse {expr!r}: {e}")
        OPS = {ast.Gt: "__gt__", ast.GtE: "__ge__", ast.Lt: "__lt__",
               ast.LtE: "__le__", ast.Eq: "__eq__", ast.NotEq: "__ne__"}
        FLIP = {"__gt__": "__lt__", "__ge__": "__le__",
                "__lt__": "__gt__", "__le__": "__ge__",
                "__eq__": "__eq__", "__ne__": "__ne__"}

        def value(n):
            if isinstance(n, ast.Name):
                if n.id not in list(self.columns):
                    raise lib.HfError(f"query: unknown column {n.id!r}")
                return self[n.id]
            if isinstance(n, ast.Constant):
                return n.value
            if isinstance(n, ast.UnaryOp) \
                    and isinstance(n.op, ast.USub) \
                    and isinstance(n.operand, ast.Constant):
                return -n.operand.value
            raise lib.HfError("query: operands must be column names or "
                              "literals")

        def build(node):
            if isinstance(node, ast.BoolOp):
                ms = [build(v) for v in node.values]
                acc = ms[0]
                for m in ms[1:]:
                    acc = (acc & m if isinstance(node.op, ast.And)
                           else acc | m)
                return acc
            if isinstance(node, ast.UnaryOp) \
                    and isinstance(node.op, ast.Not):
                return ~build(node.operand)
            if isinstance(node, ast.BinOp) \
                    and isinstance(node.op, (ast.BitAnd, ast.BitOr)):
                a, b = build(node.left), build(node.right)
                return a & b if isinstance(node.op, ast.BitAnd) else a | b
            if isinstance(node, ast.Compare):
                if len(node.ops) != 1:
                    raise lib.HfError("query: chained comparisons are a "
                                      "later round")
                opn = OPS.get(type(node.ops[0]))
                if opn is None:
                    raise lib.HfError("query: unsupported comparison")
                a, b = value(node.left), value(node.comparators[0])
                if isinstance(a, Series) and isinstance(b, Series):
                    # col-col: compare the difference against 0 (NaN
                    # rows propagate to False, the pandas rule)
                    return getattr(a - b, opn)(0)
                if isinstance(b, Series):
                    a, b, opn = b, a, FLIP[opn]
                return getattr(a, opn)(b)
            raise lib.HfError("query: unsupported expression "
                              f"{ast.dump(node)[:60]}")

        return self[build(tree)]

    @property
    def empty(self) -> bool:
        return len(self) == 0 or len(self.columns) == 0

    @property
    def size(self) -> int:
        return len(self) * len(self.columns)

    @property
    def ndim(self) -> int:
        return 2

    @property
    def values(self) -> np.ndarray:
        return self.to_pandas().to_numpy()

    def to_numpy(self) -> np.ndarray:
        return self.to_pandas().to_numpy()

    def copy(self) -> "DataFrame":
        """Columns are immutable device buffers; a copy is a new frame
        over the same ColumnRefs (copy-on-write by construction)."""
        frame = self._query_compiler._modin_frame
        from ..core.dataframe import HipDataframe
        nf = HipDataframe(list(frame._partitions), frame._index,
                          list(frame.columns), list(frame._row_lengths),
                          frame.dtypes.copy())
        return DataFrame(query_compiler=type(self._query_compiler)(nf))

    def equals(self, other) -> bool:
        if not isinstance(other, DataFrame):
            return False
        a, b = self.to_pandas(), other.to_pandas()
        return a.equals(b)

    def keys(self):
        return self.columns

    def items(self):
        for c in self.columns:
            yield c, self[c]

    def take(self, indices) -> "DataFrame":
        """pandas take(axis=0): positional row gather."""
        return self.iloc[list(indices)]

    def add_prefix(self, prefix: str) -> "DataFrame":
        return self.rename(columns={c: f"{prefix}{c}"
                                    for c in self.columns})

    def add_suffix(self, suffix: str) -> "DataFrame":
        return self.rename(columns={c: f"{c}{suffix}"
                                    for c in self.columns})

    def pop(self, col: str) -> "Series":
        """pandas pop: return the column and drop it IN PLACE."""
        out = self[col]
        self._query_compiler = self._query_compiler.getitem_column_array(
            [c for c in self.columns if c != col])
        return out

    def get(self, key, default=None):
        return self[key] if key in list(self.columns) else default

    def squeeze(self):
        if len(self.columns) == 1:
            return self[self.columns[0]]
        return self

    def sample(self, n: int = None, frac: float = None,
               random_state=None) -> "DataFrame":
        """pandas sample(replace=False): device-side draw (one uniform
        key per row; the n smallest win) — the row VALUES never leave
        the GPU.  The permutation differs from pandas' MT19937 stream
        for a given random_state (documented deviation; the sample is
        still uniform without replacement)."""
        if (n is None) == (frac is None):
            raise lib.HfError("sample: exactly one of n/frac")
        if frac is not None:
            n = int(round(frac * len(self)))
        seed = 0x5A11 if random_state is None else int(random_state)
        return DataFrame(
            query_compiler=self._query_compiler.sample_rows(n, seed))

    def nlargest(self, n: int, columns: str):
        """pandas nlargest(keep='first'): stable descending NaN-last sort
        + head(n) (NaN rows only appear once n exceeds the non-NaN
        count — pandas behavior)."""
        return self.sort_values(columns, ascending=False).head(int(n))

    def nsmallest(self, n: int, columns: str):
        return self.sort_values(columns, ascending=True).head(int(n))

    def to_pandas(self) -> pandas.DataFrame:
        return self._query_compiler.to_pandas()

    def _to_pandas(self) -> pandas.DataFrame:  # reference-compatible alias
        return self.to_pandas()

    def __repr__(self):
        return f"modin_amd.DataFrame({self.shape[0]}x{self.shape[1]} on device)"


class _StrAccessor:
    """Series.str — host-dictionary transforms + one device gather."""

    def __init__(self, s: "Series"):
        self._s = s

    def _wrap(self, qc, bool_mask=False):
        out = Series(query_compiler=qc, name=self._s.name)
        out._bool_mask = bool_mask
        return out

    def len(self):  # noqa: A003
        return self._wrap(self._s._query_compiler.str_op("len"))

    def lower(self):
        return self._wrap(self._s._query_compiler.str_op("lower"))

    def upper(self):
        return self._wrap(self._s._query_compiler.str_op("upper"))

    def contains(self, pat, regex: bool = False, na=None):
        return self._wrap(self._s._query_compiler.str_op(
            "contains", pat=pat, na=na, regex=regex),
            bool_mask=na is not None)

    def match(self, pat, na=None):
        return self._wrap(self._s._query_compiler.str_op("match",
                                                         pat=pat, na=na),
                          bool_mask=na is not None)

    def fullmatch(self, pat, na=None):
        return self._wrap(self._s._query_compiler.str_op("fullmatch",
                                                         pat=pat, na=na),
                          bool_mask=na is not None)

    def replace(self, pat, repl, regex: bool = True):
        """pandas Series.str.replace (regex default True, pandas 2)."""
        return self._wrap(self._s._query_compiler.str_op(
            "replace", pat=pat, repl=repl, regex=regex))

    def strip(self):
        return self._wrap(self._s._query_compiler.str_op("strip"))

    def lstrip(self):
        return self._wrap(self._s._query_compiler.str_op("lstrip"))

    def rstrip(self):
        return self._wrap(self._s._query_compiler.str_op("rstrip"))

    def title(self):
        return self._wrap(self._s._query_compiler.str_op("title"))

    def capitalize(self):
        return self._wrap(self._s._query_compiler.str_op("capitalize"))

    def zfill(self, width: int):
        return self._wrap(self._s._query_compiler.str_op("zfill",
                                                         width=width))

    def startswith(self, pat, na=None):
        return self._wrap(self._s._query_compiler.str_op("startswith",
                                                         pat=pat, na=na),
                          bool_mask=na is not None)

    def endswith(self, pat, na=None):
        return self._wrap(self._s._query_compiler.str_op("endswith",
                                                         pat=pat, na=na),
                          bool_mask=na is not None)


class _DtAccessor:
    """Series.dt — calendar fields of datetime64[ns] typed columns,
    computed on device with exact int64 calendar math (dataframe.dt_field)."""

    def __init__(self, s: "Series"):
        self._s = s

    def _field(self, f: str) -> "Series":
        return Series(query_compiler=self._s._query_compiler.dt_field(f),
                      name=self._s.name)

    _FREQ_NS = {"D": 86_400 * 10**9, "h": 3_600 * 10**9, "H": 3_600 * 10**9,
                "min": 60 * 10**9, "T": 60 * 10**9, "s": 10**9,
                "S": 10**9, "ms": 10**6, "us": 10**3, "ns": 1}

    def floor(self, freq: str) -> "Series":
        """pandas Series.dt.floor: truncate to the unit (D/h/min/s/ms/
        us; NaT passes through)."""
        unit = self._FREQ_NS.get(freq)
        if unit is None:
            raise lib.HfError(f"dt.floor: unsupported freq {freq!r}")
        return Series(
            query_compiler=self._s._query_compiler.dt_floor(unit),
            name=self._s.name)

    def normalize(self) -> "Series":
        """pandas Series.dt.normalize == floor('D')."""
        return self.floor("D")

    @property
    def year(self):
        return self._field("year")

    @property
    def month(self):
        return self._field("month")

    @property
    def day(self):
        return self._field("day")

    @property
    def hour(self):
        return self._field("hour")

    @property
    def minute(self):
        return self._field("minute")

    @property
    def second(self):
        return self._field("second")

    @property
    def dayofweek(self):
        return self._field("dayofweek")

    weekday = dayofweek


class Series(_HipPandasBase):
    _bool_mask = False  # comparisons set this: to_pandas lowers int64->bool

    def __init__(self, data=None, query_compiler=None, name=None):
        self.name = name
        if query_compiler is not None:
            self._query_compiler = query_compiler
            return
        if isinstance(data, pandas.Series):
            pdf = data.to_frame(name=data.name or 0)
            self.name = data.name
        else:
            raise lib.HfError("Series accepts a pandas.Series")
        self._query_compiler = HipQueryCompiler.from_pandas(pdf)

    @property
    def iloc(self):
        return _ILocIndexer(self)

    @property
    def dt(self) -> "_DtAccessor":
        return _DtAccessor(self)

    @property
    def str(self) -> "_StrAccessor":  # noqa: A003
        return _StrAccessor(self)

    def _rewrap(self, qc):
        return Series(query_compiler=qc, name=self.name)

    def _lower(self, series):
        # a Series reduction is a scalar
        return series.iloc[0]

    def __len__(self):
        return len(self._query_compiler)

    def __getitem__(self, key):
        """Boolean-mask selection: s[s > 0] (pandas Series mask form)."""
        if isinstance(key, Series):
            return Series(
                query_compiler=self._query_compiler.getitem_array(
                    key._query_compiler), name=self.name)
        raise lib.HfError("Series supports boolean-mask selection only")

    def dropna(self) -> "Series":
        qc = self._query_compiler
        return Series(query_compiler=qc.getitem_array(qc.notna()),
                      name=self.name)

    def map(self, arg) -> "Series":  # noqa: A003
        """pandas Series.map(dict): unmapped values -> NaN.  Dictionary
        columns remap host-side + one gather; int64 columns via device
        binary search over the key set."""
        if isinstance(arg, pandas.Series):
            arg = arg.to_dict()
        if not isinstance(arg, dict):
            raise lib.HfError("Series.map accepts a dict (callables are "
                              "host-bound; a later round)")
        return self._rewrap(self._query_compiler.map_dict(
            arg, keep_missing=False))

    def replace(self, to_replace, value=None) -> "Series":
        """pandas Series.replace(scalar, scalar): where(self != a, b) —
        NaN rows survive untouched (NaN != a is True).  replace(dict):
        unmapped values keep their value (device LUT path)."""
        if isinstance(to_replace, dict):
            if value is not None:
                raise lib.HfError("replace(dict) takes no value")
            return self._rewrap(self._query_compiler.map_dict(
                to_replace, keep_missing=True))
        if not (np.isscalar(to_replace) and np.isscalar(value)):
            raise lib.HfError("replace: scalar to_replace/value only "
                              "this round")
        if isinstance(to_replace, float) and np.isnan(to_replace):
            return self.fillna(value)  # pandas replace(nan, x) == fillna
        return self.where(self != to_replace, value)

    def between(self, left, right, inclusive: str = "both") -> "Series":
        """pandas Series.between: boolean mask (NaN -> False)."""
        if inclusive == "both":
            out = (self >= left) & (self <= right)
        elif inclusive == "neither":
            out = (self > left) & (self < right)
        elif inclusive == "left":
            out = (self >= left) & (self < right)
        elif inclusive == "right":
            out = (self > left) & (self <= right)
        else:
            raise lib.HfError("between: bad 'inclusive'")
        out._bool_mask = True
        return out

    def sort_values(self, ascending: bool = True, kind: str = "stable",
                    na_position: str = "last"):
        """pandas Series.sort_values (always stable)."""
        name = list(self._query_compiler._modin_frame.columns)[0]
        qc = self._query_compiler.sort_rows_by_column_values(
            name, ascending, na_position=na_position)
        return Series(query_compiler=qc, name=self.name)

    def head(self, n: int = 5):
        total = len(self._query_compiler)
        stop = max(0, total + n) if n < 0 else n  # pandas head(-n)
        return Series(
            query_compiler=self._query_compiler.take_row_range(0, stop),
            name=self.name)

    def tail(self, n: int = 5):
        total = len(self._query_compiler)
        start = min(total, -n) if n < 0 else total - n  # pandas tail(-n)
        return Series(
            query_compiler=self._query_compiler.take_row_range(start,
                                                               total),
            name=self.name)

    def to_frame(self, name=None) -> "DataFrame":
        """pandas Series.to_frame: 1-column DataFrame (zero copy)."""
        qc = self._query_compiler
        cur = list(qc._modin_frame.columns)[0]
        want = name if name is not None else (
            self.name if self.name is not None else 0)
        if want != cur:
            qc = qc.rename_columns({cur: want})
        return DataFrame(query_compiler=qc)

    def astype(self, dtype) -> "Series":
        return Series(query_compiler=self._query_compiler.astype(dtype),
                      name=self.name)

    def quantile(self, q=0.5):
        """pandas Series.quantile: scalar for scalar q, Series for a
        list (linear interpolation, NaN skipped)."""
        qs = [q] if np.isscalar(q) else list(q)
        out = self._query_compiler.quantile(qs)
        col = out.columns[0]
        return float(out[col].iloc[0]) if np.isscalar(q) else out[col]

    def any(self) -> bool:  # noqa: A003
        """pandas Series.any (skipna): non-NaN nonzero exists =
        count(notna) - count(== 0) > 0 (NaN == 0 is False on device, so
        the zero count never includes NaN rows)."""
        nz = (self == 0)._query_compiler.sum().iloc[0]
        nn = self._query_compiler.notna().sum().iloc[0]
        return bool(nn - nz > 0)

    def all(self) -> bool:  # noqa: A003
        """pandas Series.all over the 0/1 mask (NaN counts truthy, the
        pandas rule)."""
        qc = (self == 0)._query_compiler
        s = qc.sum()
        return bool(s.iloc[0] == 0)

    def mode(self) -> "Series":
        """pandas Series.mode: every value at the max multiplicity,
        sorted — composed from the device value_counts (host reshape of
        the ngroups-sized result only)."""
        vc = self.value_counts()
        vals = np.asarray(vc.index)
        cnts = np.asarray(vc)
        if not len(cnts):
            return Series(pandas.Series([], dtype=np.float64,
                                        name=self.name))
        best = vals[cnts == cnts.max()]
        return Series(pandas.Series(np.sort(best), name=self.name))

    def nlargest(self, n: int = 5):
        """pandas Series.nlargest(keep='first'): stable descending
        NaN-last sort + head(n)."""
        return self.sort_values(ascending=False).head(int(n))

    def nsmallest(self, n: int = 5):
        return self.sort_values(ascending=True).head(int(n))

    def unique(self):
        """pandas Series.unique: distinct values in FIRST-APPEARANCE order,
        NaN included at its appearance position (device groupby-min over
        global row positions)."""
        st = self._query_compiler.distinct_stats()
        order = np.argsort(st["firstpos"], kind="stable")
        vals = list(np.asarray(st["values"], dtype=object)[order])
        pos = list(st["firstpos"][order])
        if st["nan_count"]:
            i = int(np.searchsorted(np.asarray(pos), st["nan_firstpos"]))
            vals.insert(i, np.nan)
        if all(isinstance(v, (int, np.integer)) for v in vals):
            return np.array(vals, dtype=np.int64)
        return np.array(vals, dtype=object)

    def value_counts(self, normalize: bool = False):
        """pandas Series.value_counts: the device groupby supplies distinct
        values + counts + first-appearance positions; the final count-desc
        sort over the (small) distinct set is delegated to pandas
        sort_values on the appearance-ordered counts — bit-identical tie
        order with pandas (which starts from its hashtable's appearance
        order and quicksorts).  normalize=True divides by the non-NaN
        total (pandas 'proportion')."""
        st = self._query_compiler.distinct_stats()
        order = np.argsort(st["firstpos"], kind="stable")
        idx = pandas.Index(np.asarray(st["values"], dtype=object)[order],
                           name=self.name)
        counts = st["counts"][order].astype(np.int64)
        if normalize:
            tot = counts.sum()
            pre = pandas.Series(counts / tot if tot else counts * 0.0,
                                index=idx, name="proportion")
        else:
            pre = pandas.Series(counts, index=idx, name="count")
        return pre.sort_values(ascending=False)

    def nunique(self) -> int:
        return int(len(self._query_compiler.distinct_stats()["values"]))

    def isin(self, values) -> "Series":
        """Membership mask composed from EQ compares, OR-folded as
        a+b-a*b over the 0/1 masks (<= 64 values; string Series translate
        through the dictionary; a NaN entry matches NaN rows via an
        isna() mask — the pandas rule)."""
        values = list(values)
        if len(values) > 64:
            raise lib.HfError("isin supports up to 64 values this round")
        has_nan = any(isinstance(v, float) and v != v for v in values)
        values = [v for v in values
                  if not (isinstance(v, float) and v != v)]
        qc = self._query_compiler
        if not values and not has_nan:
            acc = qc.eq(float("inf"))  # all False (NaN == inf is False too)
        else:
            acc = None
            masks = [qc.eq(v) for v in values]
            if has_nan:
                # pandas isin: NaN in the value list matches NaN rows
                notna = qc.notna()
                one = type(notna).mul(notna, -1)
                masks.append(type(one).add(one, 1))  # 1 - notna
            for m in masks:
                if acc is None:
                    acc = m
                else:  # OR of 0/1 masks: a + b - a*b
                    t = type(acc).add(acc, m)
                    u = type(acc).mul(acc, m)
                    acc = type(t).sub(t, u)
        out = Series(query_compiler=acc, name=self.name)
        out._bool_mask = True
        return out

    @property
    def empty(self) -> bool:
        return len(self) == 0

    @property
    def size(self) -> int:
        return len(self)

    @property
    def ndim(self) -> int:
        return 1

    @property
    def values(self) -> np.ndarray:
        return self.to_pandas().to_numpy()

    def to_numpy(self) -> np.ndarray:
        return self.to_pandas().to_numpy()

    def copy(self) -> "Series":
        return Series(query_compiler=self._query_compiler,
                      name=self.name)

    def equals(self, other) -> bool:
        if not isinstance(other, Series):
            return False
        return self.to_pandas().equals(other.to_pandas())

    def to_pandas(self) -> pandas.Series:
        df = self._query_compiler.to_pandas()
        s = df[df.columns[0]]
        s.name = self.name
        if self._bool_mask:
            s = s.astype(bool)
        return s

    def _to_pandas(self) -> pandas.Series:
        return self.to_pandas()

    def __repr__(self):
        return f"modin_amd.Series(len={len(self)} on device)"


class DataFrameGroupBy:
    """Mirrors modin/pandas/groupby.py DataFrameGroupBy for the reduce aggs
    (sum :1330, count, mean -> _wrap_aggregation -> qc.groupby_<agg>);
    ``as_index=False`` applies the reference's reduce fix-up
    (algebra/groupby.py:278 — keys become leading columns over a fresh
    RangeIndex); ``gb[col]`` / ``gb[[cols]]`` select aggregation columns
    (SeriesGroupBy shape for a single name)."""

    def __init__(self, df: DataFrame, by, as_index: bool = True,
                 series_out: bool = False, dropna: bool = True):
        self._df = df
        self._by = by
        self._as_index = as_index
        self._series_out = series_out
        self._dropna = dropna

    def __getitem__(self, key):
        bys = list(self._by) if isinstance(self._by, (list, tuple)) \
            else [self._by]
        names = [key] if isinstance(key, str) else list(key)
        for n2 in names:
            if n2 not in list(self._df.columns):
                raise lib.HfError(f"groupby selection: column {n2!r} "
                                  "missing")
        sub = self._df[[*bys, *names]]
        return DataFrameGroupBy(sub, self._by, as_index=self._as_index,
                                series_out=isinstance(key, str),
                                dropna=self._dropna)

    def _nat_handled_df(self) -> "DataFrame":
        """NaT groupby keys: dropna=True drops the NaT rows (a device
        filter — pandas' NaN-group rule applied to the iNaT key);
        dropna=False with NaT keys is loud.  Datetime VALUE columns with
        NaT are loud (the int64 agg kernels would treat iNaT as a huge
        negative ns value)."""
        bys = (list(self._by) if isinstance(self._by, (list, tuple))
               else [self._by])
        qc = self._df._query_compiler
        frame = qc._modin_frame
        dtc = frame._dt_cols()
        frame._guard_nat("groupby values",
                         [c for c in frame.columns
                          if c in dtc and c not in bys])
        natkeys = [b for b in bys
                   if b in dtc and frame._col_has_nat(b)]
        if not natkeys:
            return self._df
        if not self._dropna:
            raise lib.HfError("groupby(dropna=False) with NaT keys is a "
                              "later round")
        mask = qc.getitem_column_array(natkeys).dropna_mask()
        return DataFrame(query_compiler=qc.getitem_array(mask))

    def _agg(self, how: str) -> DataFrame:
        qc = self._nat_handled_df()._query_compiler.groupby_agg(
            self._by, how, dropna=self._dropna)
        out = DataFrame(query_compiler=qc)
        if self._series_out and self._as_index:
            name = list(qc._modin_frame.columns)[0]
            return Series(query_compiler=qc, name=name)
        if not self._as_index:
            # reference reduce fix-up (algebra/groupby.py:278): keys become
            # leading columns over a fresh RangeIndex
            return from_pandas(out.to_pandas().reset_index())
        return out

    def sum(self):
        return self._agg("sum")

    def count(self):
        return self._agg("count")

    def mean(self):
        return self._agg("mean")

    def min(self):
        return self._agg("min")

    def max(self):
        return self._agg("max")

    def _tail_qc(self, fn_name: str, **kw):
        """Route the non-reduce aggs through the query compiler with the
        dropna flag (dropna=False rides the sentinel-NaN-key encoding)."""
        return self._nat_handled_df()._query_compiler.groupby_tail_agg(
            self._by, fn_name, dropna=self._dropna, **kw)

    def var(self, ddof: int = 1):
        return DataFrame(query_compiler=self._tail_qc("groupby_var",
                                                      ddof=ddof))

    def std(self, ddof: int = 1):
        return DataFrame(query_compiler=self._tail_qc("groupby_std",
                                                      ddof=ddof))

    def median(self):
        return self._agg("median")

    def sem(self, ddof: int = 1):
        """pandas DataFrameGroupBy.sem: std/sqrt(count) from the
        existing groupby partials (frame-wise device divide + sqrt)."""
        std = DataFrame(query_compiler=self._tail_qc("groupby_std",
                                                     ddof=ddof))
        cnt = self._agg("count")
        if isinstance(cnt, Series):
            cnt = cnt.to_frame(std.columns[0])
            cnt = DataFrame(query_compiler=cnt._query_compiler)
        # std / sqrt(cnt): sqrt via the map kernel, then frame divide
        sq = DataFrame(query_compiler=type(
            cnt._query_compiler).sqrt(
            cnt._query_compiler.astype(np.float64), 0.0))
        out = std / sq
        if self._series_out and self._as_index:
            name = list(out._query_compiler._modin_frame.columns)[0]
            return Series(query_compiler=out._query_compiler, name=name)
        return out

    def quantile(self, q: float = 0.5):
        out = DataFrame(
            query_compiler=self._tail_qc("groupby_quantile", q=float(q)))
        if self._series_out and self._as_index:
            name = list(out._query_compiler._modin_frame.columns)[0]
            return Series(query_compiler=out._query_compiler, name=name)
        if not self._as_index:
            return from_pandas(out.to_pandas().reset_index())
        return out

    def nunique(self):
        qc = self._tail_qc("groupby_nunique")
        out = DataFrame(query_compiler=qc)
        if self._series_out and self._as_index:
            name = list(qc._modin_frame.columns)[0]
            return Series(query_compiler=qc, name=name)
        if not self._as_index:
            return from_pandas(out.to_pandas().reset_index())
        return out

    def prod(self):
        return self._agg("prod")

    def first(self):
        return self._agg("first")

    def last(self):
        return self._agg("last")

    def _transform(self, how: str, **kw):
        """Same-length transforms in original row order (pandas
        DataFrameGroupBy.cumsum/cummin/cummax/cumcount/rank).  as_index is
        irrelevant (pandas keeps the caller's index for transforms)."""
        bys = (list(self._by) if isinstance(self._by, (list, tuple))
               else [self._by])
        frame = self._df._query_compiler._modin_frame
        frame._guard_nat("groupby transform", bys)
        qc = self._df._query_compiler.groupby_transform(
            self._by, how, dropna=self._dropna, **kw)
        if self._series_out or how in ("cumcount", "ngroup"):
            name = list(qc._modin_frame.columns)[0]
            return Series(query_compiler=qc,
                          name=None if how in ("cumcount", "ngroup")
                          else name)
        return DataFrame(query_compiler=qc)

    def cumsum(self):
        return self._transform("cumsum")

    def cummin(self):
        return self._transform("cummin")

    def cummax(self):
        return self._transform("cummax")

    def cumprod(self):
        return self._transform("cumprod")

    def cumcount(self):
        return self._transform("cumcount")

    def ngroup(self):
        return self._transform("ngroup")

    def shift(self, periods: int = 1):
        return self._transform("shift", periods=int(periods))

    def diff(self, periods: int = 1):
        return self._transform("diff", periods=int(periods))

    def pct_change(self, periods: int = 1):
        """pandas DataFrameGroupBy.pct_change(fill_method=None):
        x / x.shift(p within group) - 1."""
        bys = (list(self._by) if isinstance(self._by, (list, tuple))
               else [self._by])
        vals = [c for c in self._df.columns if c not in bys]
        shifted = self._transform("shift", periods=int(periods))
        base = self._df[vals[0]] if self._series_out else self._df[vals]
        return base / shifted - 1

    def ffill(self):
        """pandas DataFrameGroupBy.ffill: forward fill within groups."""
        return self._transform("ffill")

    def bfill(self):
        """pandas DataFrameGroupBy.bfill: backward fill within groups."""
        return self._transform("bfill")

    def rank(self, method: str = "average", ascending: bool = True,
             na_option: str = "keep"):
        if na_option not in ("keep", "top", "bottom"):
            raise lib.HfError(f"rank: bad na_option {na_option!r}")
        return self._transform("rank", ascending=bool(ascending),
                               method=method, na_option=na_option)

    def idxmax(self):
        """Original row label of each group's first max per column
        (all-NaN groups: NaN)."""
        qc = self._tail_qc("groupby_idxmax")
        out = DataFrame(query_compiler=qc)
        if self._series_out and self._as_index:
            name = list(qc._modin_frame.columns)[0]
            return Series(query_compiler=qc, name=name)
        return out

    def idxmin(self):
        qc = self._tail_qc("groupby_idxmin")
        out = DataFrame(query_compiler=qc)
        if self._series_out and self._as_index:
            name = list(qc._modin_frame.columns)[0]
            return Series(query_compiler=qc, name=name)
        return out

    def transform(self, func):
        """pandas DataFrameGroupBy.transform: broadcast aggregates
        ('sum'/'mean'/'count'/'min'/'max') back to every row, or the
        same-length transforms ('cumsum'/'cummin'/'cummax'/'rank') by
        name."""
        if func in ("sum", "mean", "count", "min", "max"):
            return self._transform("b" + func)
        if func in ("cumsum", "cummin", "cummax", "rank"):
            return self._transform(func)
        raise lib.HfError(f"groupby.transform({func!r}) not supported "
                          "(named aggs/transforms only)")

    def size(self):
        """pandas DataFrameGroupBy.size(): a Series of group row counts
        (NaN values included, NaN keys dropped)."""
        out = DataFrame(
            query_compiler=self._tail_qc("groupby_size")).to_pandas()
        return out["size"].rename(None)

    _AGGS = ("sum", "count", "mean", "min", "max", "var", "std",
             "median", "first", "last", "prod")

    def agg(self, how=None, **named):
        """str, list-of-str (MultiIndex columns, pandas col-major order),
        dict {column: agg}, or pandas NAMED aggregation
        (out=("col", "agg")) — composed from the single-agg kernels and
        a device-side horizontal concat (no data copies; columns
        re-label lazily)."""
        if how is None and named:
            bys = (list(self._by) if isinstance(self._by, (list, tuple))
                   else [self._by])
            base = self._nat_handled_df()
            qcs = []
            for out_name, spec in named.items():
                if not (isinstance(spec, tuple) and len(spec) == 2):
                    raise lib.HfError("named agg takes out=(column, agg)")
                col, a = spec
                sub = base[[*bys, col]]
                qc = sub._query_compiler.groupby_agg(
                    self._by, a, dropna=self._dropna)
                qcs.append(qc.rename_columns({col: out_name}))
            return DataFrame(query_compiler=qcs[0].hconcat(qcs[1:]))
        if isinstance(how, str):
            return self._agg(how)
        if isinstance(how, dict):
            qcs = []
            base = self._nat_handled_df()
            for col, a in how.items():
                if not isinstance(a, str):
                    raise lib.HfError(
                        "groupby.agg dict values must be single agg names "
                        "this round")
                bys = (list(self._by) if isinstance(self._by, (list, tuple))
                       else [self._by])
                sub = base[[*bys, col]]
                qcs.append(sub._query_compiler.groupby_agg(
                    self._by, a, dropna=self._dropna))
            return DataFrame(query_compiler=qcs[0].hconcat(qcs[1:]))
        if isinstance(how, (list, tuple)):
            bys = (list(self._by) if isinstance(self._by, (list, tuple))
                   else [self._by])
            val_cols = [c for c in self._df.columns if c not in bys]
            qcs = []
            base = self._nat_handled_df()
            for col in val_cols:  # pandas order: per column, per agg
                for a in how:
                    if not isinstance(a, str):
                        raise lib.HfError("groupby.agg list entries must "
                                          "be agg names")
                    sub = base[[*bys, col]]
                    qc = sub._query_compiler.groupby_agg(
                        self._by, a, dropna=self._dropna)
                    qcs.append(qc.rename_columns({col: (col, a)}))
            return DataFrame(query_compiler=qcs[0].hconcat(qcs[1:]))
        raise lib.HfError("groupby.agg accepts str / list / dict")


class Rolling:
    """pandas Rolling over fixed windows (mirrors pandas.core.window
    Rolling for the sum/mean/count/min/max aggs; min_periods rules as
    measured on pandas 2.3.3 — count gates on window rows, the rest on
    non-NaN observations)."""

    def __init__(self, obj, window: int, min_periods=None):
        self._obj = obj
        self._window = window
        self._min_periods = min_periods

    def _agg(self, op: str):
        qc = self._obj._query_compiler.rolling_agg(
            self._window, self._min_periods, op)
        return self._obj._rewrap(qc)

    def sum(self):
        return self._agg("sum")

    def mean(self):
        return self._agg("mean")

    def count(self):
        return self._agg("count")

    def min(self):
        return self._agg("min")

    def max(self):
        return self._agg("max")

    def var(self, ddof: int = 1):
        return _window_var(
            self._obj,
            lambda o: Rolling(o, self._window, self._min_periods),
            ddof, sqrt_=False)

    def std(self, ddof: int = 1):
        return _window_var(
            self._obj,
            lambda o: Rolling(o, self._window, self._min_periods),
            ddof, sqrt_=True)


def _window_var(obj, mk, ddof, sqrt_):
    """rolling/expanding var/std from the existing window sums:
    (Σx² − (Σx)²/n) / (n − ddof) over the window's non-NaN
    observations, NaN when n <= ddof — the same prefix-scan kernels, no
    new device code.  min_periods gating rides Σx (NaN propagates)."""
    if isinstance(obj, DataFrame):
        qcs = []
        for c in obj.columns:
            qcs.append(_window_var(obj[c], mk, ddof,
                                   sqrt_)._query_compiler)
        return DataFrame(query_compiler=qcs[0].hconcat(qcs[1:]))
    s1 = mk(obj).sum()
    s2 = mk(obj * obj).sum()
    n = mk(obj).count()
    num = (s2 - s1 * s1 / n).clip(lower=0.0)
    v = (num / (n - ddof)).where(n > ddof)
    if sqrt_:
        v = v._rewrap(type(v._query_compiler).sqrt(
            v._query_compiler, 0.0))
    return v


class Expanding:
    """pandas Expanding (growing windows) for sum/mean/count/min/max."""

    def __init__(self, obj, min_periods: int = 1):
        self._obj = obj
        self._min_periods = min_periods

    def _agg(self, op: str):
        qc = self._obj._query_compiler.expanding_agg(self._min_periods,
                                                     op)
        return self._obj._rewrap(qc)

    def var(self, ddof: int = 1):
        return _window_var(self._obj,
                           lambda o: Expanding(o, self._min_periods),
                           ddof, sqrt_=False)

    def std(self, ddof: int = 1):
        return _window_var(self._obj,
                           lambda o: Expanding(o, self._min_periods),
                           ddof, sqrt_=True)

    def sum(self):
        return self._agg("sum")

    def mean(self):
        return self._agg("mean")

    def count(self):
        return self._agg("count")

    def min(self):
        return self._agg("min")

    def max(self):
        return self._agg("max")
