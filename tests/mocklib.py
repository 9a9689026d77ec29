"""numpy mock of modin_amd.core.lib — CPU regression net for the Python
COMPOSITION layer (groupby transforms, rank, where, sort, dedup, …).

The product path has NO CPU fallback; this mock exists only inside the
test suite (tests/test_mock_compositions.py) so composition logic —
the sequences of gather/scatter/scan/filter calls in
modin_amd/core/dataframe.py — can be exercised and diffed against pandas
WITHOUT a GPU, before any gpurun call is spent.  Each function restates
the documented semantics of the corresponding hf_* entry point
(include/hipframe.h); kernel-level parity remains the GPU tier's job.

Usage: the `mocked_lib` fixture monkeypatches the functions into
modin_amd.core.lib for the duration of a test.
"""

import numpy as np

HF_INT64 = 0
HF_FLOAT64 = 1


class MockCol:
    __slots__ = ("arr",)

    def __init__(self, arr):
        self.arr = np.ascontiguousarray(arr)
        assert self.arr.dtype in (np.dtype(np.int64), np.dtype(np.float64))

    @property
    def length(self):
        return int(self.arr.size)

    @property
    def dtype_code(self):
        return HF_INT64 if self.arr.dtype == np.dtype(np.int64) \
            else HF_FLOAT64

    @property
    def np_dtype(self):
        return self.arr.dtype

    def dptr(self):
        return id(self)

    def __repr__(self):
        return f"MockCol(len={self.length}, dtype={self.arr.dtype})"


_by_ptr = {}


def _reg(col):
    _by_ptr[id(col)] = col
    return col


class MockPlan:
    __slots__ = ("keep", "n_kept")

    def __init__(self, keep):
        self.keep = keep
        self.n_kept = int(keep.sum())


def install(monkeypatch):
    """Monkeypatch the numpy restatements into modin_amd.core.lib."""
    from modin_amd.core import lib

    def put(arr):
        arr = np.ascontiguousarray(arr)
        if arr.dtype not in (np.dtype(np.int64), np.dtype(np.float64)):
            raise lib.HfError(f"unsupported dtype {arr.dtype} "
                              "(int64/float64 only)")
        return _reg(MockCol(arr))

    def get(col):
        return col.arr.copy()

    def alloc(length, dtype_code):
        dt = np.int64 if dtype_code == HF_INT64 else np.float64
        return _reg(MockCol(np.zeros(length, dtype=dt)))

    def fill_f64(dptr, value, n):
        _by_ptr[dptr].arr[:n] = value

    def fill_i64(dptr, value, n):
        _by_ptr[dptr].arr[:n] = value

    def col_slice(col, start, length):
        return _reg(MockCol(col.arr[start:start + length].copy()))

    def concat(cols):
        return _reg(MockCol(np.concatenate([c.arr for c in cols])))

    def gather(col, idx):
        return _reg(MockCol(col.arr[idx.arr]))

    def scatter(col, idx):
        out = np.empty_like(col.arr)
        out[idx.arr] = col.arr
        return _reg(MockCol(out))

    def sort_perm(keys, ascending=True):
        if ascending:
            return _reg(MockCol(np.argsort(keys.arr, kind="stable")))
        # stable descending == stable ascending on the negated key
        # (hf_sort_perm sorts key_max - key)
        return _reg(MockCol(np.argsort(keys.arr.max() - keys.arr
                                       if keys.arr.size else keys.arr,
                                       kind="stable")))

    def cumsum(col, agg_op=0):
        x = col.arr
        if col.dtype_code == HF_INT64:
            fns = {0: np.add, 1: np.minimum, 2: np.maximum, 3: np.multiply}
            return _reg(MockCol(fns[agg_op].accumulate(x)))
        nan = np.isnan(x)
        ident = {0: 0.0, 1: np.inf, 2: -np.inf, 3: 1.0}[agg_op]
        z = np.where(nan, ident, x)
        fns = {0: np.add, 1: np.minimum, 2: np.maximum, 3: np.multiply}
        acc = fns[agg_op].accumulate(z)
        return _reg(MockCol(np.where(nan, np.nan, acc)))

    def seg_cumsum(col, heads, agg_op=0):
        x = col.arr
        h = heads.arr != 0
        ident = {0: 0.0, 1: np.inf, 2: -np.inf, 3: 1.0}[agg_op] \
            if col.dtype_code == HF_FLOAT64 else \
            {0: 0, 1: np.iinfo(np.int64).max,
             2: np.iinfo(np.int64).min, 3: 1}[agg_op]
        comb = {0: lambda a, b: a + b, 1: min, 2: max,
                3: lambda a, b: a * b}[agg_op]
        out = np.empty_like(x)
        run = ident
        for i in range(x.size):
            if h[i]:
                run = ident
            xi = x[i]
            isn = col.dtype_code == HF_FLOAT64 and np.isnan(xi)
            if not isn:
                run = comb(run, xi)
            out[i] = np.nan if isn else run
        return _reg(MockCol(out))

    def filter_plan(mask):
        return MockPlan(mask.arr != 0)

    def filter_apply(plan, col):
        return _reg(MockCol(col.arr[plan.keep]))

    def filter_iota(plan, base=0):
        return _reg(MockCol(np.nonzero(plan.keep)[0].astype(np.int64)
                            + base))

    def compare_scalar(op, col, scalar):
        x = col.arr
        if op == lib.CMP_NOTNA:
            m = ~np.isnan(x) if col.dtype_code == HF_FLOAT64 \
                else np.ones(x.size, dtype=bool)
        else:
            fn = {lib.CMP_GT: np.greater, lib.CMP_GE: np.greater_equal,
                  lib.CMP_LT: np.less, lib.CMP_LE: np.less_equal,
                  lib.CMP_EQ: np.equal, lib.CMP_NE: np.not_equal}[op]
            with np.errstate(invalid="ignore"):
                m = fn(x, scalar)
            if col.dtype_code == HF_FLOAT64 and op == lib.CMP_NE:
                m = m | np.isnan(x)  # NaN != s is True (pandas ne)
        return _reg(MockCol(m.astype(np.int64)))

    def binary(op, a, b):
        x, y = a.arr, b.arr
        if a.dtype_code != b.dtype_code:
            x = x.astype(np.float64)
            y = y.astype(np.float64)
        fn = {lib.BIN_ADD: np.add, lib.BIN_SUB: np.subtract,
              lib.BIN_MUL: np.multiply, lib.BIN_DIV: np.true_divide,
              lib.BIN_MIN: np.fmin, lib.BIN_MAX: np.fmax}[op]
        if op == lib.BIN_DIV:
            x = x.astype(np.float64)
            y = y.astype(np.float64)
        with np.errstate(divide="ignore", invalid="ignore"):
            return _reg(MockCol(fn(x, y)))

    def map_scalar(op, col, scalar):
        x = col.arr
        is_int = col.dtype_code == HF_INT64
        if is_int and op in (lib.MAP_DIV, lib.MAP_RDIV, lib.MAP_FILLNA):
            x = x.astype(np.float64)
            is_int = False
        elif is_int and op in (lib.MAP_ADD, lib.MAP_SUB, lib.MAP_RSUB,
                               lib.MAP_MUL, lib.MAP_MIN, lib.MAP_MAX) \
                and isinstance(scalar, float) \
                and not float(scalar).is_integer():
            x = x.astype(np.float64)
            is_int = False
        if op == lib.MAP_CAST_F64:
            return _reg(MockCol(x.astype(np.float64)))
        if op == lib.MAP_CAST_I64:
            return _reg(MockCol(x.astype(np.int64)))
        s = (int(scalar or 0) if is_int else float(scalar or 0.0))
        with np.errstate(divide="ignore", invalid="ignore"):
            if op == lib.MAP_ADD:
                r = x + s
            elif op == lib.MAP_SUB:
                r = x - s
            elif op == lib.MAP_RSUB:
                r = s - x
            elif op == lib.MAP_MUL:
                r = x * s
            elif op == lib.MAP_DIV:
                r = x / s
            elif op == lib.MAP_RDIV:
                r = s / x
            elif op == lib.MAP_FILLNA:
                r = np.where(np.isnan(x), s, x)
            elif op == lib.MAP_ABS:
                r = np.abs(x)
            elif op == lib.MAP_NEG:
                r = -x
            elif op == lib.MAP_SQRT:
                r = np.sqrt(x)
            elif op == lib.MAP_MIN:
                r = np.where(np.isnan(x), x, np.fmin(x, s)) \
                    if not is_int else np.minimum(x, s)
            elif op == lib.MAP_MAX:
                r = np.where(np.isnan(x), x, np.fmax(x, s)) \
                    if not is_int else np.maximum(x, s)
            elif op == lib.MAP_ROUND:
                r = np.rint(x * s) / s
            else:
                raise lib.HfError(f"mock map op {op}")
        return _reg(MockCol(r))

    def cast_f64(col):
        return _reg(MockCol(col.arr.astype(np.float64)))

    class R:
        pass

    def reduce(col):
        x = col.arr
        r = R()
        if col.dtype_code == HF_FLOAT64:
            m = ~np.isnan(x)
            r.count = int(m.sum())
            r.sum = float(x[m].sum()) if r.count else 0.0
            r.mn = float(x[m].min()) if r.count else np.nan
            r.mx = float(x[m].max()) if r.count else np.nan
            r.isum, r.imn, r.imx = 0, 0, 0
        else:
            r.count = int(x.size)
            r.isum = int(x.sum()) if x.size else 0
            r.imn = int(x.min()) if x.size else 0
            r.imx = int(x.max()) if x.size else 0
            r.sum = float(r.isum)
            r.mn, r.mx = float(r.imn), float(r.imx)
        return r

    def fixup_empty(val, cnt):
        v = val.arr.astype(np.float64)
        return _reg(MockCol(np.where(cnt.arr == 0, np.nan, v)))

    def ordered_i64(col, inverse=False):
        if inverse:
            return _reg(MockCol(lib.ordered_to_f64_np(col.arr)))
        x = col.arr.astype(np.float64) + 0.0
        v = x.view(np.int64).copy()
        v[np.isnan(x)] = 0x7FF8000000000000
        neg = v < 0
        v[neg] = ~v[neg] ^ np.int64(-2**63)
        return _reg(MockCol(v))

    def search_sorted(keys, sorted_uniq):
        su = sorted_uniq.arr
        pos = np.searchsorted(su, keys.arr)
        pos = np.clip(pos, 0, max(su.size - 1, 0))
        hit = su.size > 0
        out = np.where(hit & (su[pos] == keys.arr) if su.size else False,
                       pos, -1).astype(np.int64)
        return _reg(MockCol(out))

    def cross_idx(nl, nr):
        i = np.arange(nl * nr, dtype=np.int64)
        return _reg(MockCol(i // nr)), _reg(MockCol(i % nr))

    def sync():
        pass

    def ensure_ready(gpu=None):
        pass

    for name, fn in [
        ("put", put), ("get", get), ("alloc", alloc),
        ("fill_f64", fill_f64), ("fill_i64", fill_i64),
        ("col_slice", col_slice), ("concat", concat), ("gather", gather),
        ("scatter", scatter), ("sort_perm", sort_perm),
        ("cumsum", cumsum), ("seg_cumsum", seg_cumsum),
        ("filter_plan", filter_plan), ("filter_apply", filter_apply),
        ("filter_iota", filter_iota), ("compare_scalar", compare_scalar),
        ("binary", binary), ("map_scalar", map_scalar),
        ("cast_f64", cast_f64), ("reduce", reduce),
        ("fixup_empty", fixup_empty), ("ordered_i64", ordered_i64),
        ("search_sorted", search_sorted), ("cross_idx", cross_idx),
        ("sync", sync), ("ensure_ready", ensure_ready),
    ]:
        monkeypatch.setattr(lib, name, fn)
