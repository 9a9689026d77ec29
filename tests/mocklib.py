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
        if not keys.arr.size:
            return _reg(MockCol(np.argsort(keys.arr, kind="stable")))
        if keys.dtype_code == HF_FLOAT64:
            return _reg(MockCol(np.argsort(-keys.arr, kind="stable")))
        # stable descending: mirror the kernel's mod-2^64 shifted keys
        # (span - (key - key_min)) — plain `max - key` overflows int64
        # for mixed-sign ordered keys
        u = keys.arr.astype(np.int64).view(np.uint64)
        mn = np.array([keys.arr.min()], dtype=np.int64).view(np.uint64)[0]
        shifted = u - mn
        return _reg(MockCol(np.argsort(shifted.max() - shifted,
                                       kind="stable").astype(np.int64)))

    def shuffle_dest(keys, splitters):
        spl = np.asarray(splitters, dtype=np.int64)
        return _reg(MockCol(np.searchsorted(spl, keys.arr,
                                            side="right").astype(np.int64)))

    def fill_randf64(n, seed):
        from oracle import ops as _o
        return _reg(MockCol(_o.rand_f64(seed, n)))

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
            elif op == lib.MAP_IDIV:
                r = x // s  # numpy int floordiv == Python semantics
            elif op == lib.MAP_IMOD:
                r = x % s
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

    # ---- dense-table groupby engine (hf_alloc_raw / hf_groupby_accum /
    # hf_groupby_compact and the hash/sorted variants) so the WHOLE
    # groupby-reduce composition runs on CPU ----
    _raw = {}
    _raw_next = [1 << 40]

    def alloc_raw(nbytes):
        ptr = _raw_next[0]
        _raw_next[0] += ((nbytes + 255) & ~255) + 256
        _raw[ptr] = np.zeros(nbytes, dtype=np.uint8)
        return ptr

    def free_raw(ptr):
        _raw.pop(ptr, None)

    def memset_raw(ptr, value, nbytes):
        _raw[ptr][:nbytes] = value

    def _rawview(ptr, dtype, count):
        return _raw[ptr][: count * 8].view(dtype)

    def fill_f64(dptr, value, n):
        if dptr in _raw:
            _rawview(dptr, np.float64, n)[:n] = value
        else:
            _by_ptr[dptr].arr[:n] = value

    def fill_i64(dptr, value, n):
        if dptr in _raw:
            _rawview(dptr, np.int64, n)[:n] = value
        else:
            _by_ptr[dptr].arr[:n] = value

    def groupby_accum(keys, vals, agg_op, key_min, n_slots, sums, rowcnt,
                      counts):
        k = keys.arr - key_min
        ok = (k >= 0) & (k < n_slots)
        k = k[ok]
        rc = _rawview(rowcnt, np.int64, n_slots)
        rc += np.bincount(k, minlength=n_slots)
        sv = _rawview(sums, np.float64, n_slots * max(len(vals), 1))
        cv = (_rawview(counts, np.int64, n_slots * max(len(vals), 1))
              if counts else None)
        for c, vcol in enumerate(vals):
            x = vcol.arr[ok].astype(np.float64)
            m = ~np.isnan(x)
            sl = sv[c * n_slots:(c + 1) * n_slots]
            if agg_op == 0:
                sl += np.bincount(k[m], weights=x[m], minlength=n_slots)
            else:
                fn2 = np.minimum if agg_op == 1 else np.maximum
                np_fn = fn2.at
                np_fn(sl, k[m], x[m])
            if cv is not None:
                cl = cv[c * n_slots:(c + 1) * n_slots]
                cl += np.bincount(k[m], minlength=n_slots)

    def groupby_compact(sums, rowcnt, counts, nvals, key_min, n_slots):
        rc = _rawview(rowcnt, np.int64, n_slots)
        present = np.nonzero(rc > 0)[0]
        n = present.size
        kcol = _reg(MockCol(present + key_min))
        sv = _rawview(sums, np.float64, n_slots * max(nvals, 1))
        scols = [_reg(MockCol(sv[c * n_slots:(c + 1) * n_slots]
                              [present].copy())) for c in range(nvals)]
        ccols = None
        if counts:
            cvv = _rawview(counts, np.int64, n_slots * max(nvals, 1))
            ccols = [_reg(MockCol(cvv[c * n_slots:(c + 1) * n_slots]
                                  [present].copy())) for c in range(nvals)]
        return kcol, scols, ccols, n

    def groupby_sorted(sorted_keys, vals, agg_op, want_counts):
        k = sorted_keys.arr
        if k.size == 0:
            e = _reg(MockCol(np.empty(0, dtype=np.int64)))
            return e, [], [] if want_counts else None, 0
        uniq, inv = np.unique(k, return_inverse=True)
        n = uniq.size
        kcol = _reg(MockCol(uniq))
        scols, ccols = [], []
        for vcol in vals:
            x = vcol.arr.astype(np.float64)
            m = ~np.isnan(x)
            if agg_op == 0:
                sl = np.bincount(inv[m], weights=x[m], minlength=n)
            else:
                ident = np.inf if agg_op == 1 else -np.inf
                sl = np.full(n, ident)
                (np.minimum if agg_op == 1 else np.maximum).at(
                    sl, inv[m], x[m])
            scols.append(_reg(MockCol(sl)))
            if want_counts:
                ccols.append(_reg(MockCol(
                    np.bincount(inv[m], minlength=n).astype(np.int64))))
        return kcol, scols, (ccols if want_counts else None), n

    _hash_store = {}

    def groupby_hash_accum(keys, vals, agg_op, H, tkey, sums, rowcnt,
                           counts):
        st = _hash_store.setdefault(tkey, {
            "k": [], "v": [[] for _ in vals], "agg": agg_op,
        })
        st["k"].append(keys.arr.copy())
        for c, vcol in enumerate(vals):
            st["v"][c].append(vcol.arr.astype(np.float64))

    def groupby_hash_compact(tkey, sums, rowcnt, counts, nvals, H):
        st = _hash_store.pop(tkey, {"k": [], "v": [[] for _ in range(nvals)],
                                    "agg": 0})
        k = (np.concatenate(st["k"]) if st["k"]
             else np.empty(0, dtype=np.int64))
        uniq, inv = np.unique(k, return_inverse=True)
        n = uniq.size
        agg = st["agg"]
        scols, ccols = [], []
        for c in range(nvals):
            x = (np.concatenate(st["v"][c]) if st["v"][c]
                 else np.empty(0, dtype=np.float64))
            m = ~np.isnan(x)
            if agg == 0:
                sl = np.bincount(inv[m], weights=x[m], minlength=n)
            else:
                ident = np.inf if agg == 1 else -np.inf
                sl = np.full(n, ident)
                (np.minimum if agg == 1 else np.maximum).at(sl, inv[m],
                                                            x[m])
            scols.append(_reg(MockCol(sl)))
            if counts:
                ccols.append(_reg(MockCol(
                    np.bincount(inv[m], minlength=n).astype(np.int64))))
        return (_reg(MockCol(uniq)), scols,
                (ccols if counts else None), n)

    class _MockJoin:
        pass

    def join_build(rkeys, rvals, key_min, n_slots):
        j = _MockJoin()
        j.k = rkeys.arr.copy()
        j.v = [v.arr.copy() for v in rvals]
        j.rdtypes = [v.dtype_code for v in rvals]
        j.key_min = key_min
        j.n_slots = n_slots
        return j

    def join_probe(j, lkeys):
        import pandas as _pd
        ldf = _pd.DataFrame({"k": lkeys.arr,
                             "li": np.arange(lkeys.arr.size)})
        rdf = _pd.DataFrame({"k": j.k, "ri": np.arange(j.k.size)})
        m = ldf.merge(rdf, on="k", how="inner", sort=False)
        keys = _reg(MockCol(m["k"].to_numpy().astype(np.int64)))
        lidx = _reg(MockCol(m["li"].to_numpy().astype(np.int64)))
        ri = m["ri"].to_numpy()
        rcols = [_reg(MockCol(j.v[c][ri].copy()))
                 for c in range(len(j.v))]
        return keys, lidx, rcols, int(len(m))

    def join_free(j):
        pass

    for name, fn in [
        ("put", put), ("get", get), ("alloc", alloc),
        ("fill_f64", fill_f64), ("fill_i64", fill_i64),
        ("col_slice", col_slice), ("concat", concat), ("gather", gather),
        ("scatter", scatter), ("sort_perm", sort_perm),
        ("fill_randf64", fill_randf64), ("shuffle_dest", shuffle_dest),
        ("cumsum", cumsum), ("seg_cumsum", seg_cumsum),
        ("filter_plan", filter_plan), ("filter_apply", filter_apply),
        ("filter_iota", filter_iota), ("compare_scalar", compare_scalar),
        ("binary", binary), ("map_scalar", map_scalar),
        ("cast_f64", cast_f64), ("reduce", reduce),
        ("fixup_empty", fixup_empty), ("ordered_i64", ordered_i64),
        ("search_sorted", search_sorted), ("cross_idx", cross_idx),
        ("sync", sync), ("ensure_ready", ensure_ready),
        ("alloc_raw", alloc_raw), ("free_raw", free_raw),
        ("memset_raw", memset_raw), ("groupby_accum", groupby_accum),
        ("groupby_compact", groupby_compact),
        ("groupby_sorted", groupby_sorted),
        ("groupby_hash_accum", groupby_hash_accum),
        ("groupby_hash_compact", groupby_hash_compact),
        ("join_build", join_build), ("join_probe", join_probe),
    ]:
        monkeypatch.setattr(lib, name, fn)
