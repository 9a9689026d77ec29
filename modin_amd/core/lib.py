"""ctypes binding to libhipframe.so — the C-ABI engine boundary.

This module is the modin_amd equivalent of the reference's engine wrapper
(``modin/core/execution/python/common/engine_wrapper.py:17-42`` —
``PythonWrapper.deploy`` executes the partition kernel); here "deploy" is a
ctypes call into a hand-written gfx950 HIP kernel (include/hipframe.h).

Loading the .so works on a GPU-less machine (symbols only; used by the CPU
test tier).  Any COMPUTE call requires ``ensure_ready()`` which performs
``hf_init`` and raises loudly if no MI355X is visible — there is NO CPU
fallback behind this boundary.
"""

from __future__ import annotations

import ctypes as ct
import os
import threading

import numpy as np

_LIB_NAME = "libhipframe.so"


class HfError(RuntimeError):
    """Raised when a hipframe C-ABI call fails."""


class HfReduceResult(ct.Structure):
    # mirrors hf_reduce_result in include/hipframe.h
    _fields_ = [
        ("sum", ct.c_double),
        ("count", ct.c_int64),
        ("mn", ct.c_double),
        ("mx", ct.c_double),
        ("isum", ct.c_int64),
        ("imn", ct.c_int64),
        ("imx", ct.c_int64),
    ]


# dtype codes (include/hipframe.h)
HF_INT64 = 0
HF_FLOAT64 = 1

# map ops
MAP_ADD, MAP_SUB, MAP_RSUB, MAP_MUL, MAP_DIV, MAP_RDIV, MAP_FILLNA, MAP_ABS, \
    MAP_NEG, MAP_CAST_F64, MAP_CAST_I64, MAP_SQRT, MAP_MIN, MAP_MAX, \
    MAP_ROUND, MAP_IDIV, MAP_IMOD = range(17)
# binary ops
BIN_ADD, BIN_SUB, BIN_MUL, BIN_DIV, BIN_MIN, BIN_MAX = range(6)

_NP_TO_HF = {np.dtype(np.int64): HF_INT64, np.dtype(np.float64): HF_FLOAT64}
_HF_TO_NP = {HF_INT64: np.dtype(np.int64), HF_FLOAT64: np.dtype(np.float64)}


def so_path() -> str:
    return os.path.join(os.path.dirname(__file__), "..", "csrc", _LIB_NAME)


_lock = threading.RLock()
_dll = None
_inited_gpu = None


def load() -> ct.CDLL:
    """Load the shared library and declare every exported signature."""
    global _dll
    with _lock:
        if _dll is not None:
            return _dll
        path = os.path.abspath(so_path())
        if not os.path.exists(path):
            raise HfError(
                f"{_LIB_NAME} not built at {path} — run __graft_entry__.build() "
                "(hipcc --offload-arch=gfx950); the modin_amd compute path has "
                "no CPU fallback."
            )
        dll = ct.CDLL(path)
        sig = {
            "hf_init": (ct.c_int, [ct.c_int]),
            "hf_shutdown": (ct.c_int, []),
            "hf_device_count": (ct.c_int, [ct.POINTER(ct.c_int)]),
            "hf_last_error": (ct.c_char_p, []),
            "hf_sync": (ct.c_int, []),
            "hf_put": (ct.c_int, [ct.c_void_p, ct.c_int64, ct.c_int,
                                  ct.POINTER(ct.c_void_p)]),
            "hf_get": (ct.c_int, [ct.c_void_p, ct.c_void_p]),
            "hf_col_alloc": (ct.c_int, [ct.c_int64, ct.c_int,
                                        ct.POINTER(ct.c_void_p)]),
            "hf_col_free": (ct.c_int, [ct.c_void_p]),
            "hf_col_len": (ct.c_int64, [ct.c_void_p]),
            "hf_col_dtype": (ct.c_int, [ct.c_void_p]),
            "hf_col_dptr": (ct.c_size_t, [ct.c_void_p]),
            "hf_alloc_raw": (ct.c_int, [ct.c_int64, ct.POINTER(ct.c_size_t)]),
            "hf_free_raw": (ct.c_int, [ct.c_size_t]),
            "hf_memset_raw": (ct.c_int, [ct.c_size_t, ct.c_int, ct.c_int64]),
            "hf_map_scalar": (ct.c_int, [ct.c_int, ct.c_void_p, ct.c_double,
                                         ct.POINTER(ct.c_void_p)]),
            "hf_map_scalar_i64": (ct.c_int, [ct.c_int, ct.c_void_p, ct.c_int64,
                                             ct.POINTER(ct.c_void_p)]),
            "hf_binary": (ct.c_int, [ct.c_int, ct.c_void_p, ct.c_void_p,
                                     ct.POINTER(ct.c_void_p)]),
            "hf_reduce": (ct.c_int, [ct.c_void_p, ct.POINTER(HfReduceResult)]),
            "hf_groupby_accum": (ct.c_int, [ct.c_void_p, ct.POINTER(ct.c_void_p),
                                            ct.c_int, ct.c_int, ct.c_int64,
                                            ct.c_int64, ct.c_size_t,
                                            ct.c_size_t, ct.c_size_t]),
            "hf_fill_f64": (ct.c_int, [ct.c_size_t, ct.c_double, ct.c_int64]),
            "hf_fill_i64": (ct.c_int, [ct.c_size_t, ct.c_int64, ct.c_int64]),
            "hf_fill_randint": (ct.c_int, [ct.c_void_p, ct.c_uint64,
                                           ct.c_int64, ct.c_int64]),
            "hf_fill_randf64": (ct.c_int, [ct.c_void_p, ct.c_uint64]),
            "hf_fill_randcdf": (ct.c_int, [ct.c_void_p, ct.c_uint64,
                                           ct.c_void_p]),
            "hf_groupby_hash_accum": (ct.c_int, [ct.c_void_p,
                                                 ct.POINTER(ct.c_void_p),
                                                 ct.c_int, ct.c_int,
                                                 ct.c_int64, ct.c_size_t,
                                                 ct.c_size_t, ct.c_size_t,
                                                 ct.c_size_t]),
            "hf_groupby_hash_compact": (ct.c_int, [ct.c_size_t, ct.c_size_t,
                                                   ct.c_size_t, ct.c_size_t,
                                                   ct.c_int, ct.c_int64,
                                                   ct.POINTER(ct.c_void_p),
                                                   ct.POINTER(ct.c_void_p),
                                                   ct.POINTER(ct.c_void_p),
                                                   ct.POINTER(ct.c_int64)]),
            "hf_groupby_sorted": (ct.c_int, [ct.c_void_p,
                                             ct.POINTER(ct.c_void_p),
                                             ct.c_int, ct.c_int, ct.c_int,
                                             ct.POINTER(ct.c_void_p),
                                             ct.POINTER(ct.c_void_p),
                                             ct.POINTER(ct.c_void_p),
                                             ct.POINTER(ct.c_int64)]),
            "hf_search_sorted": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                            ct.POINTER(ct.c_void_p)]),
            "hf_ordered_i64": (ct.c_int, [ct.c_void_p, ct.c_int,
                                          ct.POINTER(ct.c_void_p)]),
            "hf_cumsum": (ct.c_int, [ct.c_void_p, ct.c_int,
                                     ct.POINTER(ct.c_void_p)]),
            "hf_seg_cumsum": (ct.c_int, [ct.c_void_p, ct.c_void_p, ct.c_int,
                                         ct.POINTER(ct.c_void_p)]),
            "hf_scatter": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                      ct.POINTER(ct.c_void_p)]),
            "hf_cross_idx": (ct.c_int, [ct.c_int64, ct.c_int64,
                                        ct.POINTER(ct.c_void_p),
                                        ct.POINTER(ct.c_void_p)]),
            "hf_shuffle_dest": (ct.c_int, [ct.c_void_p,
                                           ct.POINTER(ct.c_int64), ct.c_int,
                                           ct.POINTER(ct.c_void_p)]),
            "hf_memcpy_dd": (ct.c_int, [ct.c_size_t, ct.c_size_t,
                                        ct.c_int64]),
            "hf_sort_perm": (ct.c_int, [ct.c_void_p, ct.c_int,
                                        ct.POINTER(ct.c_void_p)]),
            "hf_fixup_empty": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                          ct.POINTER(ct.c_void_p)]),
            "hf_groupby_compact": (ct.c_int, [ct.c_size_t, ct.c_size_t, ct.c_size_t,
                                              ct.c_int, ct.c_int64, ct.c_int64,
                                              ct.POINTER(ct.c_void_p),
                                              ct.POINTER(ct.c_void_p),
                                              ct.POINTER(ct.c_void_p),
                                              ct.POINTER(ct.c_int64)]),
            "hf_col_concat": (ct.c_int, [ct.POINTER(ct.c_void_p), ct.c_int,
                                         ct.POINTER(ct.c_void_p)]),
            "hf_col_slice": (ct.c_int, [ct.c_void_p, ct.c_int64, ct.c_int64,
                                        ct.POINTER(ct.c_void_p)]),
            "hf_join_build": (ct.c_int, [ct.c_void_p, ct.POINTER(ct.c_void_p),
                                         ct.c_int, ct.c_int64, ct.c_int64,
                                         ct.POINTER(ct.c_void_p)]),
            "hf_join_free": (ct.c_int, [ct.c_void_p]),
            "hf_join_probe": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                         ct.POINTER(ct.c_void_p),
                                         ct.POINTER(ct.c_void_p),
                                         ct.POINTER(ct.c_void_p),
                                         ct.POINTER(ct.c_int64)]),
            "hf_gather": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                     ct.POINTER(ct.c_void_p)]),
            "hf_compare_scalar": (ct.c_int, [ct.c_int, ct.c_void_p,
                                             ct.c_double,
                                             ct.POINTER(ct.c_void_p)]),
            "hf_filter_plan": (ct.c_int, [ct.c_void_p, ct.POINTER(ct.c_void_p),
                                          ct.POINTER(ct.c_int64)]),
            "hf_filter_apply": (ct.c_int, [ct.c_void_p, ct.c_void_p,
                                           ct.POINTER(ct.c_void_p)]),
            "hf_filter_iota": (ct.c_int, [ct.c_void_p, ct.c_int64,
                                          ct.POINTER(ct.c_void_p)]),
            "hf_filter_plan_free": (ct.c_int, [ct.c_void_p]),
            "hf_profiling": (ct.c_int, [ct.c_int]),
            "hf_kernel_stats": (ct.c_int, [ct.c_char_p, ct.POINTER(ct.c_int64),
                                           ct.POINTER(ct.c_double)]),
            "hf_kernel_stats_reset": (ct.c_int, []),
        }
        for name, (res, args) in sig.items():
            fn = getattr(dll, name)
            fn.restype = res
            fn.argtypes = args
        _dll = dll
        return dll


def exported_symbols():
    """Names include/hipframe.h declares — checked by the CPU symbol test."""
    return [
        "hf_init", "hf_shutdown", "hf_device_count", "hf_last_error", "hf_sync",
        "hf_put", "hf_get", "hf_col_alloc", "hf_col_free", "hf_col_len",
        "hf_col_dtype", "hf_col_dptr", "hf_alloc_raw", "hf_free_raw",
        "hf_memset_raw", "hf_map_scalar", "hf_map_scalar_i64", "hf_binary",
        "hf_reduce", "hf_groupby_accum", "hf_groupby_compact", "hf_fill_f64",
        "hf_fixup_empty", "hf_sort_perm", "hf_fill_i64",
        "hf_fill_randint", "hf_fill_randf64", "hf_fill_randcdf",
        "hf_groupby_hash_accum", "hf_groupby_hash_compact",
        "hf_groupby_sorted", "hf_shuffle_dest", "hf_memcpy_dd",
        "hf_search_sorted", "hf_ordered_i64", "hf_cumsum", "hf_seg_cumsum",
        "hf_scatter", "hf_cross_idx",
        "hf_col_concat", "hf_col_slice", "hf_join_build", "hf_join_free", "hf_join_probe",
        "hf_gather", "hf_compare_scalar", "hf_filter_plan", "hf_filter_apply",
        "hf_filter_iota", "hf_filter_plan_free", "hf_profiling",
        "hf_kernel_stats", "hf_kernel_stats_reset",
    ]


def _check(rc: int, what: str) -> None:
    if rc != 0:
        msg = load().hf_last_error().decode() if _dll else "?"
        raise HfError(f"{what} failed (rc={rc}): {msg}")


def ensure_ready(gpu: int | None = None) -> None:
    """Initialise the engine on the given GPU (default: MODIN_AMD_GPU or 0)."""
    global _inited_gpu
    dll = load()
    if gpu is None:
        gpu = int(os.environ.get("MODIN_AMD_GPU", "0"))
    with _lock:
        if _inited_gpu == gpu:
            return
        _check(dll.hf_init(gpu), "hf_init")
        _inited_gpu = gpu


def is_ready() -> bool:
    return _inited_gpu is not None


def shutdown() -> None:
    global _inited_gpu
    if _dll is not None and _inited_gpu is not None:
        _dll.hf_shutdown()
    _inited_gpu = None


def device_count() -> int:
    dll = load()
    n = ct.c_int(0)
    dll.hf_device_count(ct.byref(n))
    return n.value


class ColumnRef:
    """Owner of one hf_col device column (frees it on GC).

    Device columns are immutable (every op writes a new column), so per-column
    reduce partials are cached here — the lazy-metadata pattern of the
    reference (modin/core/dataframe/pandas/metadata/, ModinDtypes/index
    caches): a groupby's key-range scan runs once per column, not per query.
    """

    __slots__ = ("handle", "length", "dtype_code", "_reduce_cache")

    def __init__(self, handle: ct.c_void_p, length: int, dtype_code: int):
        self.handle = handle
        self.length = length
        self.dtype_code = dtype_code
        self._reduce_cache = None

    @property
    def np_dtype(self):
        return _HF_TO_NP[self.dtype_code]

    def dptr(self) -> int:
        return load().hf_col_dptr(self.handle)

    def __del__(self):
        try:
            if _dll is not None and _inited_gpu is not None and self.handle:
                _dll.hf_col_free(self.handle)
        except Exception:
            pass

    def __repr__(self):
        return f"ColumnRef(len={self.length}, dtype={self.np_dtype})"


def _wrap(out: ct.c_void_p, length: int, dtype_code: int) -> ColumnRef:
    return ColumnRef(out, length, dtype_code)


# ---- op wrappers -----------------------------------------------------------

def put(arr: np.ndarray) -> ColumnRef:
    ensure_ready()
    arr = np.ascontiguousarray(arr)
    dt = _NP_TO_HF.get(arr.dtype)
    if dt is None:
        raise HfError(f"unsupported dtype {arr.dtype} (int64/float64 only)")
    out = ct.c_void_p()
    _check(load().hf_put(arr.ctypes.data_as(ct.c_void_p), arr.size, dt,
                         ct.byref(out)), "hf_put")
    # the H2D copy is async from pageable memory; keep the source alive until
    # the stream drains
    _check(load().hf_sync(), "hf_sync")
    return _wrap(out, arr.size, dt)


def get(col: ColumnRef) -> np.ndarray:
    ensure_ready()
    out = np.empty(col.length, dtype=col.np_dtype)
    _check(load().hf_get(col.handle, out.ctypes.data_as(ct.c_void_p)), "hf_get")
    return out


def alloc(length: int, dtype_code: int) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_col_alloc(length, dtype_code, ct.byref(out)), "hf_col_alloc")
    return _wrap(out, length, dtype_code)


def map_scalar(op: int, col: ColumnRef, scalar) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    if col.dtype_code == HF_INT64 and op in (MAP_DIV, MAP_RDIV, MAP_FILLNA):
        col = cast_f64(col)  # pandas promotes int div to float; fillna no-ops
    elif (col.dtype_code == HF_INT64
          and op in (MAP_ADD, MAP_SUB, MAP_RSUB, MAP_MUL, MAP_MIN, MAP_MAX)
          and isinstance(scalar, float) and not scalar.is_integer()):
        col = cast_f64(col)  # pandas: int64 op non-integral float -> float64
    if op == MAP_CAST_I64:
        if col.dtype_code == HF_INT64:
            return col
        _check(load().hf_map_scalar(op, col.handle, 0.0, ct.byref(out)),
               "hf_map_scalar(cast_i64)")
        return _wrap(out, col.length, HF_INT64)
    if col.dtype_code == HF_INT64 and op != MAP_CAST_F64:
        _check(load().hf_map_scalar_i64(op, col.handle, int(scalar or 0),
                                        ct.byref(out)), "hf_map_scalar_i64")
        return _wrap(out, col.length, HF_INT64)
    _check(load().hf_map_scalar(op, col.handle, float(scalar or 0.0), ct.byref(out)),
           "hf_map_scalar")
    return _wrap(out, col.length, HF_FLOAT64)


def cast_f64(col: ColumnRef) -> ColumnRef:
    if col.dtype_code == HF_FLOAT64:
        return col
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_map_scalar(MAP_CAST_F64, col.handle, 0.0, ct.byref(out)),
           "hf_map_scalar(cast)")
    return _wrap(out, col.length, HF_FLOAT64)


def binary(op: int, a: ColumnRef, b: ColumnRef) -> ColumnRef:
    ensure_ready()
    if a.dtype_code != b.dtype_code:
        a, b = cast_f64(a), cast_f64(b)
    if a.dtype_code == HF_INT64 and op == BIN_DIV:
        a, b = cast_f64(a), cast_f64(b)
    out = ct.c_void_p()
    _check(load().hf_binary(op, a.handle, b.handle, ct.byref(out)), "hf_binary")
    return _wrap(out, a.length, a.dtype_code)


def reduce(col: ColumnRef) -> HfReduceResult:
    if col._reduce_cache is not None:
        return col._reduce_cache
    ensure_ready()
    res = HfReduceResult()
    _check(load().hf_reduce(col.handle, ct.byref(res)), "hf_reduce")
    col._reduce_cache = res
    return res


def alloc_raw(nbytes: int) -> int:
    ensure_ready()
    p = ct.c_size_t(0)
    _check(load().hf_alloc_raw(nbytes, ct.byref(p)), "hf_alloc_raw")
    return p.value


def free_raw(dptr: int) -> None:
    if _dll is not None and _inited_gpu is not None and dptr:
        _dll.hf_free_raw(dptr)


def memset_raw(dptr: int, value: int, nbytes: int) -> None:
    _check(load().hf_memset_raw(dptr, value, nbytes), "hf_memset_raw")


AGG_SUM, AGG_MIN, AGG_MAX, AGG_PROD = range(4)
AGG_OP_OF = {"sum": AGG_SUM, "count": AGG_SUM, "mean": AGG_SUM,
             "min": AGG_MIN, "max": AGG_MAX}
AGG_IDENTITY = {AGG_SUM: 0.0, AGG_MIN: float("inf"), AGG_MAX: float("-inf")}


def groupby_accum(keys: ColumnRef, vals: list, agg_op: int, key_min: int,
                  n_slots: int, sums: int, rowcnt: int, counts: int) -> None:
    ensure_ready()
    arr = (ct.c_void_p * max(len(vals), 1))(*[v.handle for v in vals])
    _check(load().hf_groupby_accum(keys.handle, arr, len(vals), agg_op,
                                   key_min, n_slots, sums, rowcnt, counts),
           "hf_groupby_accum")


def fill_f64(dptr: int, value: float, n: int) -> None:
    ensure_ready()
    _check(load().hf_fill_f64(dptr, value, n), "hf_fill_f64")


def fill_i64(dptr: int, value: int, n: int) -> None:
    ensure_ready()
    _check(load().hf_fill_i64(dptr, value, n), "hf_fill_i64")


def fill_randint(n: int, seed: int, lo: int, hi: int) -> ColumnRef:
    """Device-generated int64 column: lo + splitmix64(seed+i) % (hi-lo)
    (oracle.rand_int mirrors this bit-exactly)."""
    col = alloc(n, HF_INT64)
    _check(load().hf_fill_randint(col.handle, seed, lo, hi),
           "hf_fill_randint")
    return col


def fill_randf64(n: int, seed: int) -> ColumnRef:
    """Device-generated float64 column uniform in [0,1) (oracle.rand_f64)."""
    col = alloc(n, HF_FLOAT64)
    _check(load().hf_fill_randf64(col.handle, seed), "hf_fill_randf64")
    return col


def fill_randcdf(n: int, seed: int, cdf: ColumnRef) -> ColumnRef:
    """Device inverse-CDF draw: searchsorted(cdf, U[0,1), side='right')
    (oracle.rand_cdf)."""
    col = alloc(n, HF_INT64)
    _check(load().hf_fill_randcdf(col.handle, seed, cdf.handle),
           "hf_fill_randcdf")
    return col


def groupby_hash_accum(keys: ColumnRef, vals: list, agg_op: int, H: int,
                       tkey: int, sums: int, rowcnt: int, counts: int) -> None:
    ensure_ready()
    arr = (ct.c_void_p * max(len(vals), 1))(*[v.handle for v in vals])
    _check(load().hf_groupby_hash_accum(keys.handle, arr, len(vals), agg_op,
                                        H, tkey, sums, rowcnt, counts),
           "hf_groupby_hash_accum")


def groupby_hash_compact(tkey: int, sums: int, rowcnt: int, counts: int,
                         nvals: int, H: int):
    ensure_ready()
    out_keys = ct.c_void_p()
    out_sums = (ct.c_void_p * max(nvals, 1))()
    out_counts = (ct.c_void_p * max(nvals, 1))()
    n_groups = ct.c_int64(0)
    _check(load().hf_groupby_hash_compact(tkey, sums, rowcnt, counts, nvals,
                                          H, ct.byref(out_keys), out_sums,
                                          out_counts if counts else None,
                                          ct.byref(n_groups)),
           "hf_groupby_hash_compact")
    n = n_groups.value
    kcol = _wrap(out_keys, n, HF_INT64)
    scols = [_wrap(ct.c_void_p(out_sums[c]), n, HF_FLOAT64)
             for c in range(nvals)]
    ccols = ([_wrap(ct.c_void_p(out_counts[c]), n, HF_INT64)
              for c in range(nvals)] if counts else None)
    return kcol, scols, ccols, n


def groupby_sorted(sorted_keys: ColumnRef, vals: list, agg_op: int,
                   want_counts: bool):
    """Segmented aggregation over KEY-SORTED rows — the unbounded-cardinality
    groupby (runs of equal keys are groups).  Returns (keys, sums, counts, n)
    like groupby_hash_compact; keys come out ascending because the input is."""
    ensure_ready()
    nv = len(vals)
    arr = (ct.c_void_p * max(nv, 1))(*[v.handle for v in vals])
    out_keys = ct.c_void_p()
    out_sums = (ct.c_void_p * max(nv, 1))()
    out_counts = (ct.c_void_p * max(nv, 1))()
    n_groups = ct.c_int64(0)
    _check(load().hf_groupby_sorted(sorted_keys.handle, arr, nv, agg_op,
                                    1 if want_counts else 0,
                                    ct.byref(out_keys), out_sums,
                                    out_counts if want_counts else None,
                                    ct.byref(n_groups)),
           "hf_groupby_sorted")
    n = n_groups.value
    kcol = _wrap(out_keys, n, HF_INT64)
    scols = [_wrap(ct.c_void_p(out_sums[c]), n, HF_FLOAT64) for c in range(nv)]
    ccols = ([_wrap(ct.c_void_p(out_counts[c]), n, HF_INT64)
              for c in range(nv)] if want_counts else None)
    return kcol, scols, ccols, n


def fixup_empty(val: ColumnRef, cnt: ColumnRef) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_fixup_empty(val.handle, cnt.handle, ct.byref(out)),
           "hf_fixup_empty")
    return _wrap(out, val.length, HF_FLOAT64)


def groupby_compact(sums: int, rowcnt: int, counts: int, nvals: int,
                    key_min: int, n_slots: int):
    """Returns (keys_col, [sum_cols], [count_cols] or None, n_groups)."""
    ensure_ready()
    out_keys = ct.c_void_p()
    out_sums = (ct.c_void_p * max(nvals, 1))()
    out_counts = (ct.c_void_p * max(nvals, 1))()
    n_groups = ct.c_int64(0)
    _check(load().hf_groupby_compact(sums, rowcnt, counts, nvals, key_min,
                                     n_slots, ct.byref(out_keys), out_sums,
                                     out_counts if counts else None,
                                     ct.byref(n_groups)), "hf_groupby_compact")
    n = n_groups.value
    kcol = _wrap(out_keys, n, HF_INT64)
    scols = [_wrap(ct.c_void_p(out_sums[c]), n, HF_FLOAT64) for c in range(nvals)]
    ccols = ([_wrap(ct.c_void_p(out_counts[c]), n, HF_INT64) for c in range(nvals)]
             if counts else None)
    return kcol, scols, ccols, n


def col_slice(col: ColumnRef, start: int, length: int) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_col_slice(col.handle, start, length, ct.byref(out)),
           "hf_col_slice")
    return _wrap(out, length, col.dtype_code)


def concat(cols: list) -> ColumnRef:
    ensure_ready()
    arr = (ct.c_void_p * len(cols))(*[c.handle for c in cols])
    out = ct.c_void_p()
    _check(load().hf_col_concat(arr, len(cols), ct.byref(out)), "hf_col_concat")
    return _wrap(out, sum(c.length for c in cols), cols[0].dtype_code)


class JoinRef:
    """Owner of a built broadcast-right join structure (hf_join)."""

    __slots__ = ("handle", "nr", "rdtypes")

    def __init__(self, handle, nr, rdtypes):
        self.handle = handle
        self.nr = nr
        self.rdtypes = rdtypes

    def __del__(self):
        try:
            if _dll is not None and _inited_gpu is not None and self.handle:
                _dll.hf_join_free(self.handle)
        except Exception:
            pass


def join_build(rkeys: ColumnRef, rvals: list, key_min: int,
               n_slots: int) -> JoinRef:
    ensure_ready()
    arr = (ct.c_void_p * max(len(rvals), 1))(*[v.handle for v in rvals])
    out = ct.c_void_p()
    _check(load().hf_join_build(rkeys.handle, arr, len(rvals), key_min,
                                n_slots, ct.byref(out)), "hf_join_build")
    return JoinRef(out, len(rvals), [v.dtype_code for v in rvals])


def join_probe(j: JoinRef, lkeys: ColumnRef):
    """Returns (keys_col, lidx_col, [right_cols], n_out)."""
    ensure_ready()
    out_keys = ct.c_void_p()
    out_lidx = ct.c_void_p()
    out_r = (ct.c_void_p * max(j.nr, 1))()
    n_out = ct.c_int64(0)
    _check(load().hf_join_probe(j.handle, lkeys.handle, ct.byref(out_keys),
                                ct.byref(out_lidx), out_r, ct.byref(n_out)),
           "hf_join_probe")
    n = n_out.value
    rcols = [_wrap(ct.c_void_p(out_r[c]), n, j.rdtypes[c]) for c in range(j.nr)]
    return (_wrap(out_keys, n, HF_INT64), _wrap(out_lidx, n, HF_INT64),
            rcols, n)


# compare ops (include/hipframe.h)
CMP_GT, CMP_GE, CMP_LT, CMP_LE, CMP_EQ, CMP_NE, CMP_NOTNA = range(7)


def compare_scalar(op: int, col: ColumnRef, scalar: float) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_compare_scalar(op, col.handle, float(scalar),
                                    ct.byref(out)), "hf_compare_scalar")
    return _wrap(out, col.length, HF_INT64)


class FilterPlan:
    """Owner of an hf_filterplan; keeps the mask column alive (the plan
    borrows its device pointer)."""

    __slots__ = ("handle", "mask", "n_kept")

    def __init__(self, handle, mask, n_kept):
        self.handle = handle
        self.mask = mask
        self.n_kept = n_kept

    def __del__(self):
        try:
            if _dll is not None and _inited_gpu is not None and self.handle:
                _dll.hf_filter_plan_free(self.handle)
        except Exception:
            pass


def filter_plan(mask: ColumnRef) -> FilterPlan:
    ensure_ready()
    out = ct.c_void_p()
    n = ct.c_int64(0)
    _check(load().hf_filter_plan(mask.handle, ct.byref(out), ct.byref(n)),
           "hf_filter_plan")
    return FilterPlan(out, mask, n.value)


def filter_apply(plan: FilterPlan, col: ColumnRef) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_filter_apply(plan.handle, col.handle, ct.byref(out)),
           "hf_filter_apply")
    return _wrap(out, plan.n_kept, col.dtype_code)


def filter_iota(plan: FilterPlan, base: int) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_filter_iota(plan.handle, base, ct.byref(out)),
           "hf_filter_iota")
    return _wrap(out, plan.n_kept, HF_INT64)


def cumsum(col: ColumnRef, agg_op: int = 0) -> ColumnRef:
    """Inclusive prefix scan (pandas cumsum/cummin/cummax axis=0,
    agg_op AGG_SUM/MIN/MAX): f64 skips NaN, i64 exact."""
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_cumsum(col.handle, agg_op, ct.byref(out)),
           "hf_cumsum")
    return _wrap(out, col.length, col.dtype_code)


def seg_cumsum(col: ColumnRef, heads: ColumnRef,
               agg_op: int = 0) -> ColumnRef:
    """Segmented inclusive prefix scan: restart at rows whose `heads` entry
    is nonzero (groupby cumsum/cummin/cummax after a stable sort by key).
    f64 skips NaN, i64 exact."""
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_seg_cumsum(col.handle, heads.handle, agg_op,
                                ct.byref(out)), "hf_seg_cumsum")
    return _wrap(out, col.length, col.dtype_code)


def cross_idx(nl: int, nr: int):
    """(lidx, ridx) gather indices of the nl x nr cartesian product
    (merge how='cross')."""
    ensure_ready()
    li, ri = ct.c_void_p(), ct.c_void_p()
    _check(load().hf_cross_idx(nl, nr, ct.byref(li), ct.byref(ri)),
           "hf_cross_idx")
    return _wrap(li, nl * nr, HF_INT64), _wrap(ri, nl * nr, HF_INT64)


def scatter(col: ColumnRef, idx: ColumnRef) -> ColumnRef:
    """out[idx[i]] = col[i]; idx must be a permutation of [0, len) — the
    inverse of gather(col, idx), restoring pre-sort row order."""
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_scatter(col.handle, idx.handle, ct.byref(out)),
           "hf_scatter")
    return _wrap(out, col.length, col.dtype_code)


def ordered_i64(col: ColumnRef, inverse: bool = False) -> ColumnRef:
    """Order-preserving f64 <-> i64 bit transform (float keys on the int64
    radix machinery; -0.0 == +0.0; NaN above +inf)."""
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_ordered_i64(col.handle, 1 if inverse else 0,
                                 ct.byref(out)), "hf_ordered_i64")
    return _wrap(out, col.length, HF_FLOAT64 if inverse else HF_INT64)


def ordered_to_f64_np(keys_np):
    """Host inverse of the ordered transform (decode compacted group
    keys)."""
    s_ = np.asarray(keys_np, dtype=np.int64)
    bits = np.where(s_ < 0, ~(s_ ^ np.int64(-2**63)), s_)
    return bits.view(np.float64)


def search_sorted(keys: ColumnRef, sorted_uniq: ColumnRef) -> ColumnRef:
    """out[i] = exact-match index of keys[i] in sorted_uniq, else -1."""
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_search_sorted(keys.handle, sorted_uniq.handle,
                                   ct.byref(out)), "hf_search_sorted")
    return _wrap(out, keys.length, HF_INT64)


def shuffle_dest(keys: ColumnRef, splitters) -> ColumnRef:
    """dest[i] = #{j: splitters[j] <= key[i]} — destination-rank binning for
    the range shuffle (native int64 compares; exact over the full span)."""
    ensure_ready()
    spl = np.asarray(splitters, dtype=np.int64)
    arr = (ct.c_int64 * max(len(spl), 1))(*spl.tolist())
    out = ct.c_void_p()
    _check(load().hf_shuffle_dest(keys.handle, arr, len(spl), ct.byref(out)),
           "hf_shuffle_dest")
    return _wrap(out, keys.length, HF_INT64)


def memcpy_dd(dst_ptr: int, src_ptr: int, nbytes: int) -> None:
    """Device-to-device copy on the hipframe stream (torch-buffer interop)."""
    ensure_ready()
    _check(load().hf_memcpy_dd(dst_ptr, src_ptr, nbytes), "hf_memcpy_dd")


def sort_perm(keys: ColumnRef, ascending: bool = True) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_sort_perm(keys.handle, 1 if ascending else 0,
                               ct.byref(out)), "hf_sort_perm")
    return _wrap(out, keys.length, HF_INT64)


def gather(col: ColumnRef, idx: ColumnRef) -> ColumnRef:
    ensure_ready()
    out = ct.c_void_p()
    _check(load().hf_gather(col.handle, idx.handle, ct.byref(out)), "hf_gather")
    return _wrap(out, idx.length, col.dtype_code)


def sync() -> None:
    ensure_ready()
    _check(load().hf_sync(), "hf_sync")


def profiling(enable: bool) -> None:
    ensure_ready()
    _check(load().hf_profiling(1 if enable else 0), "hf_profiling")


def kernel_stats(name: str):
    ensure_ready()
    n = ct.c_int64(0)
    ms = ct.c_double(0.0)
    _check(load().hf_kernel_stats(name.encode(), ct.byref(n), ct.byref(ms)),
           "hf_kernel_stats")
    return n.value, ms.value


def kernel_stats_reset() -> None:
    ensure_ready()
    _check(load().hf_kernel_stats_reset(), "hf_kernel_stats_reset")
