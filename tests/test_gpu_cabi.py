"""GPU tier: direct C-ABI kernel checks against numpy (small sizes).

These bypass the host dataframe stack and exercise each hipframe entry point
through ctypes exactly as modin_amd calls it.
"""

import numpy as np
import pytest

import oracle
from modin_amd.core import lib

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _ready(gpu_ready):
    yield


def test_put_get_roundtrip():
    rng = np.random.default_rng(0)
    for arr in [rng.random(10_001), rng.integers(-5, 5, 4097).astype(np.int64),
                np.array([], dtype=np.float64)]:
        col = lib.put(arr)
        back = lib.get(col)
        np.testing.assert_array_equal(arr, back)


@pytest.mark.parametrize("op,code,scalar", [
    ("add", lib.MAP_ADD, 1.5), ("sub", lib.MAP_SUB, 2.25),
    ("rsub", lib.MAP_RSUB, 1.0), ("mul", lib.MAP_MUL, -3.5),
    ("div", lib.MAP_DIV, 2.0), ("rdiv", lib.MAP_RDIV, 7.0),
    ("fillna", lib.MAP_FILLNA, 0.5), ("abs", lib.MAP_ABS, 0.0),
    ("neg", lib.MAP_NEG, 0.0),
])
def test_map_f64_bitexact(op, code, scalar):
    import oracle
    rng = np.random.default_rng(1)
    x = rng.random(100_003) * 20 - 10
    x[rng.random(x.size) < 0.1] = np.nan
    col = lib.put(x)
    out = lib.get(lib.map_scalar(code, col, scalar))
    exp = oracle.map_op(op, x, scalar)
    np.testing.assert_array_equal(out, exp)  # single-op IEEE: bit-exact


@pytest.mark.parametrize("op,code,scalar", [
    ("add", lib.MAP_ADD, 7), ("sub", lib.MAP_SUB, -3), ("mul", lib.MAP_MUL, 11),
    ("abs", lib.MAP_ABS, 0), ("neg", lib.MAP_NEG, 0),
])
def test_map_i64_exact(op, code, scalar):
    import oracle
    rng = np.random.default_rng(2)
    x = rng.integers(-10**9, 10**9, 65_537).astype(np.int64)
    col = lib.put(x)
    out = lib.get(lib.map_scalar(code, col, scalar))
    np.testing.assert_array_equal(out, oracle.map_op(op, x, scalar))


@pytest.mark.parametrize("op,code", [
    ("add", lib.BIN_ADD), ("sub", lib.BIN_SUB), ("mul", lib.BIN_MUL),
    ("div", lib.BIN_DIV),
])
def test_binary_f64_bitexact(op, code):
    import oracle
    rng = np.random.default_rng(3)
    a = rng.random(50_001) * 10
    b = rng.random(50_001) + 0.5
    out = lib.get(lib.binary(code, lib.put(a), lib.put(b)))
    np.testing.assert_array_equal(out, oracle.binary_op(op, a, b))


def test_reduce_f64_vs_numpy():
    rng = np.random.default_rng(4)
    x = rng.random(1_000_001) * 100 - 50
    x[rng.random(x.size) < 0.05] = np.nan
    r = lib.reduce(lib.put(x))
    valid = ~np.isnan(x)
    np.testing.assert_allclose(r.sum, x[valid].sum(), rtol=1e-12)
    assert r.count == int(valid.sum())
    assert r.mn == x[valid].min() and r.mx == x[valid].max()


def test_reduce_i64_exact():
    rng = np.random.default_rng(5)
    x = rng.integers(-10**12, 10**12, 999_999).astype(np.int64)
    r = lib.reduce(lib.put(x))
    assert r.isum == int(x.sum())
    assert r.imn == int(x.min()) and r.imx == int(x.max())
    assert r.count == x.size


def test_reduce_empty_and_allnan():
    r = lib.reduce(lib.put(np.array([], dtype=np.float64)))
    assert r.count == 0 and r.sum == 0.0
    r = lib.reduce(lib.put(np.full(100, np.nan)))
    assert r.count == 0 and r.sum == 0.0


def _gb_via_cabi(keys, vals_dict, want_counts):
    key_col = lib.put(keys)
    val_cols = [lib.put(v) for v in vals_dict.values()]
    r = lib.reduce(key_col)
    kmin, kmax = r.imn, r.imx
    n_slots = kmax - kmin + 1
    nv = len(val_cols)
    sums = lib.alloc_raw(8 * nv * n_slots)
    rowcnt = lib.alloc_raw(8 * n_slots)
    counts = lib.alloc_raw(8 * nv * n_slots) if want_counts else 0
    lib.memset_raw(sums, 0, 8 * nv * n_slots)
    lib.memset_raw(rowcnt, 0, 8 * n_slots)
    if want_counts:
        lib.memset_raw(counts, 0, 8 * nv * n_slots)
    lib.groupby_accum(key_col, val_cols, lib.AGG_SUM, kmin, n_slots, sums,
                      rowcnt, counts)
    keys_out, sums_out, counts_out, n = lib.groupby_compact(
        sums, rowcnt, counts, nv, kmin, n_slots)
    got_keys = lib.get(keys_out)
    got_sums = {name: lib.get(sums_out[i]) for i, name in enumerate(vals_dict)}
    got_counts = ({name: lib.get(counts_out[i]) for i, name in enumerate(vals_dict)}
                  if want_counts else None)
    lib.free_raw(sums)
    lib.free_raw(rowcnt)
    if counts:
        lib.free_raw(counts)
    return got_keys, got_sums, got_counts


@pytest.mark.parametrize("nkeys", [1, 13, 1000])
def test_groupby_accum_compact_vs_oracle(nkeys):
    import oracle
    rng = np.random.default_rng(6)
    n = 300_000
    keys = rng.integers(-nkeys // 2, max(nkeys // 2, 1) + 1, n).astype(np.int64)
    v = rng.random(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.standard_normal(n)
    got_keys, got_sums, got_counts = _gb_via_cabi(keys, {"v": v, "w": w}, True)
    ok, osums = oracle.groupby_agg(keys, {"v": v, "w": w}, "sum")
    _, ocnts = oracle.groupby_agg(keys, {"v": v, "w": w}, "count")
    np.testing.assert_array_equal(got_keys, ok)
    # atol covers near-zero sums of the standard-normal column (cancellation
    # makes rtol meaningless there; fp error scales with sum(|x|) ~ 1e2)
    np.testing.assert_allclose(got_sums["v"], osums["v"], rtol=1e-12, atol=1e-9)
    np.testing.assert_allclose(got_sums["w"], osums["w"], rtol=1e-12, atol=1e-9)
    np.testing.assert_array_equal(got_counts["v"], ocnts["v"])
    np.testing.assert_array_equal(got_counts["w"], ocnts["w"])


def test_groupby_out_of_range_key_detected():
    keys = np.array([0, 1, 2, 99], dtype=np.int64)
    v = np.ones(4)
    key_col, val_col = lib.put(keys), lib.put(v)
    sums = lib.alloc_raw(8 * 10)
    rowcnt = lib.alloc_raw(8 * 10)
    lib.memset_raw(sums, 0, 80)
    lib.memset_raw(rowcnt, 0, 80)
    lib.groupby_accum(key_col, [val_col], lib.AGG_SUM, 0, 10, sums, rowcnt, 0)
    with pytest.raises(lib.HfError, match="outside"):
        lib.groupby_compact(sums, rowcnt, 0, 1, 0, 10)
    lib.free_raw(sums)
    lib.free_raw(rowcnt)


def test_kernel_stats_profiling():
    lib.profiling(True)
    lib.kernel_stats_reset()
    x = np.random.default_rng(7).random(1 << 20)
    col = lib.put(x)
    for _ in range(3):
        lib.map_scalar(lib.MAP_ADD, col, 1.0)
    n, ms = lib.kernel_stats("map_f64")
    assert n == 3 and ms > 0
    lib.profiling(False)


def test_shuffle_dest_kernel_vs_oracle():
    """Native-i64 splitter binning (k_shuffle_dest) — exact at magnitudes
    where an f64 compare would mis-bin (keys within 1 of a 2^62-scale
    splitter)."""
    rng = np.random.default_rng(17)
    spl = np.sort(rng.integers(-2**62, 2**62, 7)).astype(np.int64)
    keys = rng.integers(-2**62, 2**62, 100_000).astype(np.int64)
    # adversarial: keys exactly at / one-off the splitters
    edge = np.concatenate([spl, spl - 1, spl + 1]).astype(np.int64)
    keys = np.concatenate([keys, edge])
    kcol = lib.put(keys)
    dest = lib.get(lib.shuffle_dest(kcol, spl))
    np.testing.assert_array_equal(dest, oracle.shuffle_dest(keys, spl))
    # empty splitter list -> all zeros
    np.testing.assert_array_equal(
        lib.get(lib.shuffle_dest(kcol, np.empty(0, dtype=np.int64))), 0)


def test_memcpy_dd_roundtrip():
    rng = np.random.default_rng(18)
    a = rng.random(10_000)
    src = lib.put(a)
    dst = lib.alloc(10_000, lib.HF_FLOAT64)
    lib.memcpy_dd(dst.dptr(), src.dptr(), 8 * 10_000)
    np.testing.assert_array_equal(lib.get(dst), a)
