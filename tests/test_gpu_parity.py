"""GPU tier: full-stack parity — modin_amd.pandas vs the committed golden
vectors (generated from the REAL reference) and vs the oracle, plus
size-independent property checks at larger sizes.

Tolerance contract: int64 paths and group keys/counts
bit-exact; fp64 map/binary bit-exact (single IEEE op per element); fp64
sum/mean within rtol=1e-12 (both the reference and this backend reassociate
partition/atomic-order sums; the reference itself deviates ~1e-15 from
pandas — SURVEY.md §4).
"""

import numpy as np
import pandas
import pytest

import modin_amd.config as config
import modin_amd.pandas as mpd
import oracle
from modin_amd.core import lib
from tests.conftest import golden_cases, load_golden

pytestmark = pytest.mark.gpu

RTOL = 1e-12


@pytest.fixture(autouse=True)
def _ready(gpu_ready):
    yield


@pytest.fixture(params=[1, 3], ids=["np1", "np3"])
def npartitions(request):
    old = config.NPartitions.get()
    config.NPartitions.put(request.param)
    yield request.param
    config.NPartitions.put(old)


def _gb_inputs(g):
    cols = {"k": g["in_k"]}
    cols.update({k[3:]: v for k, v in g.items()
                 if k.startswith("in_") and k != "in_k"})
    return cols


@pytest.mark.parametrize("case", golden_cases("gb_"))
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_groupby_vs_golden(case, agg, npartitions):
    g = load_golden(case)
    df = mpd.DataFrame(_gb_inputs(g))
    out = getattr(df.groupby("k"), agg)().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g[f"out_{agg}_keys"])
    assert out.index.name == "k"
    for name in out.columns:
        expect = g[f"out_{agg}_{name}"]
        if agg == "count":
            np.testing.assert_array_equal(out[name].to_numpy(),
                                          expect.astype(np.int64))
        else:
            np.testing.assert_allclose(out[name].to_numpy(), expect, rtol=RTOL,
                                       atol=1e-9, equal_nan=True)


@pytest.mark.parametrize("case", golden_cases("red_"))
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_reduce_vs_golden(case, agg, npartitions):
    g = load_golden(case)
    names = [k[3:] for k in g if k.startswith("in_")]
    df = mpd.DataFrame({n: g[f"in_{n}"] for n in names})
    got = getattr(df, agg)()
    np.testing.assert_allclose(np.asarray(got, dtype=float), g[f"out_{agg}"],
                               rtol=RTOL, equal_nan=True)


def test_map_binary_vs_golden(npartitions):
    g = load_golden("map_binary")
    df = mpd.DataFrame({"v": g["in_v"], "w": g["in_w"]})
    for tag, expr in [
        ("add1", lambda d: d + 1), ("mul2", lambda d: d * 2.5),
        ("sub3", lambda d: d - 3.25), ("div2", lambda d: d / 2.0),
        ("rsub", lambda d: 1.0 - d), ("fill0", lambda d: d.fillna(0.0)),
        ("fillm1", lambda d: d.fillna(-1.5)), ("abs", lambda d: d.abs()),
        ("frame_add", lambda d: d + d), ("frame_mul", lambda d: d * d),
        ("frame_div", lambda d: d / (d + 10.0)),
    ]:
        out = expr(df).to_pandas()
        np.testing.assert_array_equal(out["v"].to_numpy(), g[f"out_{tag}_v"],
                                      err_msg=tag)
        np.testing.assert_array_equal(out["w"].to_numpy(), g[f"out_{tag}_w"],
                                      err_msg=tag)
    di = mpd.DataFrame({"i": g["in_i"]})
    np.testing.assert_array_equal((di + 7).to_pandas()["i"], g["out_iadd_i"])
    np.testing.assert_array_equal((di * -3).to_pandas()["i"], g["out_imul_i"])
    np.testing.assert_array_equal(di.abs().to_pandas()["i"], g["out_iabs_i"])


def test_roundtrip_index_dtypes(npartitions):
    rng = np.random.default_rng(11)
    pdf = pandas.DataFrame({"a": rng.integers(0, 10, 1000).astype(np.int64),
                            "b": rng.random(1000)})
    df = mpd.from_pandas(pdf)
    back = df.to_pandas()
    pandas.testing.assert_frame_equal(back, pdf)
    assert df.shape == pdf.shape
    assert list(df.dtypes) == list(pdf.dtypes)


def test_column_projection(npartitions):
    rng = np.random.default_rng(12)
    df = mpd.DataFrame({"k": rng.integers(0, 9, 500).astype(np.int64),
                        "v": rng.random(500), "w": rng.random(500)})
    s = df["v"]
    assert float(s.sum()) == pytest.approx(
        oracle.reduce_op("sum", df.to_pandas()["v"].to_numpy()), rel=RTOL)
    sub = df[["k", "v"]].to_pandas()
    assert list(sub.columns) == ["k", "v"]


# ---- property checks at sizes the oracle still runs in seconds ----

def test_property_groupby_medium():
    rng = np.random.default_rng(13)
    n = 2_000_000
    k = rng.integers(0, 10**5, n).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.groupby("k").sum().to_pandas()
    # sum of group sums == total sum (linearity), and keys are complete
    np.testing.assert_allclose(out["v"].sum(), v.sum(), rtol=1e-10)
    assert out.index.size == np.unique(k).size
    ok, osums = oracle.groupby_agg(k, {"v": v}, "sum")
    np.testing.assert_array_equal(out.index.to_numpy(), ok)
    np.testing.assert_allclose(out["v"].to_numpy(), osums["v"], rtol=RTOL)


def test_property_zipf_skew():
    rng = np.random.default_rng(14)
    n = 2_000_000
    k = np.minimum(rng.zipf(1.2, n), 10**6).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.groupby("k").sum().to_pandas()
    ok, osums = oracle.groupby_agg(k, {"v": v}, "sum")
    np.testing.assert_array_equal(out.index.to_numpy(), ok)
    np.testing.assert_allclose(out["v"].to_numpy(), osums["v"], rtol=1e-9)


def test_property_map_roundtrip_idempotence():
    rng = np.random.default_rng(15)
    v = rng.random(3_000_000)
    v[rng.random(v.size) < 0.01] = np.nan
    df = mpd.DataFrame({"v": v})
    once = df.fillna(0.0)
    twice = once.fillna(5.0)  # no NaNs left: must be identical
    np.testing.assert_array_equal(once.to_pandas()["v"].to_numpy(),
                                  twice.to_pandas()["v"].to_numpy())


def test_error_surfaces():
    rng = np.random.default_rng(16)
    # float groupby keys ride the ordered transform (was a loud error in
    # early round 1; now parity-checked here against pandas)
    kf = rng.random(100)
    vf = rng.random(100)
    pdf = pandas.DataFrame({"k": kf, "v": vf})
    out = mpd.DataFrame(pdf).groupby("k").sum().to_pandas()
    exp = pdf.groupby("k").sum()
    np.testing.assert_array_equal(out.index.to_numpy(),
                                  exp.index.to_numpy())
    np.testing.assert_allclose(out["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=RTOL)
    dfi = mpd.DataFrame({"k": rng.integers(0, 5, 100).astype(np.int64),
                         "v": rng.random(100)})
    # prod is supported since round 2 — genuinely-missing aggs stay loud
    with pytest.raises(lib.HfError, match="not implemented"):
        dfi.groupby("k").agg("sem")
    # key range beyond the dense-table cap routes to the hash path
    old = config.MaxGroupbySlots.get()
    config.MaxGroupbySlots.put(10)
    try:
        wide = mpd.DataFrame({"k": np.array([0, 10**7, 0], dtype=np.int64),
                              "v": np.array([1.0, 2.0, 3.0])})
        out = wide.groupby("k").sum().to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(), [0, 10**7])
        np.testing.assert_allclose(out["v"].to_numpy(), [4.0, 2.0])
    finally:
        config.MaxGroupbySlots.put(old)


@pytest.mark.parametrize("case", golden_cases("mg_"))
def test_merge_vs_golden(case, npartitions):
    g = load_golden(case)
    lcols = {k[len("in_l_"):]: v for k, v in g.items() if k.startswith("in_l_")}
    rcols = {k[len("in_r_"):]: v for k, v in g.items() if k.startswith("in_r_")}
    left = mpd.DataFrame({"k": g["in_lk"], **lcols})
    right = mpd.DataFrame({"k": g["in_rk"], **rcols})
    out = left.merge(right, on="k").to_pandas()
    expect_cols = [str(c) for c in g["out_columns"]]
    assert list(out.columns) == expect_cols
    assert (out.index == pandas.RangeIndex(len(out))).all()
    for c in expect_cols:
        np.testing.assert_array_equal(out[c].to_numpy(), g[f"out_{c}"],
                                      err_msg=c)
        assert out[c].dtype == g[f"out_{c}"].dtype


def test_merge_error_surfaces():
    rng = np.random.default_rng(21)
    right = mpd.DataFrame({"k": rng.integers(0, 5, 10).astype(np.int64),
                           "w": rng.random(10)})
    ok = mpd.DataFrame({"k": rng.integers(0, 5, 10).astype(np.int64),
                        "v": rng.random(10)})
    with pytest.raises(lib.HfError, match="forbids"):
        ok.merge(right, on="k", how="cross")
    with pytest.raises(lib.HfError, match="required"):
        ok.merge(right)


def test_filter_vs_golden(npartitions):
    g = load_golden("flt_basic")
    df = mpd.DataFrame({"v": g["in_v"], "w": g["in_w"], "i": g["in_i"]})
    for tag, mask in [
        ("gt", df["v"] > 0.25), ("le", df["v"] <= 0.5),
        ("eq", df["i"] == 3), ("ne", df["v"] != 0.0),
        ("none", df["v"] > 2.0),
    ]:
        m = mask.to_pandas()
        assert m.dtype == bool
        np.testing.assert_array_equal(m.to_numpy(), g[f"out_mask_{tag}"],
                                      err_msg=tag)
        out = df[mask].to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_idx_{tag}"], err_msg=tag)
        for c in ("v", "w", "i"):
            np.testing.assert_array_equal(out[c].to_numpy(),
                                          g[f"out_{tag}_{c}"],
                                          err_msg=f"{tag}/{c}")
            assert out[c].dtype == g[f"out_{tag}_{c}"].dtype


def test_filter_then_groupby(npartitions):
    """Composition: df[df.v > .5].groupby('k').sum() against the oracle."""
    rng = np.random.default_rng(33)
    n = 300_000
    k = rng.integers(0, 500, n).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df[df["v"] > 0.5].groupby("k").sum().to_pandas()
    keep = v > 0.5
    ok, osums = oracle.groupby_agg(k[keep], {"v": v[keep]}, "sum")
    np.testing.assert_array_equal(out.index.to_numpy(), ok)
    np.testing.assert_allclose(out["v"].to_numpy(), osums["v"], rtol=RTOL)


def test_int_div_promotes():
    rng = np.random.default_rng(40)
    i = rng.integers(1, 100, 1000).astype(np.int64)
    df = mpd.DataFrame({"i": i})
    out = (df / 4).to_pandas()
    np.testing.assert_array_equal(out["i"].to_numpy(), i / 4)
    assert out["i"].dtype == np.float64


@pytest.mark.parametrize("span", [8191, 8192, 8193, 10001, 4096 * 3 + 1])
def test_groupby_dense_radix_boundary(span):
    """Exercise both sides of the dense/radix switch and bucket-edge keys
    (range not a multiple of GB_RANGE; negative key_min)."""
    rng = np.random.default_rng(41)
    n = 400_000
    k = (rng.integers(0, span, n) - span // 2).astype(np.int64)
    # pin the exact range ends so n_slots == span
    k[0], k[1] = -(span // 2), span - 1 - span // 2
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.groupby("k").sum().to_pandas()
    ok, osums = oracle.groupby_agg(k, {"v": v}, "sum")
    np.testing.assert_array_equal(out.index.to_numpy(), ok)
    np.testing.assert_allclose(out["v"].to_numpy(), osums["v"], rtol=RTOL,
                               atol=1e-9)


def test_dropna_and_mask_algebra(npartitions):
    rng = np.random.default_rng(42)
    n = 20_000
    v = rng.random(n)
    v[rng.random(n) < 0.15] = np.nan
    w = rng.random(n)
    k = rng.integers(0, 7, n).astype(np.int64)
    df = mpd.DataFrame({"k": k, "v": v, "w": w})
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    pandas.testing.assert_frame_equal(df.dropna().to_pandas(), pdf.dropna())
    m = (df["v"] > 0.3) & (df["w"] <= 0.9)
    pm = (pdf["v"] > 0.3) & (pdf["w"] <= 0.9)
    np.testing.assert_array_equal(m.to_pandas().to_numpy(), pm.to_numpy())
    m2 = (df["v"] > 0.5) | ~(df["w"] > 0.2)
    pm2 = (pdf["v"] > 0.5) | ~(pdf["w"] > 0.2)
    np.testing.assert_array_equal(m2.to_pandas().to_numpy(), pm2.to_numpy())
    pandas.testing.assert_frame_equal(df[m2].to_pandas(), pdf[pm2])
    s = df["v"].isna()
    np.testing.assert_array_equal(s.to_pandas().to_numpy(),
                                  pdf["v"].isna().to_numpy())


def test_sort_vs_golden(npartitions):
    g = load_golden("srt_basic")
    df = mpd.DataFrame({"k": g["in_k"], "v": g["in_v"], "i": g["in_i"]})
    for tag, asc in [("asc", True), ("desc", False)]:
        out = df.sort_values("k", ascending=asc).to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_idx_{tag}"], err_msg=tag)
        for c in ("k", "v", "i"):
            np.testing.assert_array_equal(out[c].to_numpy(),
                                          g[f"out_{tag}_{c}"],
                                          err_msg=f"{tag}/{c}")
            assert out[c].dtype == g[f"out_{tag}_{c}"].dtype
    dfn = mpd.DataFrame({"k": g["in_kn"], "v": g["in_v"]})
    out = dfn.sort_values("k").to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_neg_idx"])
    np.testing.assert_array_equal(out["k"].to_numpy(), g["out_neg_k"])


def test_sort_property_large():
    """Stability + correctness at a multi-pass size (3 radix passes) against
    the oracle, exact."""
    rng = np.random.default_rng(50)
    n = 2_000_000
    k = rng.integers(0, 10**6, n).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.sort_values("k").to_pandas()
    perm = oracle.sort_perm(k)
    np.testing.assert_array_equal(out.index.to_numpy(), perm)
    np.testing.assert_array_equal(out["k"].to_numpy(), k[perm])
    np.testing.assert_array_equal(out["v"].to_numpy(), v[perm])
    # single-key edge (zero radix passes)
    one = mpd.DataFrame({"k": np.full(1000, 7, dtype=np.int64),
                         "v": rng.random(1000)})
    res = one.sort_values("k").to_pandas()
    np.testing.assert_array_equal(res.index.to_numpy(), np.arange(1000))


@pytest.mark.parametrize("case", golden_cases("hh_"))
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_hash_groupby_vs_golden(case, agg, npartitions):
    """Unbounded key ranges: the open-addressing hash path + wide sort."""
    g = load_golden(case)
    df = mpd.DataFrame(_gb_inputs(g))
    out = getattr(df.groupby("k"), agg)().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g[f"out_{agg}_keys"])
    for name in out.columns:
        expect = g[f"out_{agg}_{name}"]
        if agg == "count":
            np.testing.assert_array_equal(out[name].to_numpy(),
                                          expect.astype(np.int64))
        else:
            np.testing.assert_allclose(out[name].to_numpy(), expect,
                                       rtol=RTOL, atol=1e-9, equal_nan=True)


def test_hash_groupby_property_medium():
    """1M rows, ~500K distinct huge keys (exercises probe collisions and the
    grow-retry) vs the oracle."""
    rng = np.random.default_rng(91)
    n = 1_000_000
    k = rng.integers(-2**62, 2**62, n).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.groupby("k").sum().to_pandas()
    # oracle via pandas-free numpy on huge range: sort-based
    order = np.argsort(k, kind="stable")
    ks, vs = k[order], v[order]
    uk, starts = np.unique(ks, return_index=True)
    sums = np.add.reduceat(vs, starts)
    np.testing.assert_array_equal(out.index.to_numpy(), uk)
    np.testing.assert_allclose(out["v"].to_numpy(), sums, rtol=RTOL,
                               atol=1e-9)


@pytest.fixture
def force_sorted_groupby(monkeypatch):
    """Route EVERY groupby through the sort-based segmented path: shrink the
    dense cap to 0 and make the hash entry delegate straight to
    ``_groupby_sorted`` (the production unbounded-cardinality fallback)."""
    from modin_amd import config
    from modin_amd.core.partition_manager import HipDataframePartitionManager

    monkeypatch.setattr(config.MaxGroupbySlots, "_value", 0)
    monkeypatch.setattr(
        HipDataframePartitionManager, "_groupby_hash",
        classmethod(lambda cls, key_cols, vcpp, total_rows, want_counts,
                    agg_op: cls._groupby_sorted(key_cols, vcpp, want_counts,
                                                agg_op)))


@pytest.mark.parametrize("case", golden_cases("gb_") + golden_cases("hh_"))
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_sorted_groupby_vs_golden(case, agg, npartitions,
                                  force_sorted_groupby):
    """The sort-based general groupby (unbounded cardinality) must match the
    same golden fixtures as the dense/hash paths — every key shape, NaNs,
    multi-partition concat included."""
    g = load_golden(case)
    df = mpd.DataFrame(_gb_inputs(g))
    out = getattr(df.groupby("k"), agg)().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g[f"out_{agg}_keys"])
    for name in out.columns:
        expect = g[f"out_{agg}_{name}"]
        if agg == "count":
            np.testing.assert_array_equal(out[name].to_numpy(),
                                          expect.astype(np.int64))
        else:
            np.testing.assert_allclose(out[name].to_numpy(), expect,
                                       rtol=RTOL, atol=1e-9, equal_nan=True)


def test_sorted_groupby_property_medium(force_sorted_groupby):
    """2M rows over the full int64 span through sort+segagg, vs a numpy
    sort-based oracle; exercises multi-tile runs and the wide radix sort."""
    rng = np.random.default_rng(93)
    n = 2_000_000
    k = rng.integers(-2**62, 2**62, n).astype(np.int64)
    k[rng.random(n) < 0.3] = 42  # one giant run spanning many tiles
    v = rng.random(n)
    v[rng.random(n) < 0.05] = np.nan
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.groupby("k").sum().to_pandas()
    order = np.argsort(k, kind="stable")
    ks, vs = k[order], np.nan_to_num(v[order])
    uk, starts = np.unique(ks, return_index=True)
    sums = np.add.reduceat(vs, starts)
    np.testing.assert_array_equal(out.index.to_numpy(), uk)
    np.testing.assert_allclose(out["v"].to_numpy(), sums, rtol=RTOL,
                               atol=1e-9)
    outm = df.groupby("k").min().to_pandas()
    mins = np.minimum.reduceat(np.where(np.isnan(v[order]), np.inf, v[order]),
                               starts)
    has = np.logical_or.reduceat(~np.isnan(v[order]), starts)
    mins = np.where(has, mins, np.nan)
    np.testing.assert_allclose(outm["v"].to_numpy(), mins, rtol=0,
                               equal_nan=True)


def test_sort_huge_range(npartitions):
    """sort_values across a 2^61 key span (wide radix path)."""
    rng = np.random.default_rng(92)
    n = 300_000
    k = rng.integers(-2**60, 2**60, n).astype(np.int64)
    v = rng.random(n)
    df = mpd.DataFrame({"k": k, "v": v})
    out = df.sort_values("k").to_pandas()
    perm = oracle.sort_perm(k)
    np.testing.assert_array_equal(out.index.to_numpy(), perm)
    np.testing.assert_array_equal(out["k"].to_numpy(), k[perm])
    np.testing.assert_array_equal(out["v"].to_numpy(), v[perm])
    outd = df.sort_values("k", ascending=False).to_pandas()
    permd = oracle.sort_perm(k, ascending=False)
    np.testing.assert_array_equal(outd.index.to_numpy(), permd)


def test_multi_column_sort_vs_golden(npartitions):
    """sort_values by a key LIST (stable LSD composition of radix passes),
    mixed per-key ascending, vs the reference."""
    g = load_golden("srt_multi")
    df = mpd.DataFrame({"a": g["in_a"], "b": g["in_b"], "v": g["in_v"]})
    for tag, by, asc in [("ab", ["a", "b"], True),
                         ("ab_desc", ["a", "b"], False),
                         ("ab_mixed", ["a", "b"], [True, False]),
                         ("ba", ["b", "a"], True)]:
        out = df.sort_values(by, ascending=asc).to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{tag}_idx"], err_msg=tag)
        np.testing.assert_array_equal(out["a"].to_numpy(),
                                      g[f"out_{tag}_a"], err_msg=tag)
        np.testing.assert_array_equal(out["b"].to_numpy(),
                                      g[f"out_{tag}_b"], err_msg=tag)


def test_int_groupby_dtype_preserved_vs_golden(npartitions):
    """pandas dtype parity: sum/min/max of an int64 value column stay
    int64, mean float64, count int64 — values AND dtypes vs the
    reference."""
    g = load_golden("gbi_intvals")
    df = mpd.DataFrame({"k": g["in_k"], "iv": g["in_iv"],
                        "fv": g["in_fv"]})
    for agg in ("sum", "count", "mean", "min", "max"):
        out = getattr(df.groupby("k"), agg)().to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{agg}_keys"])
        for cn in ("iv", "fv"):
            expect = g[f"out_{agg}_{cn}"]
            assert out[cn].dtype == expect.dtype, \
                f"{agg}/{cn}: {out[cn].dtype} != {expect.dtype}"
            if expect.dtype == np.int64:
                np.testing.assert_array_equal(out[cn].to_numpy(), expect,
                                              err_msg=f"{agg}/{cn}")
            else:
                np.testing.assert_allclose(out[cn].to_numpy(), expect,
                                           rtol=RTOL, atol=1e-9,
                                           equal_nan=True,
                                           err_msg=f"{agg}/{cn}")
    # beyond-2^53 guard is loud, not silently wrong
    big = mpd.DataFrame({"k": np.zeros(4, dtype=np.int64),
                         "iv": np.full(4, 1 << 60, dtype=np.int64)})
    with pytest.raises(lib.HfError, match="2\\^53"):
        big.groupby("k").sum().to_pandas()


def test_var_std_size_vs_golden(npartitions):
    """frame/groupby var+std (ddof 0 and 1; single-row and all-NaN groups
    -> NaN) and groupby size, vs the reference."""
    g = load_golden("gbv_moments")
    df = mpd.DataFrame({"k": g["in_k"], "v": g["in_v"], "w": g["in_w"]})
    gb = df.groupby("k")
    for ddof in (0, 1):
        for name, res in [(f"var{ddof}", gb.var(ddof=ddof)),
                          (f"std{ddof}", gb.std(ddof=ddof))]:
            out = res.to_pandas()
            np.testing.assert_array_equal(out.index.to_numpy(),
                                          g[f"out_{name}_keys"])
            for cn in ("v", "w"):
                np.testing.assert_allclose(
                    out[cn].to_numpy(), g[f"out_{name}_{cn}"],
                    rtol=1e-9, atol=1e-12, equal_nan=True,
                    err_msg=f"{name}/{cn}")
        # frame-level (golden order is [k, v, w]; we take v, w)
        fv = df[["v", "w"]].var(ddof=ddof)
        fs = df[["v", "w"]].std(ddof=ddof)
        np.testing.assert_allclose(np.asarray(fv),
                                   g[f"out_frame_var{ddof}"][1:],
                                   rtol=1e-9, equal_nan=True)
        np.testing.assert_allclose(np.asarray(fs),
                                   g[f"out_frame_std{ddof}"][1:],
                                   rtol=1e-9, equal_nan=True)
    sz = df.groupby("k").size()
    np.testing.assert_array_equal(np.asarray(sz.index), g["out_size_keys"])
    np.testing.assert_array_equal(np.asarray(sz), g["out_size"])


def test_agg_list_dict_forms_vs_golden(npartitions):
    """groupby.agg(['sum','mean']) -> MultiIndex columns in pandas
    col-major order; agg({'v':'sum','w':'max'}) -> per-column aggs."""
    g = load_golden("gba_forms")
    df = mpd.DataFrame({"k": g["in_k"], "v": g["in_v"], "w": g["in_w"]})
    out = df.groupby("k").agg(["sum", "mean"]).to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_list_keys"])
    got_cols = [f"{c}|{a}" for c, a in out.columns]
    assert got_cols == list(g["out_list_cols"]), got_cols
    np.testing.assert_allclose(out.to_numpy(), g["out_list_vals"],
                               rtol=RTOL, atol=1e-9, equal_nan=True)
    out2 = df.groupby("k").agg({"v": "sum", "w": "max"}).to_pandas()
    np.testing.assert_array_equal(out2.index.to_numpy(), g["out_dict_keys"])
    assert list(out2.columns) == list(g["out_dict_cols"])
    np.testing.assert_allclose(out2.to_numpy(), g["out_dict_vals"],
                               rtol=RTOL, atol=1e-9, equal_nan=True)


def test_multikey_groupby_vs_golden(npartitions):
    """groupby(by=[a, b]): the combined-key fold through the single-key
    router; MultiIndex result in pandas lexicographic order, size
    included."""
    g = load_golden("gbm_ints")
    df = mpd.DataFrame({"a": g["in_a"], "b": g["in_b"], "v": g["in_v"],
                        "w": g["in_w"]})
    for agg in ("sum", "count", "mean", "min", "max"):
        out = getattr(df.groupby(["a", "b"]), agg)().to_pandas()
        np.testing.assert_array_equal(
            out.index.get_level_values(0).to_numpy(), g[f"out_{agg}_ka"],
            err_msg=f"{agg} ka")
        np.testing.assert_array_equal(
            out.index.get_level_values(1).to_numpy(), g[f"out_{agg}_kb"],
            err_msg=f"{agg} kb")
        assert out.index.names == ["a", "b"]
        for cn in ("v", "w"):
            expect = g[f"out_{agg}_{cn}"]
            np.testing.assert_allclose(out[cn].to_numpy(), expect,
                                       rtol=RTOL, atol=1e-9,
                                       equal_nan=True,
                                       err_msg=f"{agg}/{cn}")
    sz = df.groupby(["a", "b"]).size()
    np.testing.assert_array_equal(np.asarray(sz), g["out_size"])


def test_multikey_groupby_string_int_vs_golden(npartitions):
    """groupby(by=[string, int]): dictionary codes fold with the int key;
    NaN string keys drop the whole row (pandas dropna)."""
    from tests.test_gpu_strings import assert_str_equal, dec
    g = load_golden("gbm_strint")
    df = mpd.DataFrame({"s": dec(g["in_s"]), "g": g["in_g"],
                        "v": g["in_v"]})
    for agg in ("sum", "mean"):
        out = getattr(df.groupby(["s", "g"]), agg)().to_pandas()
        assert_str_equal(out.index.get_level_values(0).to_numpy(),
                         g[f"out_{agg}_ks"], f"{agg} ks")
        np.testing.assert_array_equal(
            out.index.get_level_values(1).to_numpy(), g[f"out_{agg}_kg"])
        np.testing.assert_allclose(out["v"].to_numpy(),
                                   g[f"out_{agg}_v"], rtol=RTOL,
                                   atol=1e-9, equal_nan=True)


def test_pipeline_filter_merge_groupby_sort(npartitions):
    """Integration chain: filter -> merge -> groupby -> sort, checked against
    the same chain on pandas."""
    rng = np.random.default_rng(60)
    n = 120_000
    left = {"k": rng.integers(0, 300, n).astype(np.int64),
            "v": rng.random(n)}
    right = {"k": rng.integers(0, 300, 700).astype(np.int64),
             "w": rng.random(700)}
    df = mpd.DataFrame(left)
    rf = mpd.DataFrame(right)
    out = (df[df["v"] > 0.25]
           .merge(rf, on="k")
           .groupby("k").sum()
           .to_pandas())
    pdf = pandas.DataFrame(left)
    prf = pandas.DataFrame(right)
    expect = (pdf[pdf["v"] > 0.25]
              .merge(prf, on="k")
              .groupby("k").sum())
    np.testing.assert_array_equal(out.index.to_numpy(),
                                  expect.index.to_numpy())
    for c in ("v", "w"):
        np.testing.assert_allclose(out[c].to_numpy(), expect[c].to_numpy(),
                                   rtol=RTOL, atol=1e-9)
    # and a sort on top of a merge result
    m = df.merge(rf, on="k").sort_values("k").to_pandas()
    pm = pdf.merge(prf, on="k").sort_values("k", kind="stable")
    np.testing.assert_array_equal(m["k"].to_numpy(), pm["k"].to_numpy())
    np.testing.assert_array_equal(m["w"].to_numpy(), pm["w"].to_numpy())


def test_concat_vs_pandas(npartitions):
    rng = np.random.default_rng(70)
    a = {"k": rng.integers(0, 9, 1000).astype(np.int64), "v": rng.random(1000)}
    b = {"k": rng.integers(0, 9, 500).astype(np.int64), "v": rng.random(500)}
    out = mpd.concat([mpd.DataFrame(a), mpd.DataFrame(b)]).to_pandas()
    expect = pandas.concat([pandas.DataFrame(a), pandas.DataFrame(b)])
    pandas.testing.assert_frame_equal(out, expect)
    out2 = mpd.concat([mpd.DataFrame(a), mpd.DataFrame(b)],
                      ignore_index=True)
    expect2 = pandas.concat([pandas.DataFrame(a), pandas.DataFrame(b)],
                            ignore_index=True)
    pandas.testing.assert_frame_equal(out2.to_pandas(), expect2)
    # concat result feeds the groupby path
    g = out2.groupby("k").sum().to_pandas()
    ge = expect2.groupby("k").sum()
    np.testing.assert_allclose(g["v"].to_numpy(), ge["v"].to_numpy(),
                               rtol=RTOL)


def test_head_tail_astype_rename_reset(npartitions):
    rng = np.random.default_rng(80)
    data = {"a": rng.integers(0, 50, 2000).astype(np.int64),
            "b": rng.random(2000) * 10}
    df = mpd.DataFrame(data)
    pdf = pandas.DataFrame(data)
    pandas.testing.assert_frame_equal(df.head(7).to_pandas(), pdf.head(7))
    pandas.testing.assert_frame_equal(df.tail(9).to_pandas(), pdf.tail(9))
    pandas.testing.assert_frame_equal(df.head(-3).to_pandas(),
                                      pdf.head(-3))
    pandas.testing.assert_frame_equal(df.tail(-3).to_pandas(),
                                      pdf.tail(-3))
    pandas.testing.assert_frame_equal(df.astype("float64").to_pandas(),
                                      pdf.astype("float64"))
    pandas.testing.assert_frame_equal(df.astype("int64").to_pandas(),
                                      pdf.astype("int64"))
    pandas.testing.assert_frame_equal(
        df.rename(columns={"a": "x"}).to_pandas(),
        pdf.rename(columns={"a": "x"}))
    filt = df[df["b"] > 5.0]
    pfilt = pdf[pdf["b"] > 5.0]
    pandas.testing.assert_frame_equal(filt.reset_index(drop=True).to_pandas(),
                                      pfilt.reset_index(drop=True))
    # renamed frame still computes
    out = df.rename(columns={"a": "k"}).groupby("k").sum().to_pandas()
    expect = pdf.rename(columns={"a": "k"}).groupby("k").sum()
    np.testing.assert_allclose(out["b"].to_numpy(), expect["b"].to_numpy(),
                               rtol=RTOL)


def test_groupby_keys_only(npartitions):
    """groupby on a frame whose only column is the key: result is an empty
    column set with the group index (pandas shape)."""
    rng = np.random.default_rng(90)
    k = rng.integers(0, 40, 5000).astype(np.int64)
    df = mpd.DataFrame({"k": k})
    out = df.groupby("k").sum().to_pandas()
    expect = pandas.DataFrame({"k": k}).groupby("k").sum()
    assert list(out.columns) == list(expect.columns) == []
    np.testing.assert_array_equal(out.index.to_numpy(),
                                  expect.index.to_numpy())


def test_native_extension_is_loaded():
    """Guard against a silent eager/pandas fallback: the in-tree .so must be
    mapped into this process."""
    import modin_amd.core.lib as l
    maps = open("/proc/self/maps").read()
    assert "libhipframe.so" in maps


def test_new_paths_empty_and_edge(npartitions):
    """Empty/degenerate frames through the round-1 late additions:
    multi-key groupby, var/std, size, unique/value_counts, agg forms."""
    empty = mpd.DataFrame(pandas.DataFrame({
        "a": np.array([], dtype=np.int64),
        "b": np.array([], dtype=np.int64),
        "v": np.array([], dtype=np.float64)}))
    out = empty.groupby(["a", "b"]).sum().to_pandas()
    assert len(out) == 0
    assert len(empty.groupby("a").var().to_pandas()) == 0
    assert empty.groupby("a").size().empty
    assert len(mpd.DataFrame(pandas.DataFrame(
        {"k": np.array([], dtype=np.int64)}))["k"].unique()) == 0
    # single row: var ddof=1 -> NaN, ddof=0 -> 0
    one = mpd.DataFrame(pandas.DataFrame({"k": [1], "v": [2.5]}))
    v1 = one.groupby("k").var().to_pandas()
    assert np.isnan(v1["v"].iloc[0])
    v0 = one.groupby("k").var(ddof=0).to_pandas()
    assert v0["v"].iloc[0] == 0.0
    # multi-key with one key column degenerates to single-key
    pdf = pandas.DataFrame({"a": [1, 1, 2], "v": [1.0, 2.0, 3.0]})
    df = mpd.DataFrame(pdf)
    out = df.groupby(["a"]).sum().to_pandas()
    exp = pdf.groupby(["a"]).sum()
    np.testing.assert_array_equal(out["v"].to_numpy(),
                                  exp["v"].to_numpy())
    # combined-range beyond 2^62: the sorted-heads dense-rank fold now
    # covers it (round 2) — must match pandas instead of raising
    big_p = pandas.DataFrame({
        "a": np.array([0, 2**40, 0, 2**40], dtype=np.int64),
        "b": np.array([0, 2**40, 0, 5], dtype=np.int64),
        "v": np.array([1.0, 2.0, 3.0, 4.0])})
    big = mpd.DataFrame(big_p)
    got_big = big.groupby(["a", "b"]).sum().to_pandas()
    exp_big = big_p.groupby(["a", "b"]).sum()
    assert list(got_big.index) == list(exp_big.index)
    np.testing.assert_allclose(got_big["v"].to_numpy(),
                               exp_big["v"].to_numpy(), rtol=0)


def test_merge_unbounded_span_vs_golden(npartitions):
    """Merge keys over the full int64 span: the densify path (sorted
    distinct right keys + device binary search -> CSR join in code
    space) vs the reference."""
    g = load_golden("mg2_hugespan")
    left = mpd.DataFrame({"k": g["in_lk"], "a": g["in_la"]})
    right = mpd.DataFrame({"k": g["in_rk"], "b": g["in_rb"]})
    out = left.merge(right, on="k").to_pandas()
    np.testing.assert_array_equal(out["k"].to_numpy(), g["out_k"])
    np.testing.assert_array_equal(out["a"].to_numpy(), g["out_a"])
    np.testing.assert_array_equal(out["b"].to_numpy(), g["out_b"])
    # repeat merges hit the cached densified build
    out2 = left.merge(right, on="k").to_pandas()
    np.testing.assert_array_equal(out2["k"].to_numpy(), g["out_k"])


def test_search_sorted_kernel():
    rng = np.random.default_rng(23)
    uniq = np.unique(rng.integers(-2**62, 2**62, 5000)).astype(np.int64)
    keys = np.concatenate([rng.choice(uniq, 20_000),
                           rng.integers(-2**62, 2**62, 5000)]).astype(np.int64)
    out = lib.get(lib.search_sorted(lib.put(keys), lib.put(uniq)))
    expect = np.searchsorted(uniq, keys)
    expect = np.where((expect < len(uniq)) & (uniq[np.minimum(expect,
                                                              len(uniq) - 1)]
                                              == keys), expect, -1)
    np.testing.assert_array_equal(out, expect)
    # empty inputs
    assert lib.get(lib.search_sorted(lib.put(np.empty(0, dtype=np.int64)),
                                     lib.put(uniq))).size == 0


def test_groupby_selection_and_as_index(npartitions):
    """gb[col] / gb[[cols]] selection and as_index=False (reference
    reduce fix-up algebra/groupby.py:278) vs pandas inline."""
    rng = np.random.default_rng(81)
    n = 50_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 100, n),
                            "v": rng.random(n),
                            "w": rng.random(n)})
    df = mpd.DataFrame(pdf)
    s1 = df.groupby("k")["v"].sum()
    p1 = pdf.groupby("k")["v"].sum()
    assert s1.name == "v"
    out = s1.to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), p1.index.to_numpy())
    np.testing.assert_allclose(out.to_numpy(), p1.to_numpy(), rtol=RTOL)
    d2 = df.groupby("k")[["w", "v"]].mean().to_pandas()
    p2 = pdf.groupby("k")[["w", "v"]].mean()
    assert list(d2.columns) == list(p2.columns)
    np.testing.assert_allclose(d2.to_numpy(), p2.to_numpy(), rtol=RTOL)
    d3 = df.groupby("k", as_index=False).sum().to_pandas()
    p3 = pdf.groupby("k", as_index=False).sum()
    assert list(d3.columns) == list(p3.columns)
    np.testing.assert_array_equal(d3["k"].to_numpy(), p3["k"].to_numpy())
    np.testing.assert_allclose(d3["v"].to_numpy(), p3["v"].to_numpy(),
                               rtol=RTOL)
    d4 = df.groupby(["k"], as_index=False)["v"].sum()
    p4 = pdf.groupby(["k"], as_index=False)["v"].sum()
    got4 = d4.to_pandas() if hasattr(d4, "to_pandas") else d4
    np.testing.assert_allclose(got4["v"].to_numpy(), p4["v"].to_numpy(),
                               rtol=RTOL)


def test_float_keys_vs_golden(npartitions):
    """Float sort/groupby/unique keys via the ordered f64<->i64 bit
    transform: NaN-last sorts both directions, NaN-dropped groups,
    -0.0 == +0.0, exact key decode."""
    g = load_golden("flt_keys")
    df = mpd.DataFrame({"f": g["in_f"], "w": g["in_w"], "v": g["in_v"]})
    for tag, by, asc in [("f_asc", "f", True), ("f_desc", "f", False),
                         ("fw", ["f", "w"], True),
                         ("wf_mixed", ["w", "f"], [True, False])]:
        out = df.sort_values(by, ascending=asc).to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{tag}_idx"], err_msg=tag)
        np.testing.assert_array_equal(out["f"].to_numpy(),
                                      g[f"out_{tag}_f"], err_msg=tag)
    for agg in ("sum", "mean", "count"):
        out = getattr(df.groupby("f"), agg)().to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_gb_{agg}_keys"],
                                      err_msg=f"gb {agg} keys")
        for cn in ("w", "v"):
            expect = g[f"out_gb_{agg}_{cn}"]
            np.testing.assert_allclose(out[cn].to_numpy(), expect,
                                       rtol=RTOL, atol=1e-9,
                                       equal_nan=True,
                                       err_msg=f"gb {agg}/{cn}")
    u = df["f"].unique()
    expu = g["out_unique"]
    assert len(u) == len(expu)
    np.testing.assert_array_equal(np.asarray(u, dtype=np.float64),
                                  expu.astype(np.float64))
    vc = df["f"].value_counts()
    np.testing.assert_array_equal(vc.index.to_numpy().astype(np.float64),
                                  g["out_vc_idx"].astype(np.float64))
    np.testing.assert_array_equal(vc.to_numpy(), g["out_vc"])
    assert df["f"].nunique() == int(g["out_nunique"][0])


def test_median_vs_golden(npartitions):
    """df.median and groupby.median (int/string/multi keys, NaN values,
    even/odd group sizes) vs the reference."""
    from tests.test_gpu_strings import assert_str_equal, dec
    g = load_golden("med_cases")
    df = mpd.DataFrame({"k": g["in_k"], "s": dec(g["in_s"]),
                        "v": g["in_v"], "w": g["in_w"]})
    fm = df[["k", "v", "w"]].median()
    np.testing.assert_allclose(np.asarray(fm), g["out_frame_median"],
                               rtol=RTOL)
    out = df[["k", "v", "w"]].groupby("k").median().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_gbk_keys"])
    for cn in ("v", "w"):
        np.testing.assert_allclose(out[cn].to_numpy(),
                                   g[f"out_gbk_{cn}"], rtol=RTOL,
                                   atol=1e-12, equal_nan=True,
                                   err_msg=f"gbk/{cn}")
    out = df[["s", "v", "w"]].groupby("s").median().to_pandas()
    assert_str_equal(out.index.to_numpy(), g["out_gbs_keys"], "gbs keys")
    for cn in ("v", "w"):
        np.testing.assert_allclose(out[cn].to_numpy(),
                                   g[f"out_gbs_{cn}"], rtol=RTOL,
                                   atol=1e-12, equal_nan=True)
    out = df.groupby(["k", "s"]).median().to_pandas()
    np.testing.assert_array_equal(
        out.index.get_level_values(0).to_numpy(), g["out_gbks_ka"])
    assert_str_equal(out.index.get_level_values(1).to_numpy(),
                     g["out_gbks_kb"], "gbks kb")
    for cn in ("v", "w"):
        np.testing.assert_allclose(out[cn].to_numpy(),
                                   g[f"out_gbks_{cn}"], rtol=RTOL,
                                   atol=1e-12, equal_nan=True)


def test_series_mask_and_dropna(npartitions):
    rng = np.random.default_rng(86)
    v = rng.standard_normal(10_000)
    v[rng.random(10_000) < 0.1] = np.nan
    ps = pandas.Series(v, name="v")
    df = mpd.DataFrame({"v": v})
    s_ = df["v"]
    got = s_[s_ > 0.5].to_pandas()
    exp = ps[ps > 0.5]
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    gd = s_.dropna().to_pandas()
    ed = ps.dropna()
    np.testing.assert_array_equal(gd.index.to_numpy(), ed.index.to_numpy())
    np.testing.assert_array_equal(gd.to_numpy(), ed.to_numpy())


def test_left_merge_vs_golden(npartitions):
    """merge(how='left'): unmatched lefts keep NaN rights, int right
    columns become float64 iff NaNs were introduced, pandas left row
    order; huge-span (densify) keys too."""
    g = load_golden("mgl_basic")
    left = mpd.DataFrame({"k": g["in_lk"], "a": g["in_la"]})
    right = mpd.DataFrame({"k": g["in_rk"], "b": g["in_rb"],
                           "i": g["in_ri"]})
    out = left.merge(right, on="k", how="left").to_pandas()
    np.testing.assert_array_equal(out["k"].to_numpy(), g["out_k"])
    np.testing.assert_allclose(out["a"].to_numpy(), g["out_a"], rtol=0)
    np.testing.assert_allclose(out["b"].to_numpy(), g["out_b"], rtol=0,
                               equal_nan=True)
    assert out["i"].dtype == np.float64  # NaNs introduced
    np.testing.assert_allclose(out["i"].to_numpy(), g["out_i"], rtol=0,
                               equal_nan=True)

    g2 = load_golden("mgl_allmatch")
    l2 = mpd.DataFrame({"k": g2["in_lk"],
                        "a": np.zeros(len(g2["in_lk"]))})
    r2 = mpd.DataFrame({"k": g2["in_rk"], "i": g2["in_ri"]})
    o2 = l2.merge(r2, on="k", how="left").to_pandas()
    assert o2["i"].dtype == np.int64  # all matched: int stays int
    np.testing.assert_array_equal(o2["k"].to_numpy(), g2["out_k"])
    np.testing.assert_array_equal(o2["i"].to_numpy(), g2["out_i"])

    g4 = load_golden("mgl_huge")
    l4 = mpd.DataFrame({"k": g4["in_lk"],
                        "a": np.zeros(len(g4["in_lk"]))})
    r4 = mpd.DataFrame({"k": g4["in_rk"], "b": g4["in_rb"]})
    o4 = l4.merge(r4, on="k", how="left").to_pandas()
    np.testing.assert_array_equal(o4["k"].to_numpy(), g4["out_k"])
    np.testing.assert_allclose(o4["b"].to_numpy(), g4["out_b"], rtol=0,
                               equal_nan=True)


def test_quantile_describe_vs_pandas(npartitions):
    """df.quantile (scalar and list q, linear interpolation, NaN skipped)
    and df.describe vs pandas inline (pandas is the reference's own
    arbiter — SURVEY §8c)."""
    rng = np.random.default_rng(87)
    n = 30_000
    v = rng.standard_normal(n) * 7
    v[rng.random(n) < 0.1] = np.nan
    w = rng.integers(-1000, 1000, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for q in (0.5, 0.25, 0.99, 0.0, 1.0):
        got = df.quantile(q)
        exp = pdf.quantile(q)
        np.testing.assert_allclose(np.asarray(got), exp.to_numpy(),
                                   rtol=RTOL, err_msg=str(q))
    got = df.quantile([0.1, 0.5, 0.9])
    exp = pdf.quantile([0.1, 0.5, 0.9])
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=RTOL)
    gd = df.describe()
    ed = pdf.describe()
    assert list(gd.index) == list(ed.index)
    assert list(gd.columns) == list(ed.columns)
    np.testing.assert_allclose(gd.to_numpy(), ed.to_numpy(), rtol=1e-9)


def test_outer_right_merge_vs_golden(npartitions):
    """merge how='outer' (key-sorted union, NaN fills both directions)
    and how='right' (right row order, suffix roles preserved) vs the
    reference."""
    for how in ("outer", "right"):
        g = load_golden(f"mgo_{how}")
        left = mpd.DataFrame({"k": g["in_lk"], "a": g["in_la"],
                              "w": g["in_lw"]})
        right = mpd.DataFrame({"k": g["in_rk"], "b": g["in_rb"],
                               "w": g["in_rw"]})
        out = left.merge(right, on="k", how=how).to_pandas()
        assert list(out.columns) == list(g["out_cols"]), how
        np.testing.assert_array_equal(out["k"].to_numpy(), g["out_k"],
                                      err_msg=how)
        for cn, gk in [("a", "out_a"), ("w_x", "out_wx"),
                       ("b", "out_b"), ("w_y", "out_wy")]:
            np.testing.assert_allclose(out[cn].to_numpy(), g[gk],
                                       rtol=0, equal_nan=True,
                                       err_msg=f"{how}/{cn}")


def test_shift_diff_idx_vs_pandas(npartitions):
    rng = np.random.default_rng(88)
    n = 20_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-100, 100, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for p_ in (1, 3, -2, 0, n + 5):
        got = df.shift(p_).to_pandas()
        exp = pdf.shift(p_)
        np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(),
                                   rtol=0, equal_nan=True,
                                   err_msg=f"shift {p_}")
        gd = df.diff(p_).to_pandas()
        ed = pdf.diff(p_)
        np.testing.assert_allclose(gd.to_numpy(), ed.to_numpy(),
                                   rtol=0, equal_nan=True,
                                   err_msg=f"diff {p_}")
    im = df.idxmax()
    em = pdf.idxmax()
    np.testing.assert_array_equal(np.asarray(im), em.to_numpy())
    ii = df.idxmin()
    ei = pdf.idxmin()
    np.testing.assert_array_equal(np.asarray(ii), ei.to_numpy())
    sv = df["v"]
    assert sv.idxmax() == pdf["v"].idxmax()
    assert sv.idxmin() == pdf["v"].idxmin()


def test_cumsum_vs_pandas(npartitions):
    """Device three-phase prefix scan: i64 exact, f64 NaN-skipping, at a
    multi-tile size."""
    rng = np.random.default_rng(89)
    n = 1_000_000  # ~245 tiles
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-100, 100, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    got = df.cumsum().to_pandas()
    exp = pdf.cumsum()
    assert got["w"].dtype == np.int64
    np.testing.assert_array_equal(got["w"].to_numpy(), exp["w"].to_numpy())
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=1e-12, atol=1e-9, equal_nan=True)
    s_ = df["v"].cumsum().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(), pdf["v"].cumsum().to_numpy(),
                               rtol=1e-12, atol=1e-9, equal_nan=True)
    for name, fn in (("cummax", "cummax"), ("cummin", "cummin")):
        got2 = getattr(df, fn)().to_pandas()
        exp2 = getattr(pdf, fn)()
        assert got2["w"].dtype == np.int64
        np.testing.assert_array_equal(got2["w"].to_numpy(),
                                      exp2["w"].to_numpy(), err_msg=name)
        np.testing.assert_allclose(got2["v"].to_numpy(),
                                   exp2["v"].to_numpy(), rtol=0,
                                   equal_nan=True, err_msg=name)


def test_groupby_quantile_vs_pandas(npartitions):
    rng = np.random.default_rng(91)
    n = 40_000
    k = rng.integers(0, 60, n)
    v = rng.standard_normal(n) * 5
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"k": k, "v": v})
    df = mpd.DataFrame(pdf)
    for q in (0.5, 0.25, 0.9, 0.0, 1.0):
        got = df.groupby("k").quantile(q).to_pandas()
        exp = pdf.groupby("k").quantile(q)
        np.testing.assert_array_equal(got.index.to_numpy(),
                                      exp.index.to_numpy())
        np.testing.assert_allclose(got["v"].to_numpy(),
                                   exp["v"].to_numpy(), rtol=1e-12,
                                   atol=1e-12, equal_nan=True,
                                   err_msg=str(q))


def test_clip_axis1_vs_pandas(npartitions):
    """clip bounds and axis=1 reductions (sum/mean/min/max/count row
    folds, NaN-skipping) vs pandas."""
    rng = np.random.default_rng(92)
    n = 30_000
    a = rng.standard_normal(n) * 10
    a[rng.random(n) < 0.1] = np.nan
    b = rng.standard_normal(n) * 10
    b[rng.random(n) < 0.1] = np.nan
    w = rng.integers(-50, 50, n)
    pdf = pandas.DataFrame({"a": a, "b": b, "w": w})
    df = mpd.DataFrame(pdf)
    got = df.clip(-5, 5).to_pandas()
    exp = pdf.clip(-5, 5)
    assert list(got.dtypes) == list(exp.dtypes)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    got = df.clip(lower=0).to_pandas()
    exp = pdf.clip(lower=0)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    for op in ("sum", "mean", "min", "max", "count"):
        g = getattr(df, op)(axis=1).to_pandas()
        e = getattr(pdf, op)(axis=1)
        np.testing.assert_allclose(g.to_numpy().astype(float),
                                   e.to_numpy().astype(float), rtol=1e-12,
                                   atol=1e-12, equal_nan=True,
                                   err_msg=op)
    # all-int frame keeps int64 sums/extremes
    pint = pandas.DataFrame({"x": w, "y": rng.integers(0, 9, n)})
    dint = mpd.DataFrame(pint)
    gs = dint.sum(axis=1).to_pandas()
    assert gs.dtype == np.int64
    np.testing.assert_array_equal(gs.to_numpy(),
                                  pint.sum(axis=1).to_numpy())


def test_groupby_transforms_vs_pandas(npartitions):
    """groupby cumsum/cummin/cummax/cumcount in ORIGINAL row order
    (segmented scan after a stable key sort + inverse-permutation
    scatter), incl. NaN keys (no group -> NaN, pandas dropna=True), NaN
    values (stay NaN, don't advance the state) and the dtype rule (int64
    survives only when every key is valid)."""
    rng = np.random.default_rng(93)
    n = 200_000
    k = rng.integers(0, 500, n).astype(np.float64)
    k[rng.random(n) < 0.02] = np.nan
    v = rng.standard_normal(n) * 3
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-50, 50, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for how in ("cumsum", "cummin", "cummax"):
        got = getattr(df.groupby("k"), how)().to_pandas()
        exp = getattr(pdf.groupby("k"), how)()
        assert list(got.columns) == list(exp.columns)
        for c in exp.columns:
            np.testing.assert_allclose(
                got[c].to_numpy(), exp[c].to_numpy(),
                rtol=1e-12 if how == "cumsum" else 0, atol=1e-9,
                equal_nan=True, err_msg=f"{how}/{c}")
    got_cc = df.groupby("k").cumcount().to_pandas()
    exp_cc = pdf.groupby("k").cumcount()
    np.testing.assert_allclose(got_cc.to_numpy(), exp_cc.to_numpy(),
                               rtol=0, equal_nan=True)
    # int keys, no NaN anywhere: int64 results, exact
    pdf2 = pandas.DataFrame({"k": rng.integers(0, 50, 5000),
                             "w": rng.integers(-9, 9, 5000)})
    df2 = mpd.DataFrame(pdf2)
    for how in ("cumsum", "cummin", "cummax"):
        got2 = getattr(df2.groupby("k"), how)().to_pandas()
        exp2 = getattr(pdf2.groupby("k"), how)()
        assert got2["w"].dtype == np.int64, how
        np.testing.assert_array_equal(got2["w"].to_numpy(),
                                      exp2["w"].to_numpy(), err_msg=how)
    got2 = df2.groupby("k").cumcount().to_pandas()
    assert got2.dtype == np.int64
    np.testing.assert_array_equal(got2.to_numpy(),
                                  pdf2.groupby("k").cumcount().to_numpy())
    # Series selection form
    s_ = df2.groupby("k")["w"].cumsum().to_pandas()
    pd_s = pdf2.groupby("k")["w"].cumsum()
    assert s_.name == "w"
    np.testing.assert_array_equal(s_.to_numpy(), pd_s.to_numpy())


def test_groupby_rank_vs_pandas(npartitions):
    """groupby.rank: methods average/min/first, ascending both ways,
    na_option='keep' (NaN values rank NaN; ties averaged within the
    group)."""
    rng = np.random.default_rng(94)
    n = 60_000
    k = rng.integers(0, 200, n)
    v = rng.integers(-20, 20, n).astype(np.float64)  # many exact ties
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-5, 5, n)  # int64 values with heavy ties
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for method in ("average", "min", "first"):
        for asc in (True, False):
            got = df.groupby("k").rank(method=method,
                                       ascending=asc).to_pandas()
            exp = pdf.groupby("k").rank(method=method, ascending=asc)
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=0,
                    equal_nan=True, err_msg=f"{method}/asc={asc}/{c}")
    s_ = df.groupby("k")["v"].rank().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(),
                               pdf.groupby("k")["v"].rank().to_numpy(),
                               rtol=0, equal_nan=True)


def test_groupby_transform_multikey_vs_pandas(npartitions):
    """Multi-key transforms (string + int keys): group identity via the
    per-column run-head OR — no combined-key fold, so any span works."""
    rng = np.random.default_rng(95)
    n = 30_000
    a = rng.choice(["x", "y", "zz", "w"], n)
    b = rng.integers(-3, 4, n)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    df = mpd.DataFrame(pdf)
    got = df.groupby(["a", "b"]).cumsum().to_pandas()
    exp = pdf.groupby(["a", "b"]).cumsum()
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=1e-12, atol=1e-9, equal_nan=True)
    got_cc = df.groupby(["a", "b"]).cumcount().to_pandas()
    np.testing.assert_array_equal(
        got_cc.to_numpy(), pdf.groupby(["a", "b"]).cumcount().to_numpy())
    got_r = df.groupby(["a", "b"]).rank().to_pandas()
    exp_r = pdf.groupby(["a", "b"]).rank()
    np.testing.assert_allclose(got_r["v"].to_numpy(),
                               exp_r["v"].to_numpy(), rtol=0,
                               equal_nan=True)


def test_groupby_transform_edge_cases(npartitions):
    """Single group, all-NaN keys, one row, empty value set."""
    pdf = pandas.DataFrame({"k": [7, 7, 7], "v": [1.0, np.nan, 2.0]})
    df = mpd.DataFrame(pdf)
    np.testing.assert_allclose(
        df.groupby("k").cumsum().to_pandas()["v"].to_numpy(),
        pdf.groupby("k").cumsum()["v"].to_numpy(), rtol=0, equal_nan=True)
    pdf2 = pandas.DataFrame({"k": [np.nan, np.nan], "v": [1.0, 2.0]})
    df2 = mpd.DataFrame(pdf2)
    np.testing.assert_allclose(
        df2.groupby("k").cumsum().to_pandas()["v"].to_numpy(),
        pdf2.groupby("k").cumsum()["v"].to_numpy(), rtol=0, equal_nan=True)
    pdf3 = pandas.DataFrame({"k": [1], "v": [5.0]})
    df3 = mpd.DataFrame(pdf3)
    np.testing.assert_allclose(
        df3.groupby("k").rank().to_pandas()["v"].to_numpy(),
        pdf3.groupby("k").rank()["v"].to_numpy(), rtol=0)


def test_scatter_seg_cumsum_kernels():
    """Kernel-level checks: hf_scatter inverts hf_gather; hf_seg_cumsum
    equals a per-segment host scan (sum/min/max, i64 + f64 with NaN)."""
    rng = np.random.default_rng(96)
    n = 50_000
    perm_np = rng.permutation(n).astype(np.int64)
    x = rng.standard_normal(n)
    cx = lib.put(x)
    cp = lib.put(perm_np)
    gathered = lib.gather(cx, cp)
    back = lib.scatter(gathered, cp)
    np.testing.assert_array_equal(lib.get(back), x)
    # segmented scan vs host reference
    heads = (rng.random(n) < 0.001).astype(np.int64)
    heads[0] = 1
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    seg_id = np.cumsum(heads) - 1
    for op, name in ((lib.AGG_SUM, "sum"), (lib.AGG_MIN, "min"),
                     (lib.AGG_MAX, "max")):
        got = lib.get(lib.seg_cumsum(lib.put(v), lib.put(heads), op))
        exp = pandas.Series(v).groupby(seg_id).transform(
            {"sum": "cumsum", "min": "cummin", "max": "cummax"}[name]
        ).to_numpy()
        np.testing.assert_allclose(got, exp, rtol=1e-12, atol=1e-9,
                                   equal_nan=True, err_msg=name)
    w = rng.integers(-1000, 1000, n)
    got = lib.get(lib.seg_cumsum(lib.put(w), lib.put(heads), lib.AGG_SUM))
    exp = pandas.Series(w).groupby(seg_id).cumsum().to_numpy()
    np.testing.assert_array_equal(got, exp)


def test_sort_na_position_first(npartitions):
    """sort_values(na_position='first'): NaN float keys and NaN string
    (dict code −1) keys lead the result for both directions; stability
    preserved."""
    rng = np.random.default_rng(97)
    n = 30_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    s = rng.choice(["b", "a", "cc"], n).astype(object)
    s[rng.random(n) < 0.1] = None
    w = rng.integers(0, 100, n)
    pdf = pandas.DataFrame({"v": v, "s": s, "w": w})
    df = mpd.DataFrame(pdf)
    for by in ("v", "s"):
        for asc in (True, False):
            got = df.sort_values(by, ascending=asc,
                                 na_position="first").to_pandas()
            exp = pdf.sort_values(by, ascending=asc, kind="stable",
                                  na_position="first")
            np.testing.assert_array_equal(got.index.to_numpy(),
                                          exp.index.to_numpy(),
                                          err_msg=f"{by}/asc={asc}")
            np.testing.assert_array_equal(got["w"].to_numpy(),
                                          exp["w"].to_numpy())
    s2 = df["v"].sort_values(na_position="first").to_pandas()
    e2 = pdf["v"].sort_values(na_position="first", kind="stable")
    np.testing.assert_array_equal(s2.index.to_numpy(), e2.index.to_numpy())


def test_groupby_shift_diff_ngroup_vs_pandas(npartitions):
    """groupby.shift(p)/diff(p)/ngroup: within-run index arithmetic over
    the stable key sort; NaN keys excluded (NaN rows), run boundaries
    honoured for positive and negative periods."""
    rng = np.random.default_rng(98)
    n = 80_000
    k = rng.integers(0, 300, n).astype(np.float64)
    k[rng.random(n) < 0.02] = np.nan
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-50, 50, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for p in (1, 2, -1, 3, -2):
        got = df.groupby("k").shift(p).to_pandas()
        exp = pdf.groupby("k").shift(p)
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"shift({p})/{c}")
        got_d = df.groupby("k").diff(p).to_pandas()
        exp_d = pdf.groupby("k").diff(p)
        for c in exp_d.columns:
            np.testing.assert_allclose(got_d[c].to_numpy(),
                                       exp_d[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"diff({p})/{c}")
    got_ng = df.groupby("k").ngroup().to_pandas()
    exp_ng = pdf.groupby("k").ngroup()
    # pandas ngroup marks NaN-key rows -1 (int) unless they force floats;
    # measured on 2.3.3: float64 with NaN when NaN keys exist
    np.testing.assert_allclose(got_ng.to_numpy(),
                               exp_ng.to_numpy().astype(np.float64),
                               rtol=0, equal_nan=True)
    # int keys: exact int64 ngroup, multi-key lex order
    pdf2 = pandas.DataFrame({"a": rng.integers(0, 9, 5000),
                             "b": rng.choice(["p", "q", "r"], 5000),
                             "v": rng.standard_normal(5000)})
    df2 = mpd.DataFrame(pdf2)
    got2 = df2.groupby(["a", "b"]).ngroup().to_pandas()
    exp2 = pdf2.groupby(["a", "b"]).ngroup()
    assert got2.dtype == np.int64
    np.testing.assert_array_equal(got2.to_numpy(), exp2.to_numpy())
    s_ = df2.groupby(["a", "b"])["v"].shift(1).to_pandas()
    np.testing.assert_allclose(
        s_.to_numpy(), pdf2.groupby(["a", "b"])["v"].shift(1).to_numpy(),
        rtol=0, equal_nan=True)


def test_merge_float_nan_keys_vs_pandas(npartitions):
    """Float64 merge keys through the ordered bit transform: every NaN
    canonicalizes to ONE key so NaN==NaN matches — pandas merge
    semantics; inner + left + outer, mixed int64/float64 key promotion."""
    rng = np.random.default_rng(99)
    nl, nr_ = 20_000, 5_000
    lk = rng.choice(np.r_[rng.standard_normal(300), np.nan], nl)
    rk = rng.choice(np.r_[rng.standard_normal(400), np.nan], nr_)
    pl = pandas.DataFrame({"k": lk, "a": rng.standard_normal(nl)})
    pr = pandas.DataFrame({"k": rk, "b": rng.standard_normal(nr_)})
    ml, mr = mpd.DataFrame(pl), mpd.DataFrame(pr)
    for how in ("inner", "left", "outer"):
        got = ml.merge(mr, on="k", how=how).to_pandas()
        exp = pl.merge(pr, on="k", how=how)
        assert len(got) == len(exp), how
        # row order differs only by pandas' internal ordering: compare as
        # sorted multisets over all columns
        gs = got.sort_values(["k", "a", "b"],
                             na_position="last").reset_index(drop=True)
        es = exp.sort_values(["k", "a", "b"],
                             na_position="last").reset_index(drop=True)
        for c in ("k", "a", "b"):
            np.testing.assert_allclose(gs[c].to_numpy(), es[c].to_numpy(),
                                       rtol=0, equal_nan=True,
                                       err_msg=f"{how}/{c}")
    # mixed int64 left / float64 right key: promotes to float64
    pl2 = pandas.DataFrame({"k": rng.integers(0, 50, 1000), "a": rng.random(1000)})
    pr2 = pandas.DataFrame({"k": rng.integers(0, 50, 300).astype(np.float64),
                            "b": rng.random(300)})
    got2 = mpd.DataFrame(pl2).merge(mpd.DataFrame(pr2), on="k").to_pandas()
    exp2 = pl2.merge(pr2, on="k")
    assert got2["k"].dtype == np.float64
    assert len(got2) == len(exp2)
    np.testing.assert_allclose(
        np.sort(got2["b"].to_numpy()), np.sort(exp2["b"].to_numpy()),
        rtol=0)


def test_merge_nan_string_keys_vs_pandas(npartitions):
    """Dictionary (string) merge keys with NaN: code −1 matches code −1
    (pandas NaN==NaN); right categories absent from the left dictionary
    recode to −2 and never match."""
    rng = np.random.default_rng(100)
    lk = rng.choice(["a", "b", "c", None], 2000)
    rk = rng.choice(["b", "c", "d", None], 500)
    pl = pandas.DataFrame({"k": lk, "a": rng.standard_normal(2000)})
    pr = pandas.DataFrame({"k": rk, "b": rng.standard_normal(500)})
    got = mpd.DataFrame(pl).merge(mpd.DataFrame(pr), on="k").to_pandas()
    exp = pl.merge(pr, on="k")
    assert len(got) == len(exp)
    gs = got.sort_values(["k", "a", "b"]).reset_index(drop=True)
    es = exp.sort_values(["k", "a", "b"]).reset_index(drop=True)
    np.testing.assert_array_equal(gs["k"].fillna("<NA>").to_numpy(),
                                  es["k"].fillna("<NA>").to_numpy())
    np.testing.assert_allclose(gs["a"].to_numpy(), es["a"].to_numpy(),
                               rtol=0)
    np.testing.assert_allclose(gs["b"].to_numpy(), es["b"].to_numpy(),
                               rtol=0)


def test_merge_cross_vs_pandas(npartitions):
    """merge(how='cross'): cartesian product, '_x'/'_y' suffixes on every
    colliding name, left-major row order (pandas order)."""
    rng = np.random.default_rng(101)
    pl = pandas.DataFrame({"k": rng.integers(0, 5, 200),
                           "a": rng.standard_normal(200),
                           "s": rng.choice(["x", "y"], 200)})
    pr = pandas.DataFrame({"k": rng.integers(0, 5, 30),
                           "b": rng.standard_normal(30)})
    got = mpd.DataFrame(pl).merge(mpd.DataFrame(pr), how="cross").to_pandas()
    exp = pl.merge(pr, how="cross")
    assert list(got.columns) == list(exp.columns)
    assert list(got.dtypes) == list(exp.dtypes)
    for c in exp.columns:
        if exp[c].dtype == object:
            np.testing.assert_array_equal(got[c].to_numpy(),
                                          exp[c].to_numpy())
        else:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=0,
                                       equal_nan=True, err_msg=c)


def test_groupby_transform_broadcast_vs_pandas(npartitions):
    """gb.transform('sum'/'mean'/'count'/'min'/'max'): per-group aggregate
    broadcast to every row (NaN-value rows receive the group result; NaN
    keys get NaN; all-NaN groups: sum 0.0, min/max/mean NaN; int64
    survives only with valid keys)."""
    rng = np.random.default_rng(102)
    n = 100_000
    k = rng.integers(0, 400, n).astype(np.float64)
    k[rng.random(n) < 0.02] = np.nan
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-50, 50, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for agg in ("sum", "mean", "count", "min", "max"):
        got = df.groupby("k").transform(agg).to_pandas()
        exp = pdf.groupby("k").transform(agg)
        for c in exp.columns:
            np.testing.assert_allclose(
                got[c].to_numpy().astype(float),
                exp[c].to_numpy().astype(float),
                rtol=1e-12 if agg in ("sum", "mean") else 0, atol=1e-9,
                equal_nan=True, err_msg=f"{agg}/{c}")
    # all-NaN group + int dtype rule with valid keys
    pdf2 = pandas.DataFrame({"k": [1, 1, 2, 2, 3],
                             "v": [np.nan, np.nan, 1.0, 2.0, 5.0],
                             "w": [1, 2, 3, 4, 5]})
    df2 = mpd.DataFrame(pdf2)
    for agg in ("sum", "mean", "count", "min", "max"):
        got2 = df2.groupby("k").transform(agg).to_pandas()
        exp2 = pdf2.groupby("k").transform(agg)
        assert list(got2.dtypes) == list(exp2.dtypes), agg
        for c in exp2.columns:
            np.testing.assert_allclose(
                got2[c].to_numpy().astype(float),
                exp2[c].to_numpy().astype(float), rtol=0,
                equal_nan=True, err_msg=f"{agg}/{c}")
    s_ = df2.groupby("k")["v"].transform("mean").to_pandas()
    np.testing.assert_allclose(
        s_.to_numpy(), pdf2.groupby("k")["v"].transform("mean").to_numpy(),
        rtol=0, equal_nan=True)


def test_groupby_idxmax_idxmin_vs_pandas(npartitions):
    """gb.idxmax/idxmin: first original row label of the group extreme;
    all-NaN groups -> NaN (float64 column); ties resolve to first
    occurrence; NaN keys dropped."""
    rng = np.random.default_rng(103)
    n = 50_000
    k = rng.integers(0, 150, n).astype(np.float64)
    k[rng.random(n) < 0.02] = np.nan
    v = rng.integers(-6, 6, n).astype(np.float64)  # heavy ties
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-100, 100, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for fn in ("idxmax", "idxmin"):
        got = getattr(df.groupby("k"), fn)().to_pandas()
        exp = getattr(pdf.groupby("k"), fn)()
        np.testing.assert_array_equal(got.index.to_numpy(),
                                      exp.index.to_numpy())
        for c in exp.columns:
            np.testing.assert_allclose(
                got[c].to_numpy().astype(float),
                exp[c].to_numpy().astype(float), rtol=0, equal_nan=True,
                err_msg=f"{fn}/{c}")
    # all-NaN group + int columns + selection form
    pdf2 = pandas.DataFrame({"k": [1, 1, 2, 2, 3],
                             "v": [3.0, 3.0, np.nan, np.nan, 1.0],
                             "w": [9, 2, 3, 4, 5]})
    df2 = mpd.DataFrame(pdf2)
    for fn in ("idxmax", "idxmin"):
        got2 = getattr(df2.groupby("k"), fn)().to_pandas()
        exp2 = getattr(pdf2.groupby("k"), fn)()
        assert list(got2.dtypes) == list(exp2.dtypes), fn
        for c in exp2.columns:
            np.testing.assert_allclose(
                got2[c].to_numpy().astype(float),
                exp2[c].to_numpy().astype(float), rtol=0, equal_nan=True,
                err_msg=f"{fn}/{c}")
    s_ = df2.groupby("k")["w"].idxmax().to_pandas()
    np.testing.assert_array_equal(
        s_.to_numpy(), pdf2.groupby("k")["w"].idxmax().to_numpy())
    # multi-key (no NaN keys)
    pdf3 = pandas.DataFrame({"a": rng.integers(0, 5, 3000),
                             "b": rng.choice(["x", "y"], 3000),
                             "v": rng.standard_normal(3000)})
    df3 = mpd.DataFrame(pdf3)
    got3 = df3.groupby(["a", "b"]).idxmax().to_pandas()
    exp3 = pdf3.groupby(["a", "b"]).idxmax()
    np.testing.assert_array_equal(got3["v"].to_numpy(),
                                  exp3["v"].to_numpy())


def test_frame_rank_vs_pandas(npartitions):
    """DataFrame.rank / Series.rank (axis=0): the one-group rank
    composition."""
    rng = np.random.default_rng(104)
    n = 40_000
    v = rng.integers(-30, 30, n).astype(np.float64)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-1000, 1000, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for method in ("average", "min", "first"):
        for asc in (True, False):
            got = df.rank(method=method, ascending=asc).to_pandas()
            exp = pdf.rank(method=method, ascending=asc)
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=0,
                    equal_nan=True, err_msg=f"{method}/{asc}/{c}")
    s_ = df["v"].rank().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(), pdf["v"].rank().to_numpy(),
                               rtol=0, equal_nan=True)


def test_duplicated_drop_duplicates_vs_pandas(npartitions):
    """duplicated/drop_duplicates keep='first': cumcount>0 over ALL
    subset columns with dropna=False (NaN==NaN, pandas semantics);
    drop_duplicates keeps original index labels."""
    rng = np.random.default_rng(105)
    n = 40_000
    a = rng.integers(0, 40, n).astype(np.float64)
    a[rng.random(n) < 0.05] = np.nan
    b = rng.choice(["x", "y", "zz"], n)
    w = rng.integers(0, 6, n)
    pdf = pandas.DataFrame({"a": a, "b": b, "w": w})
    df = mpd.DataFrame(pdf)
    for subset in (None, ["a"], ["a", "b"], "w"):
        got = df.duplicated(subset).to_pandas()
        exp = pdf.duplicated(subset=subset)
        np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy(),
                                      err_msg=str(subset))
    got_d = df.drop_duplicates(["a", "b"]).to_pandas()
    exp_d = pdf.drop_duplicates(subset=["a", "b"])
    np.testing.assert_array_equal(got_d.index.to_numpy(),
                                  exp_d.index.to_numpy())
    np.testing.assert_array_equal(got_d["w"].to_numpy(),
                                  exp_d["w"].to_numpy())
    s_ = df["w"].drop_duplicates().to_pandas()
    e_ = pdf["w"].drop_duplicates()
    np.testing.assert_array_equal(s_.index.to_numpy(), e_.index.to_numpy())
    np.testing.assert_array_equal(s_.to_numpy(), e_.to_numpy())


def test_where_mask_round_vs_pandas(npartitions):
    rng = np.random.default_rng(106)
    n = 30_000
    v = rng.standard_normal(n) * 10
    v[rng.random(n) < 0.1] = np.nan
    w = rng.integers(-50, 50, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    cond_p = pdf["v"] > 0
    cond_m = df["v"] > 0
    got = df.where(cond_m).to_pandas()
    exp = pdf.where(cond_p)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    got = df.where(cond_m, -1).to_pandas()
    exp = pdf.where(cond_p, -1)
    assert list(got.dtypes) == list(exp.dtypes)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    got = df.mask(cond_m).to_pandas()
    exp = pdf.mask(cond_p)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    s_ = df["v"].where(cond_m, 0.5).to_pandas()
    np.testing.assert_allclose(s_.to_numpy(),
                               pdf["v"].where(cond_p, 0.5).to_numpy(),
                               rtol=0, equal_nan=True)
    for d in (0, 1, 2, -1):
        got = df.round(d).to_pandas()
        exp = pdf.round(d)
        assert list(got.dtypes) == list(exp.dtypes)
        np.testing.assert_allclose(got["v"].to_numpy(),
                                   exp["v"].to_numpy(), rtol=0,
                                   equal_nan=True, err_msg=f"round({d})")
        np.testing.assert_array_equal(got["w"].to_numpy(),
                                      exp["w"].to_numpy())


def test_nlargest_nsmallest_vs_pandas(npartitions):
    rng = np.random.default_rng(107)
    n = 20_000
    v = rng.integers(-100, 100, n).astype(np.float64)  # ties
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(0, 9, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for k in (5, 100, n + 50):
        got = df.nlargest(k, "v").to_pandas()
        exp = pdf.nlargest(k, "v")
        if k <= n:
            np.testing.assert_array_equal(got.index.to_numpy(),
                                          exp.index.to_numpy(),
                                          err_msg=str(k))
        else:
            # k >= len: pandas' tie order is unstable/unspecified there —
            # ours is the stable sort; compare as multisets
            np.testing.assert_array_equal(
                np.sort(got.index.to_numpy()),
                np.sort(exp.index.to_numpy()), err_msg=str(k))
        got = df.nsmallest(k, "v").to_pandas()
        exp = pdf.nsmallest(k, "v")
        if k <= n:
            np.testing.assert_array_equal(got.index.to_numpy(),
                                          exp.index.to_numpy())
        else:
            np.testing.assert_array_equal(
                np.sort(got.index.to_numpy()),
                np.sort(exp.index.to_numpy()))
    s_ = df["v"].nlargest(17).to_pandas()
    e_ = pdf["v"].nlargest(17)
    np.testing.assert_array_equal(s_.index.to_numpy(), e_.index.to_numpy())
    np.testing.assert_allclose(s_.to_numpy(), e_.to_numpy(), rtol=0)


def test_pct_change_between_vs_pandas(npartitions):
    rng = np.random.default_rng(108)
    n = 20_000
    v = rng.standard_normal(n) + 5
    v[rng.random(n) < 0.05] = np.nan
    pdf = pandas.DataFrame({"v": v, "w": rng.integers(1, 9, n)})
    df = mpd.DataFrame(pdf)
    for p in (1, 3):
        got = df.pct_change(p).to_pandas()
        exp = pdf.pct_change(p, fill_method=None)
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=1e-12,
                                       atol=1e-12, equal_nan=True,
                                       err_msg=f"pct({p})/{c}")
    for inc in ("both", "neither", "left", "right"):
        got = df["v"].between(4.0, 6.0, inclusive=inc).to_pandas()
        exp = pdf["v"].between(4.0, 6.0, inclusive=inc)
        np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy(),
                                      err_msg=inc)


def test_groupby_dropna_false_vs_pandas(npartitions):
    """groupby(dropna=False): NaN keys form a real group, sorted LAST
    (pandas index order), for the reduce aggs and the transform family;
    int value dtypes survive (NaN KEYS don't force float values)."""
    rng = np.random.default_rng(109)
    n = 60_000
    kf = rng.integers(0, 50, n).astype(np.float64)
    kf[rng.random(n) < 0.05] = np.nan
    ks = rng.choice(["a", "b", "c"], n).astype(object)
    ks[rng.random(n) < 0.05] = None
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.05] = np.nan
    w = rng.integers(-40, 40, n)
    for key in ("kf", "ks"):
        pdf = pandas.DataFrame({"k": kf if key == "kf" else ks,
                                "v": v, "w": w})
        df = mpd.DataFrame(pdf)
        for agg in ("sum", "count", "mean", "min", "max"):
            got = getattr(df.groupby("k", dropna=False), agg)().to_pandas()
            exp = getattr(pdf.groupby("k", dropna=False), agg)()
            assert len(got) == len(exp), f"{key}/{agg}"
            if key == "kf":
                np.testing.assert_allclose(
                    got.index.to_numpy().astype(float),
                    exp.index.to_numpy().astype(float), rtol=0,
                    equal_nan=True, err_msg=f"{key}/{agg}")
            else:
                np.testing.assert_array_equal(
                    pandas.Series(got.index).fillna("<NA>").to_numpy(),
                    pandas.Series(exp.index).fillna("<NA>").to_numpy(),
                    err_msg=f"{key}/{agg}")
            assert list(got.dtypes) == list(exp.dtypes), f"{key}/{agg}"
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy().astype(float),
                    exp[c].to_numpy().astype(float), rtol=1e-12,
                    atol=1e-9, equal_nan=True, err_msg=f"{key}/{agg}/{c}")
        # transform family under dropna=False: NaN-key rows get REAL
        # results (they belong to the NaN group)
        got_t = df.groupby("k", dropna=False).cumsum().to_pandas()
        exp_t = pdf.groupby("k", dropna=False).cumsum()
        for c in exp_t.columns:
            np.testing.assert_allclose(
                got_t[c].to_numpy(), exp_t[c].to_numpy(), rtol=1e-12,
                atol=1e-9, equal_nan=True, err_msg=f"{key}/cumsum/{c}")
        got_n = df.groupby("k", dropna=False).ngroup().to_pandas()
        exp_n = pdf.groupby("k", dropna=False).ngroup()
        np.testing.assert_array_equal(got_n.to_numpy(), exp_n.to_numpy(),
                                      err_msg=f"{key}/ngroup")
    # round 2: var/median under dropna=False are SUPPORTED now (the
    # sentinel-encoded key route; test_groupby_dropna_false_tail_aggs
    # covers the full matrix) — spot-check they run and match pandas
    df2 = mpd.DataFrame(pandas.DataFrame(
        {"k": [1.0, 2.0, np.nan, 2.0], "v": [1.0, 2.0, 3.0, 5.0]}))
    pdf2 = pandas.DataFrame(
        {"k": [1.0, 2.0, np.nan, 2.0], "v": [1.0, 2.0, 3.0, 5.0]})
    got2 = df2.groupby("k", dropna=False).median().to_pandas()
    exp2 = pdf2.groupby("k", dropna=False).median()
    np.testing.assert_allclose(got2["v"].to_numpy(),
                               exp2["v"].to_numpy(), rtol=0)


def test_groupby_ffill_bfill_vs_pandas(npartitions):
    """groupby/frame ffill & bfill: segmented MAX over valid positions
    (composition CPU-validated on the numpy lib mock,
    tests/test_mock_compositions.py; here the same assertions run on the
    real kernels)."""
    rng = np.random.default_rng(110)
    n = 60_000
    k = rng.integers(0, 200, n).astype(np.float64)
    k[rng.random(n) < 0.03] = np.nan
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.3] = np.nan
    w = rng.integers(-40, 40, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for how in ("ffill", "bfill"):
        got = getattr(df.groupby("k"), how)().to_pandas()
        exp = getattr(pdf.groupby("k"), how)()
        assert list(got.dtypes) == list(exp.dtypes), how
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"{how}/{c}")
        got2 = getattr(df, how)().to_pandas()
        exp2 = getattr(pdf, how)()
        assert list(got2.dtypes) == list(exp2.dtypes), how
        for c in exp2.columns:
            np.testing.assert_allclose(got2[c].to_numpy(),
                                       exp2[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"frame-{how}/{c}")
    # multi-key + selection forms
    pdf2 = pandas.DataFrame({"a": rng.integers(0, 5, 5000),
                             "b": rng.choice(["x", "y"], 5000),
                             "v": np.where(rng.random(5000) < 0.4, np.nan,
                                           rng.standard_normal(5000))})
    df2 = mpd.DataFrame(pdf2)
    got3 = df2.groupby(["a", "b"]).ffill().to_pandas()
    exp3 = pdf2.groupby(["a", "b"]).ffill()
    np.testing.assert_allclose(got3["v"].to_numpy(),
                               exp3["v"].to_numpy(), rtol=0,
                               equal_nan=True)
    s_ = df2.groupby(["a", "b"])["v"].bfill().to_pandas()
    np.testing.assert_allclose(
        s_.to_numpy(), pdf2.groupby(["a", "b"])["v"].bfill().to_numpy(),
        rtol=0, equal_nan=True)


def test_groupby_pct_change_vs_pandas(npartitions):
    rng = np.random.default_rng(111)
    n = 40_000
    k = rng.integers(0, 150, n).astype(np.float64)
    k[rng.random(n) < 0.02] = np.nan
    v = rng.standard_normal(n) + 4
    v[rng.random(n) < 0.1] = np.nan
    w = rng.integers(1, 50, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for p in (1, 3):
        got = df.groupby("k").pct_change(p).to_pandas()
        exp = pdf.groupby("k").pct_change(p, fill_method=None)
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=1e-12,
                                       atol=1e-12, equal_nan=True,
                                       err_msg=f"pct({p})/{c}")
    s_ = df.groupby("k")["v"].pct_change().to_pandas()
    np.testing.assert_allclose(
        s_.to_numpy(),
        pdf.groupby("k")["v"].pct_change(fill_method=None).to_numpy(),
        rtol=1e-12, atol=1e-12, equal_nan=True)


def test_fillna_dict_replace_vs_pandas(npartitions):
    rng = np.random.default_rng(112)
    n = 20_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.15] = np.nan
    u = rng.standard_normal(n)
    u[rng.random(n) < 0.15] = np.nan
    w = rng.integers(0, 9, n)
    pdf = pandas.DataFrame({"v": v, "u": u, "w": w})
    df = mpd.DataFrame(pdf)
    got = df.fillna({"v": -1.0, "u": 2.5}).to_pandas()
    exp = pdf.fillna({"v": -1.0, "u": 2.5})
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(), exp[c].to_numpy(),
                                   rtol=0, equal_nan=True, err_msg=c)
    s_ = df["v"].replace(float(v[0]), 99.0).to_pandas()
    e_ = pdf["v"].replace(float(v[0]), 99.0)
    np.testing.assert_allclose(s_.to_numpy(), e_.to_numpy(), rtol=0,
                               equal_nan=True)
    s2 = df["w"].replace(3, -7).to_pandas()
    e2 = pdf["w"].replace(3, -7)
    assert s2.dtype == e2.dtype
    np.testing.assert_array_equal(s2.to_numpy(), e2.to_numpy())


def test_rolling_vs_pandas(npartitions):
    """rolling(w, min_periods).sum/mean/count/min/max: prefix-sum windows
    + van Herk two-scan extremes (composition CPU-validated on the mock
    tier; here on the real kernels)."""
    rng = np.random.default_rng(113)
    n = 50_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.15] = np.nan
    w = rng.integers(-30, 30, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for win, mp in ((3, None), (16, 4), (250, 1), (7, 7)):
        for op in ("sum", "mean", "count", "min", "max"):
            got = getattr(df.rolling(win, min_periods=mp), op)() \
                .to_pandas()
            exp = getattr(pdf.rolling(win, min_periods=mp), op)()
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=1e-12,
                    atol=1e-9, equal_nan=True,
                    err_msg=f"{op}/w={win}/mp={mp}/{c}")
    s_ = df["v"].rolling(5).mean().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(),
                               pdf["v"].rolling(5).mean().to_numpy(),
                               rtol=1e-12, atol=1e-12, equal_nan=True)


def test_expanding_vs_pandas(npartitions):
    rng = np.random.default_rng(114)
    n = 30_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.2] = np.nan
    w = rng.integers(-9, 9, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for mp in (1, 5):
        for op in ("sum", "mean", "count", "min", "max"):
            got = getattr(df.expanding(mp), op)().to_pandas()
            exp = getattr(pdf.expanding(mp), op)()
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=1e-12,
                    atol=1e-9, equal_nan=True,
                    err_msg=f"{op}/mp={mp}/{c}")


def test_device_rng_matches_oracle_mirror(npartitions):
    """hf_fill_randint/randf64/randcdf must be bit-exact vs the oracle's
    splitmix64 mirrors (bench.py's verify gate depends on this)."""
    n = 1_000_003
    k = lib.get(lib.fill_randint(n, 42, 0, 1_000_000))
    np.testing.assert_array_equal(k, oracle.rand_int(42, n, 0, 1_000_000))
    assert k.min() >= 0 and k.max() < 1_000_000
    v = lib.get(lib.fill_randf64(n, 777))
    np.testing.assert_array_equal(v, oracle.rand_f64(777, n))
    assert v.min() >= 0.0 and v.max() < 1.0
    cdf = oracle.zipf_cdf(10_000, 1.2)
    z = lib.get(lib.fill_randcdf(n, 5, lib.put(cdf)))
    np.testing.assert_array_equal(z, oracle.rand_cdf(5, n, cdf))
    assert z.min() >= 0 and z.max() < 10_000
    # offset streams (bench shards chunks along one stream)
    k2 = lib.get(lib.fill_randint(500, 42 + 250, 0, 1_000_000))
    np.testing.assert_array_equal(
        k2, oracle.rand_int(42 + 250, 500, 0, 1_000_000))


def test_groupby_sum_on_device_generated_frame(npartitions):
    """End-to-end bench shape at small n: device-generated frame through the
    L1 groupby, checked against the oracle regeneration."""
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition
    from modin_amd.query_compiler import HipQueryCompiler

    n, K = 2_000_000, 5000
    chunks = oracle.split_row_counts(n, 3, 1)
    parts = []
    off = 0
    for cn in chunks:
        kc = lib.fill_randint(cn, 1000 + off, 0, K)
        vc = lib.fill_randf64(cn, 2000 + off)
        parts.append(HipDataframePartition(DeviceBlock({"k": kc, "v": vc},
                                                       cn)))
        off += cn
    frame = HipDataframe(parts, pandas.RangeIndex(n), ["k", "v"], chunks,
                         pandas.Series({"k": np.dtype(np.int64),
                                        "v": np.dtype(np.float64)}))
    df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))
    out = df.groupby("k").sum().to_pandas()

    acc = np.zeros(K)
    seen = np.zeros(K, dtype=bool)
    off = 0
    for cn in chunks:
        kk = oracle.rand_int(1000 + off, cn, 0, K)
        vv = oracle.rand_f64(2000 + off, cn)
        acc += np.bincount(kk, weights=vv, minlength=K)
        seen[kk] = True
        off += cn
    exp_keys = np.nonzero(seen)[0]
    np.testing.assert_array_equal(out.index.to_numpy(), exp_keys)
    np.testing.assert_allclose(out["v"].to_numpy(), acc[exp_keys],
                               rtol=RTOL, atol=1e-9)


def test_binned_merge_matches_pandas(npartitions, monkeypatch):
    """Co-shuffled (range-binned) merge — forced via MODIN_AMD_MERGE_BINS
    on small data — must equal pandas.merge exactly (inner + left, int64
    and float/NaN keys).  Reference: range_partitioning_merge
    (merge.py:39 -> dataframe.py:4087)."""
    monkeypatch.setenv("MODIN_AMD_MERGE_BINS", "5")
    rng = np.random.default_rng(77)
    nl, nr = 40_000, 9_000
    for key_kind in ("int", "float"):
        if key_kind == "int":
            lk = (rng.integers(-10**12, 10**12, nl) // 10**7).astype(np.int64)
            rk = (rng.integers(-10**12, 10**12, nr) // 10**7).astype(np.int64)
        else:
            lk = (rng.integers(-1000, 1000, nl) / 8.0)
            rk = (rng.integers(-1000, 1000, nr) / 8.0)
            lk[rng.random(nl) < 0.05] = np.nan
            rk[rng.random(nr) < 0.05] = np.nan
        lpdf = pandas.DataFrame({"k": lk, "a": rng.standard_normal(nl),
                                 "c": rng.integers(0, 100, nl)})
        rpdf = pandas.DataFrame({"k": rk, "b": rng.standard_normal(nr),
                                 "c": rng.integers(0, 100, nr)})
        for how in ("inner", "left"):
            got = mpd.DataFrame(lpdf).merge(mpd.DataFrame(rpdf), on="k",
                                            how=how).to_pandas()
            exp = lpdf.merge(rpdf, on="k", how=how)
            assert list(got.columns) == list(exp.columns)
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(dtype=np.float64),
                    exp[c].to_numpy(dtype=np.float64), rtol=0,
                    equal_nan=True, err_msg=f"{key_kind}/{how}/{c}")


@pytest.mark.slow
def test_binned_merge_beyond_csr_cap(npartitions):
    """Organic giant-right route: > 2^27 DISTINCT right keys (the round-1
    loud-error case) now runs through the binned merge.  Multiplicity-1
    construction keeps the expectation analytic (right value = 2.5*key)."""
    nr = (1 << 27) + 2_000_000
    nl = 2_000_000
    rk = np.arange(nr, dtype=np.int64) * 3 - 5
    rv = rk.astype(np.float64) * 2.5
    rng = np.random.default_rng(5)
    sel = rng.integers(0, nr, nl)
    lk = rk[sel].copy()
    # some left keys that match nothing
    lk[::97] = -10**17
    lv = rng.standard_normal(nl)
    ldf = mpd.DataFrame({"k": lk, "a": lv})
    rdf = mpd.DataFrame({"k": rk, "b": rv})
    out = ldf.merge(rdf, on="k", how="inner").to_pandas()
    mask = lk != -10**17
    np.testing.assert_array_equal(out["k"].to_numpy(), lk[mask])
    np.testing.assert_allclose(out["a"].to_numpy(), lv[mask], rtol=0)
    np.testing.assert_allclose(out["b"].to_numpy(),
                               lk[mask].astype(np.float64) * 2.5, rtol=0)


def test_datetime64_typed_columns(npartitions):
    """datetime64[ns] rides the int64 typed-column layer (DESIGN §Round-2
    roadmap 4): round trip, sort, filter vs Timestamp scalars, groupby
    (datetime values and datetime KEY with DatetimeIndex result), merge on
    datetime keys — all vs pandas (NaT-free; NaT is a loud later-round)."""
    rng = np.random.default_rng(88)
    n = 50_000
    base = pandas.Timestamp("2021-03-01").value
    tvals = base + rng.integers(0, 10**15, n)
    t = tvals.astype("datetime64[ns]")
    k = rng.integers(0, 300, n).astype(np.int64)
    v = rng.random(n)
    pdf = pandas.DataFrame({"t": t, "k": k, "v": v})
    df = mpd.DataFrame(pdf)

    # round trip preserves dtype + values
    back = df.to_pandas()
    assert back["t"].dtype == np.dtype("datetime64[ns]")
    np.testing.assert_array_equal(back["t"].to_numpy(), t)

    # sort by the datetime column
    got = df.sort_values("t").to_pandas()
    exp = pdf.sort_values("t", kind="stable")
    np.testing.assert_array_equal(got["t"].to_numpy(), exp["t"].to_numpy())
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())

    # filter vs a Timestamp scalar
    cut = pandas.Timestamp(base + 5 * 10**14)
    got = df[df["t"] > cut].to_pandas()
    exp = pdf[pdf["t"] > cut]
    np.testing.assert_array_equal(got["t"].to_numpy(), exp["t"].to_numpy())
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=0)

    # groupby with datetime VALUES (min/max of timestamps)
    got = df.groupby("k").max().to_pandas()
    exp = pdf.groupby("k").max()
    assert got["t"].dtype == np.dtype("datetime64[ns]")
    np.testing.assert_array_equal(got["t"].to_numpy(),
                                  exp["t"].to_numpy())

    # groupby BY a datetime key -> DatetimeIndex result
    day = pdf["t"].dt.floor("D")
    pdf2 = pandas.DataFrame({"d": day, "v": v})
    df2 = mpd.DataFrame(pdf2)
    got = df2.groupby("d").sum().to_pandas()
    exp = pdf2.groupby("d").sum()
    assert got.index.dtype == np.dtype("datetime64[ns]")
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=1e-12)

    # merge on a datetime key
    keys = pandas.Series(t).iloc[:500].reset_index(drop=True)
    rpdf = pandas.DataFrame({"t": keys, "b": rng.random(500)})
    got = df.merge(mpd.DataFrame(rpdf), on="t").to_pandas()
    exp = pdf.merge(rpdf, on="t")
    assert got["t"].dtype == np.dtype("datetime64[ns]")
    np.testing.assert_array_equal(got["t"].to_numpy(), exp["t"].to_numpy())
    np.testing.assert_allclose(got["b"].to_numpy(), exp["b"].to_numpy(),
                               rtol=0)

    # NaT round-trips (iNaT bits); tz-aware stays a loud error
    nat = pandas.DataFrame({"t": pandas.to_datetime(
        ["2020-01-01", None])})
    back = mpd.DataFrame(nat).to_pandas()
    np.testing.assert_array_equal(back["t"].to_numpy(),
                                  nat["t"].to_numpy())


def test_cumprod_vs_pandas(npartitions):
    """Product scan (AGG_PROD through the segmented/linear scan kernels):
    groupby.cumprod + frame cumprod vs pandas, NaN values skip-but-stay."""
    rng = np.random.default_rng(123)
    n = 60_000
    k = rng.integers(0, 500, n).astype(np.int64)
    v = np.clip(rng.standard_normal(n), -1.5, 1.5)
    v[rng.random(n) < 0.1] = np.nan
    w = (rng.integers(0, 3, n) - 1).astype(np.int64)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    got = df.groupby("k").cumprod().to_pandas()
    exp = pdf.groupby("k").cumprod()
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(dtype=float),
                                   exp[c].to_numpy(dtype=float),
                                   rtol=1e-12, atol=1e-300, equal_nan=True,
                                   err_msg=f"gb-cumprod/{c}")
    got = df[["v", "w"]].cumprod().to_pandas()
    exp = pdf[["v", "w"]].cumprod()
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(dtype=float),
                                   exp[c].to_numpy(dtype=float),
                                   rtol=1e-12, atol=1e-300, equal_nan=True,
                                   err_msg=f"cumprod/{c}")


def test_rank_na_option_vs_pandas(npartitions):
    rng = np.random.default_rng(321)
    n = 40_000
    k = rng.integers(0, 300, n).astype(np.int64)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.15] = np.nan
    w = rng.integers(-9, 9, n).astype(np.int64)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    for na in ("top", "bottom", "keep"):
        for method in ("average", "min", "first"):
            for asc in (True, False):
                got = df.groupby("k").rank(method=method, ascending=asc,
                                           na_option=na).to_pandas()
                exp = pdf.groupby("k").rank(method=method, ascending=asc,
                                            na_option=na)
                for c in exp.columns:
                    np.testing.assert_allclose(
                        got[c].to_numpy(), exp[c].to_numpy(), rtol=0,
                        equal_nan=True,
                        err_msg=f"gb/{na}/{method}/asc={asc}/{c}")
        got = df[["v", "w"]].rank(na_option=na).to_pandas()
        exp = pdf[["v", "w"]].rank(na_option=na)
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"frame/{na}/{c}")


def test_groupby_dropna_false_tail_aggs(npartitions):
    """dropna=False for var/std/median/quantile/nunique/size/idxmax/first/
    last (round-2 completion): NaN float keys form a real trailing group
    via the sentinel-encoded key (reference pins these in
    modin/tests/pandas/test_groupby.py)."""
    rng = np.random.default_rng(55)
    n = 30_000
    k = rng.integers(0, 200, n).astype(np.float64)
    k[rng.random(n) < 0.06] = np.nan
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.integers(-50, 50, n).astype(np.int64)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    gb = df.groupby("k", dropna=False)
    egb = pdf.groupby("k", dropna=False)
    for op in ("var", "std", "median", "nunique", "first", "last"):
        got = getattr(gb, op)().to_pandas()
        exp = getattr(egb, op)()
        np.testing.assert_allclose(got.index.to_numpy(),
                                   exp.index.to_numpy(), rtol=0,
                                   equal_nan=True, err_msg=f"{op} keys")
        for c in exp.columns:
            np.testing.assert_allclose(
                got[c].to_numpy(dtype=np.float64),
                exp[c].to_numpy(dtype=np.float64), rtol=1e-12, atol=1e-9,
                equal_nan=True, err_msg=f"{op}/{c}")
    got = gb.quantile(0.25).to_pandas()
    exp = egb.quantile(0.25)
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(), exp[c].to_numpy(),
                                   rtol=1e-12, atol=1e-12, equal_nan=True,
                                   err_msg=f"quantile/{c}")
    got = gb.size()
    exp = egb.size()
    np.testing.assert_array_equal(np.asarray(got), exp.to_numpy())
    got = gb.idxmax().to_pandas()
    exp = egb.idxmax()
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(dtype=np.float64),
                                   exp[c].to_numpy(dtype=np.float64),
                                   rtol=0, equal_nan=True,
                                   err_msg=f"idxmax/{c}")


def test_multikey_idx_nan_keys(npartitions):
    """Multi-key idxmax/idxmin with NaN string keys (filter-first lift)."""
    rng = np.random.default_rng(3)
    n = 20_000
    a = rng.choice(["x", "y", "z", None], n,
                   p=[0.3, 0.3, 0.3, 0.1]).astype(object)
    b = rng.integers(0, 10, n)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    df = mpd.DataFrame(pdf)
    for mx in (True, False):
        got = (df.groupby(["a", "b"]).idxmax() if mx
               else df.groupby(["a", "b"]).idxmin()).to_pandas()
        exp = (pdf.groupby(["a", "b"]).idxmax() if mx
               else pdf.groupby(["a", "b"]).idxmin())
        assert list(got.index) == list(exp.index)
        np.testing.assert_allclose(got["v"].to_numpy().astype(float),
                                   exp["v"].to_numpy().astype(float),
                                   rtol=0, equal_nan=True)


def test_unbounded_multikey_groupby(npartitions):
    """Combined key span beyond 2^62: the sorted-heads dense-rank fold."""
    rng = np.random.default_rng(17)
    n = 200_000
    k1 = rng.integers(-2**61, 2**61, n)
    k1[rng.random(n) < 0.3] = 77
    k2 = rng.integers(-2**61, 2**61, n)
    k2[rng.random(n) < 0.3] = -5
    v = rng.standard_normal(n)
    pdf = pandas.DataFrame({"a": k1, "b": k2, "v": v})
    df = mpd.DataFrame(pdf)
    for agg in ("sum", "count", "mean"):
        got = getattr(df.groupby(["a", "b"]), agg)().to_pandas()
        exp = getattr(pdf.groupby(["a", "b"]), agg)()
        assert list(got.index) == list(exp.index), f"{agg} keys"
        np.testing.assert_allclose(got["v"].to_numpy(),
                                   exp["v"].to_numpy(), rtol=1e-12,
                                   err_msg=agg)


def test_multikey_dropna_false_vs_pandas(npartitions):
    """Multi-key groupby(dropna=False) on the GPU path (string + int key)."""
    rng = np.random.default_rng(8)
    n = 25_000
    a = rng.choice(["x", "y", None], n, p=[0.45, 0.45, 0.1]).astype(object)
    b = rng.integers(0, 6, n)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    df = mpd.DataFrame(pdf)
    for agg in ("sum", "count", "mean", "min", "max"):
        got = getattr(df.groupby(["a", "b"], dropna=False),
                      agg)().to_pandas()
        exp = getattr(pdf.groupby(["a", "b"], dropna=False), agg)()
        assert len(got) == len(exp), agg
        gi = [(x if isinstance(x, str) else "<NA>", y)
              for x, y in got.index]
        ei = [(x if isinstance(x, str) else "<NA>", y)
              for x, y in exp.index]
        assert gi == ei, f"{agg} keys"
        np.testing.assert_allclose(got["v"].to_numpy().astype(float),
                                   exp["v"].to_numpy().astype(float),
                                   rtol=1e-12, atol=1e-9, equal_nan=True)


def test_dt_accessor_vs_pandas(npartitions):
    """Series.dt calendar fields on device (exact int64 civil math)."""
    rng = np.random.default_rng(5)
    n = 60_000
    ns = rng.integers(-2 * 10**18, 2 * 10**18, n)
    t = pandas.Series(ns.astype("datetime64[ns]"), name="t")
    pdf = pandas.DataFrame({"t": t})
    df = mpd.DataFrame(pdf)
    for f in ("year", "month", "day", "hour", "minute", "second",
              "dayofweek"):
        got = getattr(df["t"].dt, f).to_pandas()
        exp = getattr(t.dt, f)
        np.testing.assert_array_equal(np.asarray(got), exp.to_numpy(),
                                      err_msg=f)
        assert np.asarray(got).dtype == exp.to_numpy().dtype, f


def test_groupby_prod_vs_pandas(npartitions):
    """groupby.prod on device: segmented PROD scan + last-valid pick."""
    rng = np.random.default_rng(12)
    n = 40_000
    k = rng.integers(0, 300, n).astype(np.int64)
    v = np.clip(rng.standard_normal(n), -1.4, 1.4)
    v[rng.random(n) < 0.1] = np.nan
    w = (rng.integers(0, 3, n) - 1).astype(np.int64)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    df = mpd.DataFrame(pdf)
    got = df.groupby("k").prod().to_pandas()
    exp = pdf.groupby("k").prod()
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(dtype=float),
                                   exp[c].to_numpy(dtype=float),
                                   rtol=1e-12, atol=1e-300,
                                   err_msg=f"prod/{c}")
    assert list(got.dtypes) == list(exp.dtypes)
    # float keys + dropna=False route
    kf = k.astype(np.float64)
    kf[rng.random(n) < 0.05] = np.nan
    pdf2 = pandas.DataFrame({"k": kf, "v": v})
    df2 = mpd.DataFrame(pdf2)
    got = df2.groupby("k", dropna=False).prod().to_pandas()
    exp = pdf2.groupby("k", dropna=False).prod()
    np.testing.assert_allclose(got.index.to_numpy(), exp.index.to_numpy(),
                               rtol=0, equal_nan=True)
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=1e-12, atol=1e-300)


def test_iloc_vs_pandas(npartitions):
    rng = np.random.default_rng(2)
    n = 50_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 999, n),
                            "v": rng.random(n)})
    df = mpd.DataFrame(pdf)
    pandas.testing.assert_frame_equal(df.iloc[1717:42000].to_pandas(),
                                      pdf.iloc[1717:42000])
    sel = rng.integers(-n, n, 5000).tolist()
    pandas.testing.assert_frame_equal(df.iloc[sel].to_pandas(),
                                      pdf.iloc[sel])
    assert df.iloc[4242]["k"] == pdf.iloc[4242]["k"]
    assert abs(df["v"].iloc[-3] - pdf["v"].iloc[-3]) < 1e-15


def test_sort_index_vs_pandas(npartitions):
    rng = np.random.default_rng(4)
    n = 30_000
    pdf = pandas.DataFrame({"v": rng.random(n)},
                           index=rng.integers(0, 500, n))
    df = mpd.DataFrame(pdf)
    for asc in (True, False):
        got = df.sort_index(ascending=asc).to_pandas()
        pandas.testing.assert_frame_equal(got,
                                          pdf.sort_index(ascending=asc))


def test_merge_big_duplicate_keys(npartitions):
    """Per-key right multiplicity beyond the 4096 in-thread fixup: the
    sorted-build fallback keeps pandas match order (round 2 — removes the
    last r1 merge cap)."""
    rng = np.random.default_rng(31)
    nr = 60_000
    rk = np.full(nr, 7, dtype=np.int64)
    m = rng.random(nr) < 0.2
    rk[m] = rng.integers(0, 50, int(m.sum()))
    rv = rng.random(nr)
    lk = np.array([7, 3, 7, 99, 12], dtype=np.int64)
    lv = rng.random(5)
    lpdf = pandas.DataFrame({"k": lk, "a": lv})
    rpdf = pandas.DataFrame({"k": rk, "b": rv})
    for how in ("inner", "left"):
        got = mpd.DataFrame(lpdf).merge(mpd.DataFrame(rpdf), on="k",
                                        how=how).to_pandas()
        exp = lpdf.merge(rpdf, on="k", how=how)
        assert len(got) == len(exp)
        for c in ("k", "a", "b"):
            np.testing.assert_allclose(
                got[c].to_numpy(), exp[c].to_numpy(), rtol=0,
                equal_nan=True, err_msg=f"{how}/{c}")


def test_loc_vs_pandas(npartitions):
    rng = np.random.default_rng(6)
    n = 30_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 9, n),
                            "v": rng.random(n),
                            "w": rng.integers(-5, 5, n)})
    df = mpd.DataFrame(pdf)
    m = pdf["v"] > 0.5
    pandas.testing.assert_frame_equal(
        df.loc[df["v"] > 0.5].to_pandas(), pdf.loc[m])
    pandas.testing.assert_frame_equal(
        df.loc[df["v"] > 0.5, ["k", "w"]].to_pandas(),
        pdf.loc[m, ["k", "w"]])
    pandas.testing.assert_frame_equal(df.loc[100:2000].to_pandas(),
                                      pdf.loc[100:2000])


def test_astype_datetime_round_trip(npartitions):
    rng = np.random.default_rng(44)
    n = 20_000
    ns = rng.integers(0, 2 * 10**18, n)
    pdf = pandas.DataFrame({"x": ns})
    df = mpd.DataFrame(pdf)
    as_dt = df.astype("datetime64[ns]").to_pandas()
    pandas.testing.assert_frame_equal(as_dt, pdf.astype("datetime64[ns]"))
    t = pandas.DataFrame({"t": ns.astype("datetime64[ns]")})
    back = mpd.DataFrame(t).astype("int64").to_pandas()
    pandas.testing.assert_frame_equal(back, t.astype("int64"))


def test_multikey_merge_vs_pandas(npartitions):
    """merge(on=[...]) inner/left with int/float-NaN/string keys."""
    rng = np.random.default_rng(21)
    nl, nr = 20_000, 8_000
    lpdf = pandas.DataFrame({
        "a": rng.integers(0, 40, nl),
        "b": (rng.integers(-50, 50, nl) / 8.0),
        "s": rng.choice(["u", "v", "w"], nl).astype(object),
        "x": rng.random(nl)})
    lpdf.loc[rng.random(nl) < 0.05, "b"] = np.nan
    rpdf = pandas.DataFrame({
        "a": rng.integers(0, 40, nr),
        "b": (rng.integers(-50, 50, nr) / 8.0),
        "s": rng.choice(["u", "v", "z"], nr).astype(object),
        "y": rng.random(nr), "x": rng.random(nr)})
    rpdf.loc[rng.random(nr) < 0.05, "b"] = np.nan
    for keys in (["a", "b"], ["a", "s"], ["a", "b", "s"]):
        for how in ("inner", "left"):
            got = mpd.DataFrame(lpdf).merge(
                mpd.DataFrame(rpdf), on=keys, how=how).to_pandas()
            exp = lpdf.merge(rpdf, on=keys, how=how)
            assert list(got.columns) == list(exp.columns), (keys, how)
            assert len(got) == len(exp), (keys, how)
            for c in exp.columns:
                g, e = got[c].to_numpy(), exp[c].to_numpy()
                if e.dtype == object:
                    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
                    assert same.all(), (keys, how, c)
                else:
                    np.testing.assert_allclose(
                        g.astype(float), e.astype(float), rtol=0,
                        equal_nan=True, err_msg=f"{keys}/{how}/{c}")


def test_merge_left_on_right_on_vs_pandas(npartitions):
    """merge(left_on=, right_on=): rewrite over a zero-copy key alias —
    both key columns survive as payload (pandas keep-keys rule), NaN
    fills the unmatched side's key, suffixes on other collisions; all
    four hows; int, float-NaN and dictionary (string) keys."""
    rng = np.random.default_rng(123)
    nl, nr = 30_000, 9_000
    lpdf = pandas.DataFrame({
        "a": rng.choice(np.r_[rng.standard_normal(200), np.nan], nl),
        "v": rng.standard_normal(nl),
        "c": rng.integers(0, 9, nl)})
    rpdf = pandas.DataFrame({
        "b": rng.choice(np.r_[rng.standard_normal(250), np.nan], nr),
        "w": rng.standard_normal(nr),
        "c": rng.integers(10, 19, nr)})
    for how in ("inner", "left", "right", "outer"):
        got = mpd.DataFrame(lpdf).merge(
            mpd.DataFrame(rpdf), left_on="a", right_on="b",
            how=how).to_pandas()
        exp = lpdf.merge(rpdf, left_on="a", right_on="b", how=how)
        assert list(got.columns) == list(exp.columns), how
        assert len(got) == len(exp), how
        order = ["a", "b", "v", "w", "c_x", "c_y"]
        gs = got.sort_values(order, na_position="last").reset_index(drop=True)
        es = exp.sort_values(order, na_position="last").reset_index(drop=True)
        for c in exp.columns:
            np.testing.assert_allclose(
                gs[c].to_numpy().astype(float),
                es[c].to_numpy().astype(float), rtol=0,
                equal_nan=True, err_msg=f"{how}/{c}")
    # dictionary keys under different names, mismatched dictionaries
    lp = pandas.DataFrame({"s": rng.choice(["a", "b", "c", None], 4000),
                           "v": rng.standard_normal(4000)})
    rp = pandas.DataFrame({"t": rng.choice(["b", "c", "d", None], 1500),
                           "w": rng.standard_normal(1500)})
    for how in ("inner", "left"):
        got = mpd.DataFrame(lp).merge(mpd.DataFrame(rp), left_on="s",
                                      right_on="t", how=how).to_pandas()
        exp = lp.merge(rp, left_on="s", right_on="t", how=how)
        assert list(got.columns) == list(exp.columns)
        assert len(got) == len(exp), how
        gs = got.sort_values(["s", "t", "v", "w"]).reset_index(drop=True)
        es = exp.sort_values(["s", "t", "v", "w"]).reset_index(drop=True)
        for c in ("s", "t"):
            np.testing.assert_array_equal(
                gs[c].fillna("<NA>").to_numpy(),
                es[c].fillna("<NA>").to_numpy(), err_msg=f"{how}/{c}")
        for c in ("v", "w"):
            np.testing.assert_allclose(gs[c].to_numpy(), es[c].to_numpy(),
                                       rtol=0, equal_nan=True,
                                       err_msg=f"{how}/{c}")
    # int keys, len-1 list form, error surfaces
    lp2 = pandas.DataFrame({"i": rng.integers(0, 40, 2000),
                            "v": rng.standard_normal(2000)})
    rp2 = pandas.DataFrame({"j": rng.integers(0, 40, 700),
                            "w": rng.standard_normal(700)})
    got = mpd.DataFrame(lp2).merge(mpd.DataFrame(rp2), left_on=["i"],
                                   right_on=["j"]).to_pandas()
    exp = lp2.merge(rp2, left_on="i", right_on="j")
    assert len(got) == len(exp)
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_allclose(np.sort(got["w"].to_numpy()),
                               np.sort(exp["w"].to_numpy()), rtol=0)
    import modin_amd.core.lib as hl
    with pytest.raises(hl.HfError):
        mpd.DataFrame(lp2).merge(mpd.DataFrame(rp2), left_on="i")
    with pytest.raises(hl.HfError):
        mpd.DataFrame(lp2).merge(mpd.DataFrame(rp2), on="i", left_on="i",
                                 right_on="j")


def test_setitem_insert_assign_vs_pandas(npartitions):
    """Column assignment on device: derived columns (zero host traffic),
    scalar broadcasts via device fill, host arrays (strings dictionary-
    encode, datetimes tag), insert-at-position, assign."""
    rng = np.random.default_rng(124)
    n = 50_000
    pdf = pandas.DataFrame({"a": rng.integers(0, 100, n),
                            "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    exp = pdf.copy()
    df["v"] = df["v"] * 3.0
    exp["v"] = exp["v"] * 3.0
    df["w"] = df["a"] - df["v"]
    exp["w"] = exp["a"] - exp["v"]
    df["k7"] = 7
    exp["k7"] = 7
    df["nn"] = np.nan
    exp["nn"] = np.nan
    df["s"] = "zz"
    exp["s"] = "zz"
    strs = rng.choice(["p", "q", None], n)
    df["t"] = strs
    exp["t"] = strs
    ts = pandas.Series(
        pandas.to_datetime("2024-01-01")
        + pandas.to_timedelta(rng.integers(0, 10_000, n), unit="m"))
    df["d"] = ts
    exp["d"] = ts
    got = df.to_pandas()
    assert list(got.columns) == list(exp.columns)
    assert got["d"].dtype == exp["d"].dtype
    for c in exp.columns:
        g, e = got[c].to_numpy(), exp[c].to_numpy()
        if e.dtype == object:
            same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
            assert same.all(), c
        elif c == "d":
            np.testing.assert_array_equal(g, e)
        else:
            np.testing.assert_allclose(g.astype(float), e.astype(float),
                                       rtol=0, equal_nan=True, err_msg=c)
    # derived column keeps working downstream: groupby over the new key
    g1 = df[["k7", "a", "v"]].groupby("k7").sum().to_pandas()
    e1 = exp[["k7", "a", "v"]].groupby("k7").sum()
    assert len(g1) == 1
    np.testing.assert_allclose(g1.to_numpy(), e1.to_numpy(), rtol=1e-12)
    # insert + assign
    df.insert(0, "z", df["a"] * 2)
    exp.insert(0, "z", exp["a"] * 2)
    assert list(df.columns) == list(exp.columns)
    out = df.assign(q=lambda d: d["z"] + 1)
    expq = exp.assign(q=lambda d: d["z"] + 1)
    np.testing.assert_allclose(
        out.to_pandas()["q"].to_numpy().astype(float),
        expq["q"].to_numpy().astype(float), rtol=0)


def test_series_map_replace_dict_vs_pandas(npartitions):
    """Series.map(dict)/replace(dict): int64 via hf_search_sorted LUT
    (coverage-dependent result dtype), strings via host-dictionary remap
    + one gather."""
    rng = np.random.default_rng(125)
    n = 80_000
    pdf = pandas.DataFrame({"i": rng.integers(-50, 50, n),
                            "s": rng.choice(["aa", "bb", "cc", None], n)})
    df = mpd.DataFrame(pdf)
    full = {k: int(k) * 3 - 7 for k in range(-50, 50)}
    got = df["i"].map(full).to_pandas()
    exp = pdf["i"].map(full)
    assert got.dtype == exp.dtype
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    part = {k: float(k) / 4 for k in range(-10, 10)}
    got = df["i"].map(part).to_pandas()
    exp = pdf["i"].map(part)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy().astype(float),
                               rtol=0, equal_nan=True)
    rep = {0: 1000, -7: 7000, 13: -13000}
    got = df["i"].replace(rep).to_pandas()
    exp = pdf["i"].replace(rep)
    assert got.dtype == exp.dtype
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    got = df["s"].map({"aa": "x", "cc": "y"}).to_pandas()
    exp = pdf["s"].map({"aa": "x", "cc": "y"})
    g, e = got.to_numpy(), exp.to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()
    got = df["s"].replace({"bb": "BB"}).to_pandas()
    exp = pdf["s"].replace({"bb": "BB"})
    g, e = got.to_numpy(), exp.to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()
    got = df["s"].map({"aa": 1.5, "bb": 2, "cc": 3}).to_pandas()
    exp = pdf["s"].map({"aa": 1.5, "bb": 2, "cc": 3})
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy().astype(float),
                               rtol=0, equal_nan=True)


def test_melt_pivot_table_vs_pandas(npartitions):
    """melt on device (value alias + variable fill + concat) and
    pivot_table (device multi-key groupby + host unstack of the reduced
    table), NaN values included."""
    rng = np.random.default_rng(126)
    n = 60_000
    pdf = pandas.DataFrame({
        "id": rng.integers(0, 50, n),
        "g": rng.choice(["r", "s", "t", "u"], n),
        "x": rng.standard_normal(n),
        "y": rng.integers(-5, 5, n),
        "z": rng.standard_normal(n)})
    pdf.loc[rng.random(n) < 0.1, "x"] = np.nan
    df = mpd.DataFrame(pdf)
    got = df.melt(id_vars=["id", "g"]).to_pandas()
    exp = pdf.melt(id_vars=["id", "g"])
    assert list(got.columns) == list(exp.columns)
    assert len(got) == len(exp)
    for c in exp.columns:
        g, e = got[c].to_numpy(), exp[c].to_numpy()
        if e.dtype == object:
            np.testing.assert_array_equal(g, e, err_msg=c)
        else:
            np.testing.assert_allclose(g.astype(float), e.astype(float),
                                       rtol=0, equal_nan=True, err_msg=c)
    for aggfunc in ("mean", "sum", "min", "max"):
        got = df.pivot_table(values="x", index="id", columns="g",
                             aggfunc=aggfunc).to_pandas()
        exp = pdf.pivot_table(values="x", index="id", columns="g",
                              aggfunc=aggfunc)
        assert list(got.columns) == list(exp.columns), aggfunc
        np.testing.assert_array_equal(np.asarray(got.index),
                                      np.asarray(exp.index))
        np.testing.assert_allclose(got.to_numpy().astype(float),
                                   exp.to_numpy().astype(float),
                                   rtol=1e-12, equal_nan=True,
                                   err_msg=aggfunc)


def test_sample_device_draw(npartitions):
    """sample(n) on device: exact count, no duplicates, values/labels
    consistent with the source, seed-reproducible, and the mock tier's
    oracle-RNG mirror agrees with the device draw."""
    rng = np.random.default_rng(127)
    n = 200_000
    pdf = pandas.DataFrame({"a": np.arange(n),
                            "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    got = df.sample(n=1000, random_state=11).to_pandas()
    assert len(got) == 1000
    assert got["a"].is_unique
    np.testing.assert_allclose(
        got["v"].to_numpy(), pdf.loc[got["a"].to_numpy(), "v"].to_numpy(),
        rtol=0)
    got2 = df.sample(n=1000, random_state=11).to_pandas()
    np.testing.assert_array_equal(got["a"].to_numpy(), got2["a"].to_numpy())
    # device RNG == oracle mirror: the selected set is the argsort of
    # the oracle's uniform stream
    from oracle import ops as oops
    keys = oops.rand_f64(11, n)
    exp_sel = np.argsort(keys, kind="stable")[:1000]
    np.testing.assert_array_equal(np.sort(got["a"].to_numpy()),
                                  np.sort(exp_sel))
    assert len(df.sample(frac=0.5)) == n // 2


def test_floordiv_mod_vs_pandas(npartitions):
    """int64 // and % with int scalars: HF_MAP_IDIV/IMOD Python floor
    semantics, exact over negatives."""
    rng = np.random.default_rng(128)
    pdf = pandas.DataFrame({"a": rng.integers(-10**12, 10**12, 100_000)})
    df = mpd.DataFrame(pdf)
    for k in (7, -7, 86_400_000_000_000):
        np.testing.assert_array_equal(
            (df["a"] // k).to_pandas().to_numpy(),
            (pdf["a"] // k).to_numpy(), err_msg=f"//{k}")
        np.testing.assert_array_equal(
            (df["a"] % k).to_pandas().to_numpy(),
            (pdf["a"] % k).to_numpy(), err_msg=f"%{k}")


def test_corr_cov_vs_pandas(npartitions):
    """corr/cov: pairwise-complete masked moments on device (NaN-
    propagating z = x + 0*y + NaN-skipping reduce)."""
    rng = np.random.default_rng(129)
    n = 150_000
    pdf = pandas.DataFrame({"x": rng.standard_normal(n),
                            "y": rng.standard_normal(n),
                            "z": rng.standard_normal(n),
                            "w": rng.integers(-50, 50, n)})
    pdf["y"] += 0.7 * pdf["x"]
    pdf.loc[rng.random(n) < 0.15, "x"] = np.nan
    pdf.loc[rng.random(n) < 0.15, "y"] = np.nan
    df = mpd.DataFrame(pdf)
    np.testing.assert_allclose(df.corr().to_numpy(),
                               pdf.corr().to_numpy(), rtol=1e-9,
                               equal_nan=True)
    np.testing.assert_allclose(df.cov().to_numpy(),
                               pdf.cov().to_numpy(), rtol=1e-9,
                               equal_nan=True)


def test_concat_column_alignment_vs_pandas(npartitions):
    """concat with mismatched columns: NaN fills, appearance-order
    union, int->float promotion (pandas outer-align rules)."""
    rng = np.random.default_rng(130)
    p1 = pandas.DataFrame({"a": rng.integers(0, 90, 20_000),
                           "v": rng.standard_normal(20_000),
                           "s": rng.choice(["x", "y", None], 20_000)})
    p2 = pandas.DataFrame({"a": rng.integers(0, 90, 9_000),
                           "w": rng.integers(-50, 50, 9_000)})
    got = mpd.concat([mpd.DataFrame(p1), mpd.DataFrame(p2)],
                     ignore_index=True).to_pandas()
    exp = pandas.concat([p1, p2], ignore_index=True)
    assert list(got.columns) == list(exp.columns)
    assert list(got.dtypes) == list(exp.dtypes)
    for c in exp.columns:
        g, e = got[c].to_numpy(), exp[c].to_numpy()
        if e.dtype == object:
            same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
            assert same.all(), c
        else:
            np.testing.assert_allclose(g.astype(float), e.astype(float),
                                       rtol=0, equal_nan=True, err_msg=c)
    # downstream groupby over the aligned result still runs on device
    r = got.groupby("a")["w"].count()
    gdf = mpd.concat([mpd.DataFrame(p1), mpd.DataFrame(p2)],
                     ignore_index=True)
    rg = gdf.groupby("a")["w"].count().to_pandas()
    np.testing.assert_array_equal(rg.to_numpy(), r.to_numpy())


def test_nat_semantics_vs_pandas(npartitions):
    """NaT on device (iNaT ns bits): round trip, masks, compares, dt
    fields, sort na_position, groupby-key drop, shift fill,
    fillna(Timestamp), inner merge NaT==NaT, loud guards."""
    rng = np.random.default_rng(133)
    n = 40_000
    t = pandas.Series(pandas.to_datetime("2021-03-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**6, n), unit="min"))
    t[rng.random(n) < 0.15] = pandas.NaT
    pdf = pandas.DataFrame({"t": t, "v": rng.standard_normal(n),
                            "k": rng.integers(0, 70, n)})
    df = mpd.DataFrame(pdf)
    back = df.to_pandas()
    np.testing.assert_array_equal(back["t"].to_numpy(),
                                  pdf["t"].to_numpy())
    np.testing.assert_array_equal(df["t"].notna().to_pandas().to_numpy(),
                                  pdf["t"].notna().to_numpy())
    got = df.dropna().to_pandas()
    exp = pdf.dropna()
    assert len(got) == len(exp)
    np.testing.assert_array_equal(got["t"].to_numpy(),
                                  exp["t"].to_numpy())
    ts = pandas.Timestamp("2021-06-01")
    for op in ("__gt__", "__ge__", "__lt__", "__le__", "__eq__",
               "__ne__"):
        g = getattr(df["t"], op)(ts).to_pandas().to_numpy()
        e = getattr(pdf["t"], op)(ts).to_numpy()
        np.testing.assert_array_equal(g.astype(bool), e, err_msg=op)
    for f in ("year", "month", "day", "dayofweek", "hour", "minute",
              "second"):
        g = getattr(df["t"].dt, f).to_pandas()
        e = getattr(pdf["t"].dt, f)
        np.testing.assert_allclose(g.to_numpy().astype(float),
                                   e.to_numpy().astype(float), rtol=0,
                                   equal_nan=True, err_msg=f)
    for asc in (True, False):
        for nap in ("last", "first"):
            g = df.sort_values("t", ascending=asc,
                               na_position=nap).to_pandas()
            e = pdf.sort_values("t", ascending=asc, na_position=nap,
                                kind="stable")
            np.testing.assert_array_equal(g["t"].to_numpy(),
                                          e["t"].to_numpy(),
                                          err_msg=f"{asc}/{nap}")
            np.testing.assert_array_equal(np.asarray(g.index),
                                          e.index.to_numpy())
    g = df.groupby("t").sum().to_pandas()
    e = pdf.groupby("t").sum()
    assert len(g) == len(e)
    assert g.index.dtype == e.index.dtype
    np.testing.assert_array_equal(g.index.to_numpy(), e.index.to_numpy())
    np.testing.assert_allclose(g["v"].to_numpy(), e["v"].to_numpy(),
                               rtol=1e-12)
    g = df.groupby("t").size()
    e = pdf.groupby("t").size()
    np.testing.assert_array_equal(np.asarray(g), e.to_numpy())
    with pytest.raises(lib.HfError):
        df.groupby("t", dropna=False).sum()
    with pytest.raises(lib.HfError):
        df.groupby("k").min()  # NaT in a VALUE column
    g = df[["t"]].shift(-3).to_pandas()
    e = pdf[["t"]].shift(-3)
    assert g["t"].dtype == e["t"].dtype
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    fv = pandas.Timestamp("1999-12-31 23:59:59.123456789")
    g = df["t"].fillna(fv).to_pandas()
    e = pdf["t"].fillna(fv)
    assert g.dtype == e.dtype
    np.testing.assert_array_equal(g.to_numpy(), e.to_numpy())
    # inner merge on a NaT-bearing datetime key: NaT==NaT matches
    rp = pandas.DataFrame({"t": pandas.concat(
        [t.iloc[:300], pandas.Series([pandas.NaT])],
        ignore_index=True), "b": rng.random(301)})
    g = df.merge(mpd.DataFrame(rp), on="t", how="inner").to_pandas()
    e = pdf.merge(rp, on="t", how="inner")
    assert len(g) == len(e)
    assert g["t"].dtype == e["t"].dtype
    np.testing.assert_allclose(np.sort(g["b"].to_numpy()),
                               np.sort(e["b"].to_numpy()), rtol=0)
    # left merge on the datetime KEY is fine (key never fills; right
    # payload fills NaN)
    g = df.merge(mpd.DataFrame(rp), on="t", how="left").to_pandas()
    e = pdf.merge(rp, on="t", how="left")
    assert len(g) == len(e)
    assert g["t"].dtype == e["t"].dtype
    np.testing.assert_allclose(
        np.sort(g["b"].to_numpy()), np.sort(e["b"].to_numpy()),
        rtol=0, equal_nan=True)
    # ... but a datetime PAYLOAD column that would need NaT fills is loud
    rp2 = rp.copy()
    rp2["d2"] = pandas.Timestamp("2000-01-01")
    with pytest.raises(lib.HfError):
        df.merge(mpd.DataFrame(rp2), on="t", how="left")


def test_series_extras_vs_pandas(npartitions):
    """Series tail/to_frame/astype/quantile/any/all/mode on device."""
    rng = np.random.default_rng(135)
    n = 30_000
    pdf = pandas.DataFrame({"a": rng.integers(0, 50, n),
                            "v": rng.standard_normal(n)})
    pdf.loc[rng.random(n) < 0.1, "v"] = np.nan
    df = mpd.DataFrame(pdf)
    pandas.testing.assert_series_equal(df["v"].tail(9).to_pandas(),
                                       pdf["v"].tail(9))
    assert abs(df["v"].quantile(0.75) - pdf["v"].quantile(0.75)) < 1e-12
    assert df["v"].any() == pdf["v"].any()
    assert df["v"].all() == pdf["v"].all()
    assert (df["v"] > 100).any() == (pdf["v"] > 100).any()
    got, exp = df["a"].mode().to_pandas(), pdf["a"].mode()
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    g = df["a"].astype(np.float64).to_pandas()
    assert g.dtype == np.float64


def test_window_var_std_vs_pandas(npartitions):
    """rolling/expanding var/std composed from the window prefix-scan
    sums — no new kernels."""
    rng = np.random.default_rng(136)
    n = 60_000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.12] = np.nan
    pdf = pandas.DataFrame({"v": v})
    df = mpd.DataFrame(pdf)
    for w_, mp in ((16, None), (7, 3)):
        for op in ("var", "std"):
            g = getattr(df["v"].rolling(w_, min_periods=mp),
                        op)().to_pandas()
            e = getattr(pdf["v"].rolling(w_, min_periods=mp), op)()
            np.testing.assert_allclose(g.to_numpy(), e.to_numpy(),
                                       rtol=1e-7, atol=1e-9,
                                       equal_nan=True,
                                       err_msg=f"{w_}/{mp}/{op}")
    g = df["v"].expanding(2).std().to_pandas()
    e = pdf["v"].expanding(2).std()
    np.testing.assert_allclose(g.to_numpy(), e.to_numpy(), rtol=1e-7,
                               atol=1e-9, equal_nan=True)


def test_duplicated_keep_vs_pandas(npartitions):
    """duplicated/drop_duplicates keep='last'/False via device row
    reversal + mask OR."""
    rng = np.random.default_rng(137)
    n = 50_000
    pdf = pandas.DataFrame({
        "k": rng.integers(0, 900, n),
        "s": rng.choice(["a", "b", "c", None], n),
        "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    for keep in ("first", "last", False):
        for subs in (["k"], ["k", "s"], None):
            g = df.duplicated(subs, keep=keep).to_pandas().to_numpy()
            e = pdf.duplicated(subset=subs, keep=keep).to_numpy()
            np.testing.assert_array_equal(g.astype(bool), e,
                                          err_msg=f"{keep}/{subs}")
            gd = df.drop_duplicates(subs, keep=keep).to_pandas()
            ed = pdf.drop_duplicates(subset=subs, keep=keep)
            assert len(gd) == len(ed), (keep, subs)
            np.testing.assert_array_equal(np.asarray(gd.index),
                                          ed.index.to_numpy())


def test_cut_qcut_vs_pandas(npartitions):
    """cut/qcut: one hf_shuffle_dest pass over ordered edges; codes
    exact (labels=False), Interval labels match pandas' rounded breaks
    for int bins; groupby over the binned column."""
    rng = np.random.default_rng(138)
    n = 100_000
    v = rng.standard_normal(n) * 10
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"v": v, "w": rng.random(n)})
    df = mpd.DataFrame(pdf)
    for bins in (4, 10, [-40.0, -5.0, 0.0, 5.0, 40.0]):
        for right in (True, False):
            g = mpd.cut(df["v"], bins, right=right,
                        labels=False).to_pandas()
            e = pandas.cut(pdf["v"], bins, right=right, labels=False)
            np.testing.assert_allclose(
                g.to_numpy(), e.to_numpy().astype(float), rtol=0,
                equal_nan=True, err_msg=f"{bins}/{right}")
    g = mpd.cut(df["v"], 6).to_pandas()
    e = pandas.cut(pdf["v"], 6).astype(object)
    same = (pandas.isna(g.to_numpy()) & pandas.isna(e.to_numpy())) \
        | (g.to_numpy() == e.to_numpy())
    assert same.all()
    for q in (4, 10):
        g = mpd.qcut(df["v"], q, labels=False).to_pandas()
        e = pandas.qcut(pdf["v"], q, labels=False)
        np.testing.assert_allclose(g.to_numpy(),
                                   e.to_numpy().astype(float), rtol=0,
                                   equal_nan=True, err_msg=str(q))
    df["bin"] = mpd.cut(df["v"], [-40.0, 0.0, 40.0])
    pdf["bin"] = pandas.cut(pdf["v"], [-40.0, 0.0, 40.0])
    got = df[["bin", "w"]].groupby("bin").sum().to_pandas()
    exp = pdf[["bin", "w"]].groupby("bin", observed=True).sum()
    np.testing.assert_allclose(got["w"].to_numpy(), exp["w"].to_numpy(),
                               rtol=1e-12)


def test_set_reset_index_vs_pandas(npartitions):
    """set_index (lazy DeviceIndex, datetime tag kept) and
    reset_index(drop=False)."""
    rng = np.random.default_rng(139)
    n = 40_000
    t = pandas.Series(pandas.to_datetime("2022-01-01")
                      + pandas.to_timedelta(rng.integers(0, 10**5, n),
                                            unit="s"))
    pdf = pandas.DataFrame({"k": rng.integers(0, 900, n),
                            "t": t, "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    got = df.set_index("t").to_pandas()
    exp = pdf.set_index("t")
    assert got.index.dtype == exp.index.dtype
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    assert list(got.columns) == list(exp.columns)
    g2 = df.set_index("k").reset_index().to_pandas()
    e2 = pdf.set_index("k").reset_index()
    assert list(g2.columns) == list(e2.columns)
    np.testing.assert_array_equal(g2["k"].to_numpy(), e2["k"].to_numpy())
    np.testing.assert_allclose(g2["v"].to_numpy(), e2["v"].to_numpy(),
                               rtol=0, equal_nan=True)


def test_multikey_merge_right_vs_pandas(npartitions):
    """merge(on=[a,b], how='right'): swapped-left composition over the
    multi-key fold."""
    rng = np.random.default_rng(142)
    nl, nr = 20_000, 7_000
    lpdf = pandas.DataFrame({
        "a": rng.integers(0, 60, nl),
        "b": (rng.integers(-20, 20, nl) / 4.0),
        "x": rng.standard_normal(nl), "c": rng.integers(0, 5, nl)})
    rpdf = pandas.DataFrame({
        "a": rng.integers(0, 60, nr),
        "b": (rng.integers(-20, 20, nr) / 4.0),
        "y": rng.standard_normal(nr), "c": rng.integers(5, 9, nr)})
    got = mpd.DataFrame(lpdf).merge(mpd.DataFrame(rpdf),
                                    on=["a", "b"], how="right").to_pandas()
    exp = lpdf.merge(rpdf, on=["a", "b"], how="right")
    assert list(got.columns) == list(exp.columns)
    assert len(got) == len(exp)
    order = ["a", "b", "x", "y", "c_x", "c_y"]
    gs = got.sort_values(order, na_position="last").reset_index(drop=True)
    es = exp.sort_values(order, na_position="last").reset_index(drop=True)
    for c in exp.columns:
        np.testing.assert_allclose(gs[c].to_numpy().astype(float),
                                   es[c].to_numpy().astype(float),
                                   rtol=0, equal_nan=True, err_msg=c)


def test_named_agg_vs_pandas(npartitions):
    rng = np.random.default_rng(144)
    n = 50_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 500, n),
                            "v": rng.standard_normal(n),
                            "w": rng.integers(-9, 9, n)})
    df = mpd.DataFrame(pdf)
    got = df.groupby("k").agg(total=("v", "sum"), hi=("w", "max"),
                              m=("v", "mean")).to_pandas()
    exp = pdf.groupby("k").agg(total=("v", "sum"), hi=("w", "max"),
                               m=("v", "mean"))
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy().astype(float),
                                   exp[c].to_numpy().astype(float),
                                   rtol=1e-12, err_msg=c)


def test_dropna_how_subset_vc_normalize(npartitions):
    rng = np.random.default_rng(145)
    n = 40_000
    pdf = pandas.DataFrame({"a": rng.standard_normal(n),
                            "b": rng.standard_normal(n),
                            "k": rng.integers(0, 50, n)})
    pdf.loc[rng.random(n) < 0.3, "a"] = np.nan
    pdf.loc[rng.random(n) < 0.3, "b"] = np.nan
    df = mpd.DataFrame(pdf)
    for how in ("any", "all"):
        for subset in (None, ["a"], ["a", "b"]):
            g = df.dropna(how=how, subset=subset).to_pandas()
            e = pdf.dropna(how=how, subset=subset)
            assert len(g) == len(e), (how, subset)
            np.testing.assert_array_equal(np.asarray(g.index),
                                          e.index.to_numpy(),
                                          err_msg=f"{how}/{subset}")
    g = df["k"].value_counts(normalize=True)
    e = pdf["k"].value_counts(normalize=True)
    np.testing.assert_array_equal(np.asarray(g.index).astype(np.int64),
                                  e.index.to_numpy())
    np.testing.assert_allclose(np.asarray(g), e.to_numpy(), rtol=1e-12)


def test_query_vs_pandas(npartitions):
    """query: mask-algebra composition from parsed expressions
    (comparisons, col-col via subtract, and/or/not, strings)."""
    rng = np.random.default_rng(146)
    n = 60_000
    pdf = pandas.DataFrame({"a": rng.integers(0, 100, n),
                            "b": rng.integers(0, 100, n),
                            "v": rng.standard_normal(n),
                            "s": rng.choice(["x", "y", "z"], n)})
    pdf.loc[rng.random(n) < 0.1, "v"] = np.nan
    df = mpd.DataFrame(pdf)
    for expr in ("a > 50", "v <= 0.5 and a != 3", "a > b",
                 "(a > 5 or b < 2) and not v > 0", "s != 'y'"):
        g = df.query(expr).to_pandas()
        e = pdf.query(expr)
        assert len(g) == len(e), expr
        np.testing.assert_array_equal(np.asarray(g.index),
                                      e.index.to_numpy(), err_msg=expr)


def test_filter_selectdtypes_dtfloor(npartitions):
    rng = np.random.default_rng(147)
    n = 30_000
    t = pandas.Series(pandas.to_datetime("1965-01-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**7, n), unit="min"))
    t[rng.random(n) < 0.1] = pandas.NaT
    pdf = pandas.DataFrame({"aa": rng.integers(0, 5, n),
                            "ab": rng.standard_normal(n), "t": t})
    df = mpd.DataFrame(pdf)
    assert list(df.filter(like="a").columns) == ["aa", "ab"]
    assert list(df.select_dtypes(include="number").columns) == \
        list(pdf.select_dtypes(include="number").columns)
    for freq in ("D", "h", "min", "s"):
        g = df["t"].dt.floor(freq).to_pandas()
        e = pdf["t"].dt.floor(freq)
        assert g.dtype == e.dtype
        np.testing.assert_array_equal(g.to_numpy(), e.to_numpy(),
                                      err_msg=freq)
    # normalized column groups by calendar day
    d2 = mpd.DataFrame(query_compiler=df["t"].dt.normalize()
                       ._query_compiler)
    d2["v"] = df["ab"]
    g = d2.groupby("t").count().to_pandas()
    p2 = pandas.DataFrame({"t": pdf["t"].dt.normalize(),
                           "v": pdf["ab"]})
    e = p2.groupby("t").count()
    np.testing.assert_array_equal(g.index.to_numpy(), e.index.to_numpy())
    np.testing.assert_array_equal(g["v"].to_numpy(), e["v"].to_numpy())


def test_where_datetime_nat_fill(npartitions):
    """where/mask over datetime columns: NaT (or Timestamp) fills via an
    int64 blend, dtype kept."""
    rng = np.random.default_rng(149)
    n = 30_000
    t = pandas.Series(pandas.to_datetime("2020-01-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**6, n), unit="min"))
    t[rng.random(n) < 0.1] = pandas.NaT
    pdf = pandas.DataFrame({"t": t, "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    m, pm = df["v"] > 0, pdf["v"] > 0
    g = df[["t"]].where(m).to_pandas()
    e = pdf[["t"]].where(pm)
    assert g["t"].dtype == e["t"].dtype
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    fv = pandas.Timestamp("1999-01-01 03:04:05.000000006")
    g = df[["t"]].mask(m, fv).to_pandas()
    e = pdf[["t"]].mask(pm, fv)
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    with pytest.raises(lib.HfError):
        df[["v"]].where(m, fv)


def test_concat_axis1_numeric_only(npartitions):
    rng = np.random.default_rng(150)
    n = 30_000
    p1 = pandas.DataFrame({"a": rng.integers(0, 90, n),
                           "s": rng.choice(["x", "y"], n)})
    p2 = pandas.DataFrame({"b": rng.standard_normal(n)})
    got = mpd.concat([mpd.DataFrame(p1), mpd.DataFrame(p2)],
                     axis=1).to_pandas()
    exp = pandas.concat([p1, p2], axis=1)
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_allclose(got["b"].to_numpy(), exp["b"].to_numpy(),
                               rtol=0)
    df = mpd.DataFrame(p1)
    g = df.sum(numeric_only=True)
    e = p1.sum(numeric_only=True)
    np.testing.assert_allclose(np.asarray(g),
                               e.to_numpy().astype(float), rtol=0)


def test_sem_vs_pandas(npartitions):
    rng = np.random.default_rng(151)
    n = 40_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 300, n),
                            "v": rng.standard_normal(n),
                            "w": rng.standard_normal(n)})
    pdf.loc[rng.random(n) < 0.1, "v"] = np.nan
    df = mpd.DataFrame(pdf)
    np.testing.assert_allclose(np.asarray(df.sem()),
                               pdf.sem().to_numpy(), rtol=1e-10)
    g = df.groupby("k").sem().to_pandas()
    e = pdf.groupby("k").sem()
    for c in e.columns:
        np.testing.assert_allclose(g[c].to_numpy(), e[c].to_numpy(),
                                   rtol=1e-9, equal_nan=True, err_msg=c)
