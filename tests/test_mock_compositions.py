"""CPU tier: the Python COMPOSITION layer over the numpy lib mock.

These run the REAL modin_amd/core/dataframe.py composition code (the
sequences of gather/scatter/scan/filter calls behind groupby transforms,
rank, where/mask/round, sort and dedup) against pandas, with
tests/mocklib.py standing in for the HIP kernels.  Kernel parity stays on
the GPU tier; this tier catches composition regressions (metadata,
ordering, NaN bookkeeping) before any gpurun call is spent.  Paths whose
kernels are NOT mocked (groupby aggregations, merge) are exercised on the
GPU tier only.
"""

import numpy as np
import pandas
import pytest

from tests import mocklib
from modin_amd.core.lib import HfError as _HfErr


@pytest.fixture()
def mlib(monkeypatch):
    mocklib.install(monkeypatch)
    import modin_amd.pandas as mpd
    yield mpd


def _frames(rng, n=4000, nan_keys=True):
    k = rng.integers(0, 40, n).astype(np.float64)
    if nan_keys:
        k[rng.random(n) < 0.05] = np.nan
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.integers(-30, 30, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    return pdf


def test_mock_groupby_transforms(mlib):
    rng = np.random.default_rng(41)
    pdf = _frames(rng)
    df = mlib.DataFrame(pdf)
    for how in ("cumsum", "cummin", "cummax"):
        got = getattr(df.groupby("k"), how)().to_pandas()
        exp = getattr(pdf.groupby("k"), how)()
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=1e-12,
                                       atol=1e-9, equal_nan=True,
                                       err_msg=f"{how}/{c}")
    np.testing.assert_allclose(
        df.groupby("k").cumcount().to_pandas().to_numpy(),
        pdf.groupby("k").cumcount().to_numpy(), rtol=0, equal_nan=True)
    np.testing.assert_allclose(
        df.groupby("k").ngroup().to_pandas().to_numpy(),
        pdf.groupby("k").ngroup().to_numpy().astype(float), rtol=0,
        equal_nan=True)
    for p in (1, -2):
        got = df.groupby("k").shift(p).to_pandas()
        exp = pdf.groupby("k").shift(p)
        for c in exp.columns:
            np.testing.assert_allclose(got[c].to_numpy(),
                                       exp[c].to_numpy(), rtol=0,
                                       equal_nan=True,
                                       err_msg=f"shift({p})/{c}")
    for agg in ("sum", "mean", "count", "min", "max"):
        got = df.groupby("k").transform(agg).to_pandas()
        exp = pdf.groupby("k").transform(agg)
        for c in exp.columns:
            np.testing.assert_allclose(
                got[c].to_numpy().astype(float),
                exp[c].to_numpy().astype(float), rtol=1e-12, atol=1e-9,
                equal_nan=True, err_msg=f"t-{agg}/{c}")


def test_mock_rank_and_frame_rank(mlib):
    rng = np.random.default_rng(43)
    pdf = _frames(rng, 3000)
    df = mlib.DataFrame(pdf)
    for method in ("average", "min", "first"):
        for asc in (True, False):
            got = df.groupby("k").rank(method=method,
                                       ascending=asc).to_pandas()
            exp = pdf.groupby("k").rank(method=method, ascending=asc)
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=0,
                    equal_nan=True, err_msg=f"{method}/{asc}/{c}")
    got = df.rank().to_pandas()
    exp = pdf.rank()
    for c in ("v", "w"):
        np.testing.assert_allclose(got[c].to_numpy(), exp[c].to_numpy(),
                                   rtol=0, equal_nan=True, err_msg=c)


def test_mock_where_round_sort_dedup(mlib):
    rng = np.random.default_rng(47)
    pdf = _frames(rng, 3000, nan_keys=False)
    df = mlib.DataFrame(pdf)
    cond_p = pdf["v"] > 0
    got = df.where(df["v"] > 0, -2.5).to_pandas()
    exp = pdf.where(cond_p, -2.5)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    got = df.mask(df["v"] > 0).to_pandas()
    exp = pdf.mask(cond_p)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    for d in (0, 2, -1):
        got = df.round(d).to_pandas()
        exp = pdf.round(d)
        np.testing.assert_allclose(got["v"].to_numpy(),
                                   exp["v"].to_numpy(), rtol=0,
                                   equal_nan=True, err_msg=f"round{d}")
        np.testing.assert_array_equal(got["w"].to_numpy(),
                                      exp["w"].to_numpy())
    for by, asc, napos in (("w", True, "last"), ("v", False, "last"),
                           ("v", True, "first")):
        got = df.sort_values(by, ascending=asc,
                             na_position=napos).to_pandas()
        exp = pdf.sort_values(by, ascending=asc, kind="stable",
                              na_position=napos)
        np.testing.assert_array_equal(got.index.to_numpy(),
                                      exp.index.to_numpy(),
                                      err_msg=f"{by}/{asc}/{napos}")
    got = df.duplicated(["w"]).to_pandas()
    exp = pdf.duplicated(subset=["w"])
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    got = df.drop_duplicates(["w"]).to_pandas()
    exp = pdf.drop_duplicates(subset=["w"])
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())


def test_mock_ffill_bfill(mlib):
    rng = np.random.default_rng(53)
    pdf = _frames(rng, 3000)
    df = mlib.DataFrame(pdf)
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
    s_ = df.groupby("k")["v"].ffill().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(),
                               pdf.groupby("k")["v"].ffill().to_numpy(),
                               rtol=0, equal_nan=True)


def test_mock_groupby_pct_change(mlib):
    rng = np.random.default_rng(59)
    pdf = _frames(rng, 2500)
    df = mlib.DataFrame(pdf)
    for p in (1, 2):
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


def test_mock_fillna_dict_replace(mlib):
    rng = np.random.default_rng(61)
    pdf = _frames(rng, 2000, nan_keys=False)
    df = mlib.DataFrame(pdf)
    got = df.fillna({"v": -1.5}).to_pandas()
    exp = pdf.fillna({"v": -1.5})
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(), exp[c].to_numpy(),
                                   rtol=0, equal_nan=True, err_msg=c)
    s = pdf["w"].astype(np.float64)
    s[rng.random(2000) < 0.1] = np.nan
    pdf2 = pandas.DataFrame({"x": s})
    df2 = mlib.DataFrame(pdf2)
    got = df2["x"].replace(7.0, -99.0).to_pandas()
    exp = pdf2["x"].replace(7.0, -99.0)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0,
                               equal_nan=True)
    got = df2["x"].replace(np.nan, 0.5).to_pandas()
    exp = pdf2["x"].replace(np.nan, 0.5)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0)


def test_mock_rolling(mlib):
    rng = np.random.default_rng(67)
    n = 3000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.15] = np.nan
    w = rng.integers(-20, 20, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mlib.DataFrame(pdf)
    for win, mp in ((1, None), (3, None), (3, 1), (16, 4), (100, None),
                    (7, 7)):
        for op in ("sum", "mean", "count", "min", "max"):
            got = getattr(df.rolling(win, min_periods=mp), op)() \
                .to_pandas()
            exp = getattr(pdf.rolling(win, min_periods=mp), op)()
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=1e-12,
                    atol=1e-12, equal_nan=True,
                    err_msg=f"{op}/w={win}/mp={mp}/{c}")
    s_ = df["v"].rolling(5).mean().to_pandas()
    np.testing.assert_allclose(s_.to_numpy(),
                               pdf["v"].rolling(5).mean().to_numpy(),
                               rtol=1e-12, atol=1e-12, equal_nan=True)


def test_mock_expanding(mlib):
    rng = np.random.default_rng(71)
    n = 2000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.2] = np.nan
    w = rng.integers(-9, 9, n)
    pdf = pandas.DataFrame({"v": v, "w": w})
    df = mlib.DataFrame(pdf)
    for mp in (1, 3, 50):
        for op in ("sum", "mean", "count", "min", "max"):
            got = getattr(df.expanding(mp), op)().to_pandas()
            exp = getattr(pdf.expanding(mp), op)()
            for c in exp.columns:
                np.testing.assert_allclose(
                    got[c].to_numpy(), exp[c].to_numpy(), rtol=1e-12,
                    atol=1e-12, equal_nan=True,
                    err_msg=f"{op}/mp={mp}/{c}")


def test_mock_cumprod(mlib):
    """groupby.cumprod + frame cumprod (product scan, AGG_PROD): NaN values
    stay NaN and do not advance the running product (pandas semantics)."""
    rng = np.random.default_rng(99)
    pdf = _frames(rng, n=3000)
    # keep magnitudes tame so f64 products stay finite
    pdf["v"] = np.clip(pdf["v"], -1.5, 1.5)
    pdf["w"] = (pdf["w"] % 3) - 1
    df = mlib.DataFrame(pdf)
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


def test_mock_rank_na_option(mlib):
    """rank na_option='top'/'bottom': NaN values take the lowest/highest
    shared ranks (one tie run at the NaN sentinel) — vs pandas for every
    (method, ascending) combination, frame and groupby forms."""
    rng = np.random.default_rng(7)
    pdf = _frames(rng, n=2500)
    df = mlib.DataFrame(pdf)
    for na in ("top", "bottom", "keep"):
        for method in ("average", "min", "first"):
            for asc in (True, False):
                got = df.groupby("k").rank(
                    method=method, ascending=asc,
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


def test_mock_unbounded_multikey_groupby(mlib):
    """Multi-key groupby whose combined span exceeds 2^62 takes the
    sorted-heads dense-rank fold (no tuple hash) — vs pandas."""
    rng = np.random.default_rng(17)
    n = 5000
    k1 = rng.integers(-2**61, 2**61, n)
    k1[rng.random(n) < 0.3] = 77  # heavy dup runs
    k2 = rng.integers(-2**61, 2**61, n)
    k2[rng.random(n) < 0.3] = -5
    v = rng.standard_normal(n)
    pdf = pandas.DataFrame({"a": k1, "b": k2, "v": v})
    df = mlib.DataFrame(pdf)
    for agg in ("sum", "count", "mean"):
        got = getattr(df.groupby(["a", "b"]), agg)().to_pandas()
        exp = getattr(pdf.groupby(["a", "b"]), agg)()
        assert list(got.index) == list(exp.index), f"{agg} keys"
        np.testing.assert_allclose(got["v"].to_numpy(),
                                   exp["v"].to_numpy(), rtol=1e-12,
                                   err_msg=agg)


def test_mock_multikey_idx_nan_keys(mlib):
    """Multi-key idxmax/idxmin with NaN keys: filter-first composition
    (pinned prototype lifted) vs pandas."""
    rng = np.random.default_rng(3)
    n = 8000
    a = rng.choice(["x", "y", "z", None], n,
                   p=[0.3, 0.3, 0.3, 0.1]).astype(object)
    b = rng.integers(0, 10, n)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    df = mlib.DataFrame(pdf)
    for mx in (True, False):
        got = (df.groupby(["a", "b"]).idxmax() if mx
               else df.groupby(["a", "b"]).idxmin()).to_pandas()
        exp = (pdf.groupby(["a", "b"]).idxmax() if mx
               else pdf.groupby(["a", "b"]).idxmin())
        assert list(got.index) == list(exp.index)
        np.testing.assert_allclose(got["v"].to_numpy().astype(float),
                                   exp["v"].to_numpy().astype(float),
                                   rtol=0, equal_nan=True)


def test_mock_multikey_dropna_false(mlib):
    """Multi-key groupby(dropna=False): NaN string keys are an extra
    top-of-code-space level per column (sorts last per level)."""
    rng = np.random.default_rng(8)
    n = 9000
    a = rng.choice(["x", "y", None], n, p=[0.45, 0.45, 0.1]).astype(object)
    b = rng.integers(0, 6, n)
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    df = mlib.DataFrame(pdf)
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


def test_mock_dt_accessor(mlib):
    """Series.dt fields: exact int64 civil-calendar math vs pandas
    (incl. pre-1970 dates — negative ns exercise floor div/mod)."""
    rng = np.random.default_rng(5)
    n = 4000
    ns = rng.integers(-2 * 10**18, 2 * 10**18, n)
    t = pandas.Series(ns.astype("datetime64[ns]"), name="t")
    pdf = pandas.DataFrame({"t": t})
    df = mlib.DataFrame(pdf)
    for f in ("year", "month", "day", "hour", "minute", "second",
              "dayofweek"):
        got = getattr(df["t"].dt, f).to_pandas()
        exp = getattr(t.dt, f)
        np.testing.assert_array_equal(np.asarray(got), exp.to_numpy(),
                                      err_msg=f)


def test_mock_str_accessor(mlib):
    """Series.str len/lower/upper/contains/startswith/endswith: host
    dictionary transform + device gather, vs pandas."""
    rng = np.random.default_rng(9)
    n = 4000
    words = np.array(["Apple", "beta", "Ba", "apple", "CAT", "ca t"],
                     dtype=object)
    sv = words[rng.integers(0, len(words), n)]
    sv[rng.random(n) < 0.1] = np.nan
    t = pandas.Series(sv, name="s")
    df = mlib.DataFrame(pandas.DataFrame({"s": t}))
    got = df["s"].str.len().to_pandas()
    np.testing.assert_allclose(np.asarray(got, dtype=float),
                               t.str.len().to_numpy(dtype=float),
                               rtol=0, equal_nan=True)
    for op in ("lower", "upper"):
        got = getattr(df["s"].str, op)().to_pandas()
        exp = getattr(t.str, op)()
        same = (pandas.isna(np.asarray(got)) & pandas.isna(exp).to_numpy()
                ) | (np.asarray(got) == exp.to_numpy())
        assert same.all(), op
    for op, pat in (("contains", "a"), ("startswith", "a"),
                    ("endswith", "t")):
        got = getattr(df["s"].str, op)(pat, na=False).to_pandas()
        exp = getattr(t.str, op)(pat, na=False)
        np.testing.assert_array_equal(
            np.asarray(got, dtype=bool), exp.to_numpy(dtype=bool),
            err_msg=op)
        got = getattr(df["s"].str, op)(pat).to_pandas()
        exp2 = getattr(t.str, op)(pat)
        np.testing.assert_allclose(
            np.asarray(got, dtype=float), exp2.to_numpy(dtype=float),
            rtol=0, equal_nan=True, err_msg=f"{op}/naNone")


def test_mock_groupby_prod(mlib):
    """groupby.prod: segmented PROD scan + last-valid pick; int64 stays
    int64 (wrapping), NaN skipped, all-NaN groups give 1.0."""
    rng = np.random.default_rng(12)
    pdf = _frames(rng, n=3000)
    pdf["v"] = np.clip(pdf["v"], -1.4, 1.4)
    pdf["w"] = (pdf["w"] % 3) - 1
    df = mlib.DataFrame(pdf)
    got = df.groupby("k").prod().to_pandas()
    exp = pdf.groupby("k").prod()
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy(dtype=float),
                                   exp[c].to_numpy(dtype=float),
                                   rtol=1e-12, atol=1e-300,
                                   err_msg=f"prod/{c}")
    assert list(got.dtypes) == list(exp.dtypes)


def test_mock_iloc(mlib):
    rng = np.random.default_rng(2)
    pdf = pandas.DataFrame({"k": rng.integers(0, 9, 500),
                            "v": rng.random(500)})
    df = mlib.DataFrame(pdf)
    pandas.testing.assert_frame_equal(df.iloc[17:200].to_pandas(),
                                      pdf.iloc[17:200])
    sel = [3, 499, 0, -2, 77]
    pandas.testing.assert_frame_equal(df.iloc[sel].to_pandas(),
                                      pdf.iloc[sel])
    assert df.iloc[42]["k"] == pdf.iloc[42]["k"]
    assert abs(df["v"].iloc[-1] - pdf["v"].iloc[-1]) < 1e-15


def test_mock_sort_index(mlib):
    rng = np.random.default_rng(4)
    pdf = pandas.DataFrame({"v": rng.random(300)},
                           index=rng.integers(0, 40, 300))
    df = mlib.DataFrame(pdf)
    for asc in (True, False):
        got = df.sort_index(ascending=asc).to_pandas()
        pandas.testing.assert_frame_equal(got,
                                          pdf.sort_index(ascending=asc))


def test_mock_loc(mlib):
    rng = np.random.default_rng(6)
    pdf = pandas.DataFrame({"k": rng.integers(0, 9, 400),
                            "v": rng.random(400),
                            "w": rng.integers(-5, 5, 400)})
    df = mlib.DataFrame(pdf)
    m = pdf["v"] > 0.5
    got = df.loc[df["v"] > 0.5].to_pandas()
    pandas.testing.assert_frame_equal(got, pdf.loc[m])
    got = df.loc[df["v"] > 0.5, ["k", "w"]].to_pandas()
    pandas.testing.assert_frame_equal(got, pdf.loc[m, ["k", "w"]])
    got = df.loc[10:20].to_pandas()
    pandas.testing.assert_frame_equal(got, pdf.loc[10:20])
    got = df.loc[:, ["v"]].to_pandas()
    pandas.testing.assert_frame_equal(got, pdf.loc[:, ["v"]])


def test_mock_multikey_merge(mlib):
    """merge(on=[a,b]) inner/left: shared arithmetic fold over both
    sides; float keys canonical-NaN match; string key recode."""
    rng = np.random.default_rng(21)
    nl, nr = 6000, 2500
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
            got = mlib.DataFrame(lpdf).merge(
                mlib.DataFrame(rpdf), on=keys, how=how).to_pandas()
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


def test_mock_merge_left_on_right_on(mlib):
    """merge(left_on=, right_on=): both key columns survive (pandas
    keep-keys rule), suffixes over every other collision, NaN fill on
    the unmatched side's key for left/right/outer."""
    rng = np.random.default_rng(22)
    nl, nr = 5000, 1800
    lpdf = pandas.DataFrame({
        "a": (rng.integers(0, 60, nl)).astype(np.float64),
        "v": rng.random(nl),
        "c": rng.integers(0, 9, nl)})
    lpdf.loc[rng.random(nl) < 0.05, "a"] = np.nan
    rpdf = pandas.DataFrame({
        "b": (rng.integers(0, 80, nr)).astype(np.float64),
        "w": rng.random(nr),
        "c": rng.integers(10, 19, nr)})
    rpdf.loc[rng.random(nr) < 0.05, "b"] = np.nan
    for how in ("inner", "left", "right", "outer"):
        got = mlib.DataFrame(lpdf).merge(
            mlib.DataFrame(rpdf), left_on="a", right_on="b",
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
    # same-named keys collapse to the `on` form (one key column)
    got = mlib.DataFrame(lpdf).merge(
        mlib.DataFrame(rpdf.rename(columns={"b": "a"})),
        left_on="a", right_on="a", how="inner").to_pandas()
    exp = lpdf.merge(rpdf.rename(columns={"b": "a"}), on="a", how="inner")
    assert list(got.columns) == list(exp.columns)
    assert len(got) == len(exp)


def test_mock_setitem_insert_assign(mlib):
    """df[c] = series/array/scalar, insert(loc), assign: replace-or-
    append positional assignment; scalar broadcast; multi-partition
    value re-slice."""
    rng = np.random.default_rng(23)
    n = 3000
    pdf = pandas.DataFrame({"a": rng.integers(0, 50, n),
                            "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    exp = pdf.copy()
    # derived-column assignment (replace + append)
    df["v"] = df["v"] * 2.0
    exp["v"] = exp["v"] * 2.0
    df["w"] = df["a"] + df["v"]
    exp["w"] = exp["a"] + exp["v"]
    # scalar broadcasts (int / float / NaN / string)
    df["k"] = 7
    exp["k"] = 7
    df["f"] = 2.5
    exp["f"] = 2.5
    df["nn"] = np.nan
    exp["nn"] = np.nan
    df["s"] = "hello"
    exp["s"] = "hello"
    # host array / pandas Series values
    arr = rng.standard_normal(n)
    df["h"] = arr
    exp["h"] = arr
    ps = pandas.Series(rng.integers(-9, 9, n))
    df["i"] = ps
    exp["i"] = ps
    got = df.to_pandas()
    assert list(got.columns) == list(exp.columns)
    for c in exp.columns:
        g, e = got[c].to_numpy(), exp[c].to_numpy()
        if e.dtype == object:
            np.testing.assert_array_equal(g, e, err_msg=c)
        else:
            np.testing.assert_allclose(g.astype(float), e.astype(float),
                                       rtol=0, equal_nan=True, err_msg=c)
    # insert at a position
    df2 = mlib.DataFrame(pdf)
    exp2 = pdf.copy()
    df2.insert(1, "z", df2["a"] * 10)
    exp2.insert(1, "z", exp2["a"] * 10)
    got2 = df2.to_pandas()
    assert list(got2.columns) == list(exp2.columns)
    np.testing.assert_allclose(got2["z"].to_numpy(),
                               exp2["z"].to_numpy(), rtol=0)
    with pytest.raises(_HfErr):
        df2.insert(0, "z", 1)  # duplicate
    # assign (incl. callable) leaves the original untouched
    df3 = mlib.DataFrame(pdf)
    out = df3.assign(q=lambda d: d["v"] - 1.0, r=5)
    assert list(df3.columns) == list(pdf.columns)
    expq = pdf.assign(q=lambda d: d["v"] - 1.0, r=5)
    gotq = out.to_pandas()
    np.testing.assert_allclose(gotq["q"].to_numpy(),
                               expq["q"].to_numpy(), rtol=1e-15)
    np.testing.assert_array_equal(gotq["r"].to_numpy(),
                                  expq["r"].to_numpy())
    # length mismatch is loud
    with pytest.raises(_HfErr):
        df3["bad"] = np.zeros(n - 1)


def test_mock_series_map_replace_dict(mlib):
    """Series.map(dict) / replace(dict): int64 LUT via device binary
    search; string columns remap on the host dictionary."""
    rng = np.random.default_rng(24)
    n = 4000
    pdf = pandas.DataFrame({"i": rng.integers(0, 10, n),
                            "s": rng.choice(["a", "b", "c", None], n)})
    df = mlib.DataFrame(pdf)
    # int map, full coverage + int values -> int64
    full = {k: k * 10 for k in range(10)}
    got = df["i"].map(full).to_pandas()
    exp = pdf["i"].map(full)
    assert got.dtype == exp.dtype == np.int64
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    # int map, partial coverage -> float64 + NaN
    part = {1: 100, 3: 300.5}
    got = df["i"].map(part).to_pandas()
    exp = pdf["i"].map(part)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy().astype(float),
                               rtol=0, equal_nan=True)
    # int replace: unmapped keep their value, int64 preserved
    got = df["i"].replace({2: -2, 5: -5}).to_pandas()
    exp = pdf["i"].replace({2: -2, 5: -5})
    assert got.dtype == exp.dtype
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    # int replace with float values -> float64
    got = df["i"].replace({2: 2.5}).to_pandas()
    exp = pdf["i"].replace({2: 2.5})
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy().astype(float),
                               rtol=0)
    # string map -> string (missing cat -> NaN)
    got = df["s"].map({"a": "X", "b": "Y"}).to_pandas()
    exp = pdf["s"].map({"a": "X", "b": "Y"})
    ge = got.to_numpy(), exp.to_numpy()
    same = (pandas.isna(ge[0]) & pandas.isna(ge[1])) | (ge[0] == ge[1])
    assert same.all()
    # string map -> numeric
    got = df["s"].map({"a": 1, "b": 2, "c": 3}).to_pandas()
    exp = pdf["s"].map({"a": 1, "b": 2, "c": 3})
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy().astype(float),
                               rtol=0, equal_nan=True)
    # string replace: unmapped strings keep their value, NaN stays NaN
    got = df["s"].replace({"a": "Q"}).to_pandas()
    exp = pdf["s"].replace({"a": "Q"})
    g, e = got.to_numpy(), exp.to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()
    # module-level merge form
    l = mlib.DataFrame({"k": [1, 2], "v": [1.0, 2.0]})
    r = mlib.DataFrame({"k": [2, 3], "w": [5.0, 6.0]})
    assert len(mlib.merge(l, r, on="k")) == 1


def test_mock_melt_pivot_table(mlib):
    """melt (device-wide->long) and pivot_table (device groupby +
    host reshape of the reduced table)."""
    rng = np.random.default_rng(25)
    n = 2000
    pdf = pandas.DataFrame({
        "id": rng.integers(0, 6, n),
        "g": rng.choice(["r", "s", "t"], n),
        "x": rng.standard_normal(n),
        "y": rng.integers(-5, 5, n),
        "z": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    got = df.melt(id_vars=["id", "g"]).to_pandas()
    exp = pdf.melt(id_vars=["id", "g"])
    assert list(got.columns) == list(exp.columns)
    assert len(got) == len(exp)
    for c in exp.columns:
        g, e = got[c].to_numpy(), exp[c].to_numpy()
        if e.dtype == object:
            np.testing.assert_array_equal(g, e, err_msg=c)
        else:
            np.testing.assert_allclose(g.astype(float),
                                       e.astype(float), rtol=0,
                                       equal_nan=True, err_msg=c)
    # subset value_vars, custom names, single value dtype (no cast)
    got = df.melt(id_vars="id", value_vars=["x", "z"], var_name="V",
                  value_name="W").to_pandas()
    exp = pdf.melt(id_vars="id", value_vars=["x", "z"], var_name="V",
                   value_name="W")
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_allclose(got["W"].to_numpy(), exp["W"].to_numpy(),
                               rtol=0)
    np.testing.assert_array_equal(got["V"].to_numpy(),
                                  exp["V"].to_numpy())
    # pivot_table
    for aggfunc in ("mean", "sum", "count"):
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


def test_mock_sample(mlib):
    """sample(n/frac): exact count, no duplicates, rows drawn from the
    original (values + index labels consistent), seeded reproducible."""
    rng = np.random.default_rng(26)
    n = 5000
    pdf = pandas.DataFrame({"a": np.arange(n),
                            "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    got = df.sample(n=500, random_state=7).to_pandas()
    assert len(got) == 500
    assert got["a"].is_unique
    # each sampled row matches the original at its label
    np.testing.assert_allclose(
        got["v"].to_numpy(), pdf.loc[got["a"].to_numpy(), "v"].to_numpy(),
        rtol=0)
    np.testing.assert_array_equal(np.asarray(got.index),
                                  got["a"].to_numpy())
    got2 = df.sample(n=500, random_state=7).to_pandas()
    np.testing.assert_array_equal(got["a"].to_numpy(),
                                  got2["a"].to_numpy())
    assert len(df.sample(frac=0.25)) == n // 4
    with pytest.raises(_HfErr):
        df.sample(n=100, frac=0.5)
    with pytest.raises(_HfErr):
        df.sample(n=n + 1)


def test_mock_floordiv_mod(mlib):
    rng = np.random.default_rng(27)
    pdf = pandas.DataFrame({"a": rng.integers(-100, 100, 3000)})
    df = mlib.DataFrame(pdf)
    for k in (7, -7, 3):
        np.testing.assert_array_equal(
            (df["a"] // k).to_pandas().to_numpy(),
            (pdf["a"] // k).to_numpy(), err_msg=f"//{k}")
        np.testing.assert_array_equal(
            (df["a"] % k).to_pandas().to_numpy(),
            (pdf["a"] % k).to_numpy(), err_msg=f"%{k}")
    with pytest.raises(_HfErr):
        df["a"] // 0
    with pytest.raises(_HfErr):
        df["a"] // 2.5


def test_mock_corr_cov(mlib):
    rng = np.random.default_rng(28)
    n = 4000
    pdf = pandas.DataFrame({"x": rng.standard_normal(n),
                            "y": rng.standard_normal(n),
                            "w": rng.integers(-5, 5, n)})
    pdf["y"] += 0.5 * pdf["x"]
    pdf.loc[rng.random(n) < 0.1, "x"] = np.nan
    pdf.loc[rng.random(n) < 0.1, "y"] = np.nan
    df = mlib.DataFrame(pdf)
    got, exp = df.corr(), pdf.corr()
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(),
                               rtol=1e-10, equal_nan=True)
    got, exp = df.cov(), pdf.cov()
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(),
                               rtol=1e-10, equal_nan=True)


def test_mock_concat_column_alignment(mlib):
    """concat over frames with DIFFERENT columns: NaN-fill alignment,
    appearance-order column union, int64 -> float64 promotion where a
    column is missing anywhere (pandas rules)."""
    rng = np.random.default_rng(29)
    p1 = pandas.DataFrame({"a": rng.integers(0, 9, 500),
                           "v": rng.random(500),
                           "s": rng.choice(["x", "y"], 500)})
    p2 = pandas.DataFrame({"a": rng.integers(0, 9, 300),
                           "w": rng.integers(-5, 5, 300)})
    p3 = pandas.DataFrame({"w": rng.integers(-5, 5, 200),
                           "s": rng.choice(["y", "z"], 200),
                           "a": rng.integers(0, 9, 200)})
    got = mlib.concat([mlib.DataFrame(p) for p in (p1, p2, p3)],
                      ignore_index=True).to_pandas()
    exp = pandas.concat([p1, p2, p3], ignore_index=True)
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
    # identical columns keep the fast path + int dtype
    got2 = mlib.concat([mlib.DataFrame(p2), mlib.DataFrame(p2)],
                       ignore_index=True).to_pandas()
    exp2 = pandas.concat([p2, p2], ignore_index=True)
    assert list(got2.dtypes) == list(exp2.dtypes)
    np.testing.assert_array_equal(got2.to_numpy(), exp2.to_numpy())


def test_mock_nat_semantics(mlib):
    """NaT through the typed-column layer: iNaT bits on device; masks,
    compares, dt fields, sort na_position, groupby-key drop, shift NaT
    fill, fillna(Timestamp) — pandas semantics; unsupported ops loud."""
    rng = np.random.default_rng(30)
    n = 3000
    t = pandas.Series(pandas.to_datetime("2021-03-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**6, n), unit="min"))
    t[rng.random(n) < 0.15] = pandas.NaT
    pdf = pandas.DataFrame({"t": t, "v": rng.standard_normal(n),
                            "k": rng.integers(0, 7, n)})
    df = mlib.DataFrame(pdf)
    # round trip
    back = df.to_pandas()
    np.testing.assert_array_equal(back["t"].to_numpy(),
                                  pdf["t"].to_numpy())
    # isna / notna / dropna
    np.testing.assert_array_equal(
        df["t"].isna().to_pandas().to_numpy(),
        pdf["t"].isna().to_numpy())
    got = df.dropna().to_pandas()
    exp = pdf.dropna()
    assert len(got) == len(exp)
    np.testing.assert_array_equal(got["t"].to_numpy(),
                                  exp["t"].to_numpy())
    # ordered compares are False on NaT rows; NE is True
    ts = pandas.Timestamp("2021-06-01")
    for op in ("__gt__", "__le__", "__eq__", "__ne__"):
        g = getattr(df["t"], op)(ts).to_pandas().to_numpy()
        e = getattr(pdf["t"], op)(ts).to_numpy()
        np.testing.assert_array_equal(g.astype(bool), e, err_msg=op)
    # dt fields: NaT -> NaN, float64 (pandas rule)
    for f in ("year", "month", "dayofweek", "hour"):
        g = getattr(df["t"].dt, f).to_pandas()
        e = getattr(pdf["t"].dt, f)
        np.testing.assert_allclose(g.to_numpy().astype(float),
                                   e.to_numpy().astype(float), rtol=0,
                                   equal_nan=True, err_msg=f)
    # sort: NaT last (default) and first, both directions
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
    # groupby by the datetime key: NaT group dropped (pandas dropna=True)
    g = df.groupby("t").sum().to_pandas()
    e = pdf.groupby("t").sum()
    assert len(g) == len(e)
    np.testing.assert_array_equal(g.index.to_numpy(), e.index.to_numpy())
    np.testing.assert_allclose(g["v"].to_numpy(), e["v"].to_numpy(),
                               rtol=1e-12)
    with pytest.raises(_HfErr, match="dropna"):
        df.groupby("t", dropna=False).sum()
    # datetime VALUE column with NaT under an agg is loud
    with pytest.raises(_HfErr, match="NaT"):
        df.groupby("k").min()
    # shift keeps dtype, fills NaT
    g = df[["t"]].shift(2).to_pandas()
    e = pdf[["t"]].shift(2)
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    # fillna(Timestamp): exact ns replace
    fv = pandas.Timestamp("1999-12-31 23:59:59.123456789")
    g = df["t"].fillna(fv).to_pandas()
    e = pdf["t"].fillna(fv)
    assert g.dtype == e.dtype
    np.testing.assert_array_equal(g.to_numpy(), e.to_numpy())
    # where over datetime now fills NaT (dtype kept)
    gw = df[["t", "v"]].where(df["v"] > 0).to_pandas()
    ew = pdf[["t", "v"]].where(pdf["v"] > 0)
    assert gw["t"].dtype == ew["t"].dtype
    np.testing.assert_array_equal(gw["t"].to_numpy(), ew["t"].to_numpy())
    # unsupported ops are loud, not wrong
    for fn in (lambda: df[["t", "v"]].cumsum(),
               lambda: df[["t"]].diff(),
               lambda: df["t"].value_counts(),
               lambda: df[["t"]].astype(np.int64)):
        with pytest.raises(_HfErr):
            fn()


def test_mock_to_datetime(mlib):
    rng = np.random.default_rng(31)
    dates = ["2021-01-05", "1999-12-31", "2024-02-29", None]
    sv = rng.choice(np.array(dates, dtype=object), 2000)
    pdf = pandas.DataFrame({"s": sv})
    df = mlib.DataFrame(pdf)
    got = mlib.to_datetime(df["s"]).to_pandas()
    exp = pandas.to_datetime(pdf["s"])
    assert got.dtype == exp.dtype
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    # format + errors='coerce' (unparseable -> NaT)
    pdf2 = pandas.DataFrame(
        {"s": rng.choice(np.array(["05/01/2021", "31/12/1999", "oops"],
                                  dtype=object), 800)})
    got = mlib.to_datetime(mlib.DataFrame(pdf2)["s"],
                           format="%d/%m/%Y", errors="coerce").to_pandas()
    exp = pandas.to_datetime(pdf2["s"], format="%d/%m/%Y",
                             errors="coerce")
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    # downstream: dt fields after parsing (NaT -> NaN)
    g = mlib.to_datetime(df["s"]).dt.year.to_pandas()
    e = pandas.to_datetime(pdf["s"]).dt.year
    np.testing.assert_allclose(g.to_numpy().astype(float),
                               e.to_numpy().astype(float), rtol=0,
                               equal_nan=True)


def test_mock_series_extras(mlib):
    """Series tail/to_frame/astype/quantile/any/all/mode."""
    rng = np.random.default_rng(32)
    pdf = pandas.DataFrame({"a": rng.integers(0, 6, 400),
                            "v": rng.standard_normal(400)})
    pdf.loc[rng.random(400) < 0.1, "v"] = np.nan
    df = mlib.DataFrame(pdf)
    pandas.testing.assert_series_equal(
        df["v"].tail(7).to_pandas(), pdf["v"].tail(7))
    f = df["v"].to_frame("x").to_pandas()
    pandas.testing.assert_frame_equal(f, pdf["v"].to_frame("x"))
    g = df["a"].astype(np.float64).to_pandas()
    assert g.dtype == np.float64
    np.testing.assert_allclose(g.to_numpy(),
                               pdf["a"].to_numpy().astype(float), rtol=0)
    assert abs(df["v"].quantile(0.3) - pdf["v"].quantile(0.3)) < 1e-12
    np.testing.assert_allclose(
        np.asarray(df["v"].quantile([0.1, 0.9])),
        pdf["v"].quantile([0.1, 0.9]).to_numpy(), rtol=1e-12)
    # any/all incl. NaN-skip rule
    for data in ([0.0, np.nan], [0.0, 1.0], [np.nan], [1.0],
                 [0, 0], [2, 3]):
        ps = pandas.Series(data)
        ms = mlib.DataFrame({"x": ps})["x"]
        assert ms.any() == ps.any(), data
        assert ms.all() == ps.all(), data
    got = df["a"].mode().to_pandas()
    exp = pdf["a"].mode()
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())


def test_mock_window_var_std(mlib):
    rng = np.random.default_rng(33)
    n = 2000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.12] = np.nan
    pdf = pandas.DataFrame({"v": v, "w": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    for w_, mp in ((8, None), (5, 2), (10, 1)):
        for op in ("var", "std"):
            g = getattr(df["v"].rolling(w_, min_periods=mp), op)()
            e = getattr(pdf["v"].rolling(w_, min_periods=mp), op)()
            np.testing.assert_allclose(g.to_pandas().to_numpy(),
                                       e.to_numpy(), rtol=1e-9,
                                       atol=1e-9, equal_nan=True,
                                       err_msg=f"{w_}/{mp}/{op}")
    for mp in (1, 3):
        g = df["v"].expanding(mp).var().to_pandas()
        e = pdf["v"].expanding(mp).var()
        np.testing.assert_allclose(g.to_numpy(), e.to_numpy(),
                                   rtol=1e-9, atol=1e-9, equal_nan=True)
    # frame-wide rolling std
    g = df.rolling(6).std().to_pandas()
    e = pdf.rolling(6).std()
    for c in e.columns:
        np.testing.assert_allclose(g[c].to_numpy(), e[c].to_numpy(),
                                   rtol=1e-9, atol=1e-9, equal_nan=True,
                                   err_msg=c)


def test_mock_duplicated_keep(mlib):
    rng = np.random.default_rng(34)
    n = 3000
    pdf = pandas.DataFrame({
        "k": rng.integers(0, 60, n),
        "s": rng.choice(["a", "b", None], n),
        "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
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


def test_mock_cut_qcut(mlib):
    rng = np.random.default_rng(35)
    n = 4000
    v = rng.standard_normal(n) * 10
    v[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"v": v})
    df = mlib.DataFrame(pdf)
    # labels=False: exact code parity incl. NaN
    for bins in (4, 7, [-30.0, -5.0, 0.0, 5.0, 30.0]):
        for right in (True, False):
            g = mlib.cut(df["v"], bins, right=right,
                         labels=False).to_pandas()
            e = pandas.cut(pdf["v"], bins, right=right, labels=False)
            np.testing.assert_allclose(
                g.to_numpy(), e.to_numpy().astype(float), rtol=0,
                equal_nan=True, err_msg=f"{bins}/{right}")
    # labels=None: Interval values equal pandas' astype(object)
    g = mlib.cut(df["v"], [-30.0, -5.0, 0.0, 5.0, 30.0]).to_pandas()
    e = pandas.cut(pdf["v"], [-30.0, -5.0, 0.0, 5.0, 30.0]).astype(object)
    same = (pandas.isna(g.to_numpy()) & pandas.isna(e.to_numpy())) \
        | (g.to_numpy() == e.to_numpy())
    assert same.all()
    # int-bins Interval values match pandas' computed edges
    g = mlib.cut(df["v"], 5).to_pandas()
    e = pandas.cut(pdf["v"], 5).astype(object)
    same = (pandas.isna(g.to_numpy()) & pandas.isna(e.to_numpy())) \
        | (g.to_numpy() == e.to_numpy())
    assert same.all()
    # qcut codes
    for q in (4, 10):
        g = mlib.qcut(df["v"], q, labels=False).to_pandas()
        e = pandas.qcut(pdf["v"], q, labels=False)
        np.testing.assert_allclose(g.to_numpy(),
                                   e.to_numpy().astype(float), rtol=0,
                                   equal_nan=True, err_msg=str(q))
    # groupby over a cut column (the common binning pattern)
    pdf2 = pandas.DataFrame({"v": v, "w": rng.random(n)})
    df2 = mlib.DataFrame(pdf2)
    df2["bin"] = mlib.cut(df2["v"], [-30.0, 0.0, 30.0])
    got = df2[["bin", "w"]].groupby("bin").sum().to_pandas()
    pdf2["bin"] = pandas.cut(pdf2["v"], [-30.0, 0.0, 30.0])
    exp = pdf2[["bin", "w"]].groupby("bin", observed=True).sum()
    np.testing.assert_allclose(got["w"].to_numpy(), exp["w"].to_numpy(),
                               rtol=1e-12)


def test_mock_set_reset_index(mlib):
    rng = np.random.default_rng(36)
    n = 1500
    pdf = pandas.DataFrame({"k": rng.integers(0, 40, n),
                            "s": rng.choice(["a", "b", None], n),
                            "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    for key in ("k", "s", "v"):
        got = df.set_index(key).to_pandas()
        exp = pdf.set_index(key)
        assert list(got.columns) == list(exp.columns), key
        gi = got.index.to_numpy(dtype=object)
        ei = exp.index.to_numpy(dtype=object)
        same = (pandas.isna(gi) & pandas.isna(ei)) | (gi == ei)
        assert same.all(), key
    # drop=False keeps the column
    got = df.set_index("k", drop=False).to_pandas()
    exp = pdf.set_index("k", drop=False)
    assert list(got.columns) == list(exp.columns)
    # reset_index(drop=False): old index becomes the leading column
    g2 = df.set_index("k").reset_index().to_pandas()
    e2 = pdf.set_index("k").reset_index()
    assert list(g2.columns) == list(e2.columns)
    np.testing.assert_array_equal(g2["k"].to_numpy(), e2["k"].to_numpy())
    pandas.testing.assert_index_equal(g2.index, e2.index)
    # unnamed RangeIndex -> "index" column
    g3 = df.reset_index().to_pandas()
    e3 = pdf.reset_index()
    assert list(g3.columns) == list(e3.columns)
    np.testing.assert_array_equal(g3["index"].to_numpy(),
                                  e3["index"].to_numpy())


@pytest.mark.parametrize("seed", range(500, 540))
def test_mock_fuzz_pipeline(mlib, seed):
    """The GPU fuzz harness body over the numpy mock — a free CPU-tier
    sweep on a DIFFERENT seed range than the GPU tier runs."""
    from tests.test_gpu_fuzz import test_fuzz_pipeline
    test_fuzz_pipeline(seed)


def test_mock_str_extras_shift_astype_isin(mlib):
    """str regex/transform ops, string-column shift, string astype
    (host-dictionary parse), isin with NaN."""
    rng = np.random.default_rng(37)
    n = 2000
    s = rng.choice(["  Alpha ", "beta42", "Gamma", "7.5", None], n)
    num = rng.choice(["1", "2.5", "-3", None], n)
    pdf = pandas.DataFrame({"s": s, "num": num,
                            "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    # regex contains / match / fullmatch
    for pat in (r"a\d+", r"[GA]a?m"):
        g = df["s"].str.contains(pat, regex=True, na=False).to_pandas()
        e = pdf["s"].str.contains(pat, regex=True, na=False)
        np.testing.assert_array_equal(g.to_numpy().astype(bool),
                                      e.to_numpy(), err_msg=pat)
        g = df["s"].str.match(pat, na=False).to_pandas()
        e = pdf["s"].str.match(pat, na=False)
        np.testing.assert_array_equal(g.to_numpy().astype(bool),
                                      e.to_numpy(), err_msg=pat)
    # transforms
    for op, eop in (("strip", "strip"), ("title", "title"),
                    ("capitalize", "capitalize")):
        g = getattr(df["s"].str, op)().to_pandas().to_numpy()
        e = getattr(pdf["s"].str, op)().to_numpy()
        same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
        assert same.all(), op
    g = df["s"].str.replace(r"\d+", "#", regex=True).to_pandas().to_numpy()
    e = pdf["s"].str.replace(r"\d+", "#", regex=True).to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()
    # string shift keeps dtype + NaN fill
    g = df[["s", "v"]].shift(3).to_pandas()
    e = pdf[["s", "v"]].shift(3)
    gs, es = g["s"].to_numpy(), e["s"].to_numpy()
    same = (pandas.isna(gs) & pandas.isna(es)) | (gs == es)
    assert same.all()
    # string astype float (NaN-safe) and int (loud on NaN)
    g = df["num"].astype(np.float64).to_pandas()
    e = pdf["num"].astype(np.float64)
    np.testing.assert_allclose(g.to_numpy(), e.to_numpy(), rtol=0,
                               equal_nan=True)
    with pytest.raises(_HfErr):
        df["num"].astype(np.int64)  # NaN strings
    with pytest.raises(_HfErr):
        df["s"].astype(np.float64)  # unparseable
    # isin with NaN matches NaN rows
    g = df["v"].where(df["v"] > 0).isin([np.nan]).to_pandas()
    e = pdf["v"].where(pdf["v"] > 0).isin([np.nan])
    np.testing.assert_array_equal(g.to_numpy().astype(bool), e.to_numpy())


def test_mock_multikey_merge_right(mlib):
    rng = np.random.default_rng(38)
    nl, nr = 4000, 1500
    lpdf = pandas.DataFrame({
        "a": rng.integers(0, 30, nl),
        "b": rng.integers(0, 6, nl),
        "x": rng.random(nl), "c": rng.integers(0, 5, nl)})
    rpdf = pandas.DataFrame({
        "a": rng.integers(0, 30, nr),
        "b": rng.integers(0, 6, nr),
        "y": rng.random(nr), "c": rng.integers(5, 9, nr)})
    got = mlib.DataFrame(lpdf).merge(mlib.DataFrame(rpdf),
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


def test_mock_where_string_fill(mlib):
    rng = np.random.default_rng(39)
    n = 2000
    pdf = pandas.DataFrame({"s": rng.choice(["a", "b", None], n),
                            "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    m = df["v"] > 0
    pm = pdf["v"] > 0
    # existing-category fill and new-category fill
    for fill in ("a", "zz"):
        g = df[["s"]].where(m, fill).to_pandas()
        e = pdf[["s"]].where(pm, fill)
        gs, es = g["s"].to_numpy(), e["s"].to_numpy()
        same = (pandas.isna(gs) & pandas.isna(es)) | (gs == es)
        assert same.all(), fill
    # mask() form too
    g = df[["s"]].mask(m, "q").to_pandas()
    e = pdf[["s"]].mask(pm, "q")
    gs, es = g["s"].to_numpy(), e["s"].to_numpy()
    same = (pandas.isna(gs) & pandas.isna(es)) | (gs == es)
    assert same.all()
    with pytest.raises(_HfErr):
        df[["v"]].where(m, "x")


def test_mock_named_agg(mlib):
    rng = np.random.default_rng(40)
    n = 3000
    pdf = pandas.DataFrame({"k": rng.integers(0, 25, n),
                            "v": rng.standard_normal(n),
                            "w": rng.integers(-9, 9, n)})
    df = mlib.DataFrame(pdf)
    got = df.groupby("k").agg(total=("v", "sum"), lo=("w", "min"),
                              n=("v", "count")).to_pandas()
    exp = pdf.groupby("k").agg(total=("v", "sum"), lo=("w", "min"),
                               n=("v", "count"))
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    for c in exp.columns:
        np.testing.assert_allclose(got[c].to_numpy().astype(float),
                                   exp[c].to_numpy().astype(float),
                                   rtol=1e-12, err_msg=c)


def test_mock_dropna_how_subset_vc_normalize(mlib):
    rng = np.random.default_rng(41)
    n = 2500
    pdf = pandas.DataFrame({"a": rng.standard_normal(n),
                            "b": rng.standard_normal(n),
                            "s": rng.choice(["x", "y", None], n)})
    pdf.loc[rng.random(n) < 0.3, "a"] = np.nan
    pdf.loc[rng.random(n) < 0.3, "b"] = np.nan
    df = mlib.DataFrame(pdf)
    for how in ("any", "all"):
        for subset in (None, ["a"], ["a", "b"]):
            g = df.dropna(how=how, subset=subset).to_pandas()
            e = pdf.dropna(how=how, subset=subset)
            assert len(g) == len(e), (how, subset)
            np.testing.assert_array_equal(np.asarray(g.index),
                                          e.index.to_numpy(),
                                          err_msg=f"{how}/{subset}")
    g = df["s"].value_counts(normalize=True)
    e = pdf["s"].value_counts(normalize=True)
    np.testing.assert_array_equal(np.asarray(g.index),
                                  e.index.to_numpy())
    np.testing.assert_allclose(np.asarray(g), e.to_numpy(), rtol=1e-12)


def test_mock_misc_surface(mlib):
    rng = np.random.default_rng(42)
    pdf = pandas.DataFrame({"a": rng.integers(0, 9, 300),
                            "v": rng.standard_normal(300)})
    df = mlib.DataFrame(pdf)
    assert not df.empty and df.size == 600 and df.ndim == 2
    np.testing.assert_allclose(df.to_numpy(), pdf.to_numpy(), rtol=0)
    assert df.equals(df.copy())
    assert list(df.keys()) == list(pdf.keys())
    assert dict((k, len(v)) for k, v in df.items()) == \
        {"a": 300, "v": 300}
    g = df.take([5, 1, 7]).to_pandas()
    e = pdf.take([5, 1, 7]).reset_index(drop=True)
    np.testing.assert_allclose(g.to_numpy(), e.to_numpy(), rtol=0)
    assert list(df.add_prefix("p_").columns) == ["p_a", "p_v"]
    assert list(df.add_suffix("_s").columns) == ["a_s", "v_s"]
    d2 = df.copy()
    s = d2.pop("a")
    assert list(d2.columns) == ["v"] and s.name == "a"
    assert d2.get("missing") is None
    assert d2.squeeze().name == "v"
    sv = df["v"]
    assert sv.size == 300 and sv.ndim == 1 and not sv.empty
    assert sv.equals(sv.copy())
    np.testing.assert_allclose(sv.values, pdf["v"].to_numpy(), rtol=0)


def test_mock_query(mlib):
    rng = np.random.default_rng(43)
    n = 3000
    pdf = pandas.DataFrame({"a": rng.integers(0, 20, n),
                            "b": rng.integers(0, 20, n),
                            "v": rng.standard_normal(n),
                            "s": rng.choice(["x", "y", "z"], n)})
    pdf.loc[rng.random(n) < 0.1, "v"] = np.nan
    df = mlib.DataFrame(pdf)
    for expr in ("a > 10", "v <= 0.5 and a != 3", "a > b",
                 "(a > 5 or b < 2) and not v > 0",
                 "s == 'x'", "a == b or v > 1.5"):
        g = df.query(expr).to_pandas()
        e = pdf.query(expr)
        assert len(g) == len(e), expr
        np.testing.assert_array_equal(np.asarray(g.index),
                                      e.index.to_numpy(), err_msg=expr)
    with pytest.raises(_HfErr):
        df.query("c > 1")
    with pytest.raises(_HfErr):
        df.query("a + b > 2")


def test_mock_filter_selectdtypes_dtfloor(mlib):
    rng = np.random.default_rng(44)
    n = 1000
    t = pandas.Series(pandas.to_datetime("1965-01-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**7, n), unit="min"))
    t[rng.random(n) < 0.1] = pandas.NaT
    pdf = pandas.DataFrame({"aa": rng.integers(0, 5, n),
                            "ab": rng.standard_normal(n),
                            "s": rng.choice(["x", "y"], n), "t": t})
    df = mlib.DataFrame(pdf)
    assert list(df.filter(items=["ab", "s"]).columns) == ["ab", "s"]
    assert list(df.filter(like="a").columns) == ["aa", "ab"]
    assert list(df.filter(regex="^a.$").columns) == ["aa", "ab"]
    assert list(df.select_dtypes(include="number").columns) == \
        list(pdf.select_dtypes(include="number").columns)
    assert list(df.select_dtypes(include=object).columns) == ["s"]
    assert list(df.select_dtypes(exclude=["datetime64[ns]"]).columns) == \
        ["aa", "ab", "s"]
    # dt.floor / normalize incl. pre-1970 values and NaT
    for freq in ("D", "h", "min"):
        g = df["t"].dt.floor(freq).to_pandas()
        e = pdf["t"].dt.floor(freq)
        assert g.dtype == e.dtype
        np.testing.assert_array_equal(g.to_numpy(), e.to_numpy(),
                                      err_msg=freq)
    g = df["t"].dt.normalize().to_pandas()
    e = pdf["t"].dt.normalize()
    np.testing.assert_array_equal(g.to_numpy(), e.to_numpy())


def test_mock_get_dummies(mlib):
    rng = np.random.default_rng(45)
    n = 1500
    pdf = pandas.DataFrame({"s": rng.choice(["a", "b", "c", None], n)})
    df = mlib.DataFrame(pdf)
    got = mlib.get_dummies(df["s"]).to_pandas()
    exp = pandas.get_dummies(pdf["s"])
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got.to_numpy().astype(bool),
                                  exp.to_numpy().astype(bool))
    got = mlib.get_dummies(df["s"], prefix="p").to_pandas()
    exp = pandas.get_dummies(pdf["s"], prefix="p")
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got.to_numpy().astype(bool),
                                  exp.to_numpy().astype(bool))


def test_mock_where_datetime(mlib):
    rng = np.random.default_rng(46)
    n = 1500
    t = pandas.Series(pandas.to_datetime("2020-01-01")
                      + pandas.to_timedelta(
                          rng.integers(0, 10**6, n), unit="min"))
    t[rng.random(n) < 0.1] = pandas.NaT
    pdf = pandas.DataFrame({"t": t, "v": rng.standard_normal(n)})
    df = mlib.DataFrame(pdf)
    m, pm = df["v"] > 0, pdf["v"] > 0
    g = df[["t"]].where(m).to_pandas()
    e = pdf[["t"]].where(pm)
    assert g["t"].dtype == e["t"].dtype
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    fv = pandas.Timestamp("1999-01-01 03:04:05")
    g = df[["t"]].mask(m, fv).to_pandas()
    e = pdf[["t"]].mask(pm, fv)
    np.testing.assert_array_equal(g["t"].to_numpy(), e["t"].to_numpy())
    with pytest.raises(_HfErr):
        df[["v"]].where(m, fv)


def test_mock_concat_axis1_numeric_only(mlib):
    rng = np.random.default_rng(47)
    n = 800
    p1 = pandas.DataFrame({"a": rng.integers(0, 9, n),
                           "s": rng.choice(["x", "y"], n)})
    p2 = pandas.DataFrame({"b": rng.random(n)})
    got = mlib.concat([mlib.DataFrame(p1), mlib.DataFrame(p2)],
                      axis=1).to_pandas()
    exp = pandas.concat([p1, p2], axis=1)
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_allclose(got["b"].to_numpy(), exp["b"].to_numpy(),
                               rtol=0)
    # Series operand
    got = mlib.concat([mlib.DataFrame(p2),
                       mlib.DataFrame(p1)["a"].to_frame("a2")],
                      axis=1).to_pandas()
    assert list(got.columns) == ["b", "a2"]
    # numeric_only reductions skip string columns
    df = mlib.DataFrame(p1)
    g = df.sum(numeric_only=True)
    e = p1.sum(numeric_only=True)
    np.testing.assert_allclose(np.asarray(g), e.to_numpy().astype(float),
                               rtol=0)
    g = df.mean(numeric_only=True)
    e = p1.mean(numeric_only=True)
    np.testing.assert_allclose(np.asarray(g), e.to_numpy(), rtol=1e-12)


@pytest.mark.parametrize("n", [0, 1])
def test_mock_edge_lengths(mlib, n):
    """Empty and single-row frames through the whole op surface —
    results (columns, lengths, values) match pandas."""
    pdf = pandas.DataFrame({"k": np.arange(n, dtype=np.int64),
                            "v": np.ones(n)})
    df = mlib.DataFrame(pdf)
    checks = [
        ("gbsum", lambda: df.groupby("k").sum().to_pandas(),
         lambda: pdf.groupby("k").sum()),
        ("sort", lambda: df.sort_values("k").to_pandas(),
         lambda: pdf.sort_values("k")),
        ("cumsum", lambda: df.cumsum().to_pandas(),
         lambda: pdf.cumsum()),
        ("dedup", lambda: df.drop_duplicates(["k"]).to_pandas(),
         lambda: pdf.drop_duplicates(subset=["k"])),
        ("merge", lambda: df.merge(
            mlib.DataFrame(pdf.rename(columns={"v": "w"})),
            on="k").to_pandas(),
         lambda: pdf.merge(pdf.rename(columns={"v": "w"}), on="k")),
        ("where", lambda: df.where(df["v"] > 2).to_pandas(),
         lambda: pdf.where(pdf["v"] > 2)),
        ("shift", lambda: df.shift(1).to_pandas(),
         lambda: pdf.shift(1)),
        ("rank", lambda: df.rank().to_pandas(), lambda: pdf.rank()),
        ("filter", lambda: df[df["v"] > 0].to_pandas(),
         lambda: pdf[pdf["v"] > 0]),
        ("concat", lambda: mlib.concat([df, df],
                                       ignore_index=True).to_pandas(),
         lambda: pandas.concat([pdf, pdf], ignore_index=True)),
        ("melt", lambda: df.melt(id_vars="k").to_pandas(),
         lambda: pdf.melt(id_vars="k")),
    ]
    for name, gf, ef in checks:
        g, e = gf(), ef()
        assert list(g.columns) == list(e.columns), (n, name)
        assert len(g) == len(e), (n, name)
        if len(e):
            num = e.select_dtypes(include="number").columns
            np.testing.assert_allclose(
                g[num].to_numpy().astype(float),
                e[num].to_numpy().astype(float), rtol=0,
                equal_nan=True, err_msg=f"{n}/{name}")
    np.testing.assert_allclose(np.asarray(df.sum(), dtype=float),
                               pdf.sum().to_numpy().astype(float),
                               rtol=0)
    assert df["v"].nunique() == pdf["v"].nunique()


@pytest.mark.parametrize("seed", range(600, 620))
def test_mock_fuzz_merge(mlib, seed):
    """The GPU merge fuzzer body over the numpy mock (different seed
    range than the GPU tier)."""
    from tests.test_gpu_fuzz import test_fuzz_merge
    test_fuzz_merge(seed)


def test_mock_left_on_key_name_collision(mlib):
    """left_on/right_on where the LEFT key name also exists as a RIGHT
    payload column: pandas suffixes the key itself (a_x / a_y)."""
    rng = np.random.default_rng(48)
    l = pandas.DataFrame({"a": rng.integers(0, 10, 500),
                          "v": rng.random(500)})
    r = pandas.DataFrame({"b": rng.integers(0, 10, 200),
                          "a": rng.integers(50, 60, 200),
                          "w": rng.random(200)})
    for how in ("inner", "left", "right", "outer"):
        g = mlib.DataFrame(l).merge(mlib.DataFrame(r), left_on="a",
                                    right_on="b", how=how).to_pandas()
        e = l.merge(r, left_on="a", right_on="b", how=how)
        assert list(g.columns) == list(e.columns), how
        assert len(g) == len(e), how
        order = list(e.columns)
        gs = g.sort_values(order, na_position="last").reset_index(
            drop=True)
        es = e.sort_values(order, na_position="last").reset_index(
            drop=True)
        np.testing.assert_allclose(gs.to_numpy().astype(float),
                                   es.to_numpy().astype(float), rtol=0,
                                   equal_nan=True, err_msg=how)


def test_mock_sem(mlib):
    rng = np.random.default_rng(49)
    n = 3000
    pdf = pandas.DataFrame({"k": rng.integers(0, 30, n),
                            "v": rng.standard_normal(n),
                            "w": rng.standard_normal(n)})
    pdf.loc[rng.random(n) < 0.1, "v"] = np.nan
    df = mlib.DataFrame(pdf)
    np.testing.assert_allclose(np.asarray(df.sem()),
                               pdf.sem().to_numpy(), rtol=1e-10)
    g = df.groupby("k").sem().to_pandas()
    e = pdf.groupby("k").sem()
    np.testing.assert_array_equal(g.index.to_numpy(), e.index.to_numpy())
    for c in e.columns:
        np.testing.assert_allclose(g[c].to_numpy(), e[c].to_numpy(),
                                   rtol=1e-9, equal_nan=True, err_msg=c)
    s = df.groupby("k")["v"].sem().to_pandas()
    es = pdf.groupby("k")["v"].sem()
    np.testing.assert_allclose(s.to_numpy(), es.to_numpy(), rtol=1e-9,
                               equal_nan=True)
