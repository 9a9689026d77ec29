"""GPU tier: dictionary-encoded string columns vs golden
vectors generated from the REAL reference (oracle/make_golden.py
gen_string_cases) — groupby by string key, merge on string key, string
comparisons/filters, sort, concat with differing dictionaries.

Strings never touch the device: columns hold int64 codes (−1 = NaN) against
a host-side sorted dictionary; every device op is the same int64 kernel the
numeric paths use.
"""

import numpy as np
import pandas
import pytest

import modin_amd.config as config
import modin_amd.pandas as mpd
from modin_amd.core import lib
from tests.conftest import load_golden

pytestmark = pytest.mark.gpu

NA = "__NA__"
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


def dec(arr):
    """'<U' golden array with the NA sentinel -> object array with NaN."""
    out = arr.astype(object)
    out[arr == NA] = np.nan
    return out


def assert_str_equal(got, expect_enc, msg=""):
    got = np.asarray(got, dtype=object)
    exp = dec(expect_enc)
    assert len(got) == len(exp), f"{msg}: length {len(got)} != {len(exp)}"
    for i, (g, e) in enumerate(zip(got, exp)):
        if isinstance(e, float) and np.isnan(e):
            assert isinstance(g, float) and np.isnan(g), f"{msg}[{i}]: {g!r}"
        else:
            assert g == e, f"{msg}[{i}]: {g!r} != {e!r}"


def test_string_roundtrip(npartitions):
    g = load_golden("str_gb")
    s = dec(g["in_s"])
    df = mpd.DataFrame({"s": s, "v": g["in_v"]})
    out = df.to_pandas()
    assert out["s"].dtype == np.dtype(object)
    assert_str_equal(out["s"].to_numpy(), g["in_s"], "roundtrip")
    np.testing.assert_array_equal(out["v"].to_numpy(), g["in_v"])


@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_string_groupby_vs_golden(agg, npartitions):
    g = load_golden("str_gb")
    df = mpd.DataFrame({"s": dec(g["in_s"]), "v": g["in_v"],
                        "w": g["in_w"]})
    out = getattr(df.groupby("s"), agg)().to_pandas()
    assert_str_equal(out.index.to_numpy(), g[f"out_{agg}_keys"],
                     f"{agg} keys")
    for cn in ("v", "w"):
        expect = g[f"out_{agg}_{cn}"]
        if agg == "count":
            np.testing.assert_array_equal(out[cn].to_numpy(),
                                          expect.astype(np.int64))
        else:
            np.testing.assert_allclose(out[cn].to_numpy(), expect,
                                       rtol=RTOL, atol=1e-9, equal_nan=True)


def test_string_merge_vs_golden(npartitions):
    g = load_golden("str_merge")
    left = mpd.DataFrame({"s": dec(g["in_lk"]), "a": g["in_la"]})
    right = mpd.DataFrame({"s": dec(g["in_rk"]), "b": g["in_rb"]})
    out = left.merge(right, on="s").to_pandas()
    assert_str_equal(out["s"].to_numpy(), g["out_s"], "merge key")
    np.testing.assert_array_equal(out["a"].to_numpy(), g["out_a"])
    np.testing.assert_array_equal(out["b"].to_numpy(), g["out_b"])


def test_string_filter_vs_golden(npartitions):
    g = load_golden("str_filter")
    df = mpd.DataFrame({"s": dec(g["in_s"]), "v": g["in_v"]})
    masks = {
        "eq": df["s"] == "apple",
        "ne": df["s"] != "apple",
        "gt": df["s"] > "m",
        "le": df["s"] <= "kiwi",
        "eq_missing": df["s"] == "notthere",
    }
    for tag, m in masks.items():
        out = df[m].to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{tag}_idx"], err_msg=tag)
        assert_str_equal(out["s"].to_numpy(), g[f"out_{tag}_s"], tag)
        np.testing.assert_array_equal(out["v"].to_numpy(),
                                      g[f"out_{tag}_v"], err_msg=tag)
    out = df.dropna().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_dropna_idx"])
    assert_str_equal(out["s"].to_numpy(), g["out_dropna_s"], "dropna")


def test_string_sort_vs_golden(npartitions):
    g = load_golden("str_sort")
    df = mpd.DataFrame({"s": dec(g["in_s"]), "v": g["in_v"]})
    out = df.sort_values("s").to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_idx"])
    assert_str_equal(out["s"].to_numpy(), g["out_s"], "sorted keys")
    np.testing.assert_array_equal(out["v"].to_numpy(), g["out_v"])


def test_string_concat_vs_golden(npartitions):
    g = load_golden("str_concat")
    d1 = mpd.DataFrame({"s": dec(g["in_s1"]), "v": g["in_v1"]})
    d2 = mpd.DataFrame({"s": dec(g["in_s2"]), "v": g["in_v2"]})
    cat = mpd.concat([d1, d2], ignore_index=True)
    out = cat.to_pandas()
    assert_str_equal(out["s"].to_numpy(), g["out_cat_s"], "concat order")
    gb = cat.groupby("s").sum().to_pandas()
    assert_str_equal(gb.index.to_numpy(), g["out_gb_keys"], "concat gb keys")
    np.testing.assert_allclose(gb["v"].to_numpy(), g["out_gb_v"], rtol=RTOL,
                               atol=1e-9)


def test_string_error_surfaces(npartitions):
    df = mpd.DataFrame({"s": np.array(["a", "b", None], dtype=object),
                        "v": np.array([1.0, 2.0, 3.0])})
    with pytest.raises(lib.HfError, match="arithmetic on string"):
        (df + 1).to_pandas()
    with pytest.raises(lib.HfError, match="string column"):
        df.sum()
    with pytest.raises(lib.HfError, match="string"):
        df.groupby("v").sum().to_pandas()  # string VALUE column
    with pytest.raises(lib.HfError, match="ordering comparison"):
        (df["s"] > 3).to_pandas()
    # numeric column vs string scalar
    with pytest.raises(lib.HfError, match="string scalar"):
        (df["v"] == "a").to_pandas()


def test_string_sort_nan_multi_vs_golden(npartitions):
    """String sort keys WITH NaN (na_position='last' both directions) and
    string+int multi-key sorts.  Pinned against pandas: the reference
    itself crashes on NaN string sort keys (numpy partition over a mixed
    object array in its range-partitioning sampler) — see
    make_golden.gen_sort2_cases."""
    g = load_golden("srt_str_nan")
    df = mpd.DataFrame({"s": dec(g["in_s"]), "w": g["in_w"],
                        "v": g["in_v"]})
    for tag, by, asc in [("s_asc", "s", True), ("s_desc", "s", False),
                         ("sw", ["s", "w"], True),
                         ("ws_mixed", ["w", "s"], [False, True])]:
        out = df.sort_values(by, ascending=asc).to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{tag}_idx"], err_msg=tag)
        assert_str_equal(out["s"].to_numpy(), g[f"out_{tag}_s"], tag)
        np.testing.assert_array_equal(out["w"].to_numpy(),
                                      g[f"out_{tag}_w"], err_msg=tag)


def test_string_category_dtype_roundtrip(npartitions):
    """category dtype encodes by CATEGORY order (pandas sort semantics for
    categoricals) and round-trips through astype back to category."""
    c = pandas.Categorical(["lo", "hi", "lo", "mid"],
                           categories=["lo", "mid", "hi"], ordered=True)
    pdf = pandas.DataFrame({"c": c, "v": np.arange(4, dtype=np.float64)})
    df = mpd.DataFrame(pdf)
    out = df.to_pandas()
    assert list(out["c"]) == list(pdf["c"])
    srt = df.sort_values("c").to_pandas()
    exp = pdf.sort_values("c", kind="stable")
    assert list(srt["c"]) == list(exp["c"])
    np.testing.assert_array_equal(srt.index.to_numpy(),
                                  exp.index.to_numpy())


def test_read_parquet_vs_pandas(tmp_path, npartitions):
    """§8f.4 columnar ingestion: pyarrow parquet -> device columns; output
    matches pandas.read_parquet (strings object + NaN, nullable ints as
    float64, projection, multi-row-group files)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    rng = np.random.default_rng(71)
    n = 20_000
    s_vals = rng.choice(np.array(["red", "green", "blue", "Amber", "x y"]),
                        n).astype(object)
    s_vals[rng.random(n) < 0.07] = None
    v = rng.random(n)
    v[rng.random(n) < 0.05] = np.nan
    k = rng.integers(-1000, 1000, n)
    ni = k.astype(object)
    ni[rng.random(n) < 0.03] = None  # nullable int -> float64 in pandas
    table = pa.table({
        "s": pa.array(s_vals),
        "sd": pa.array(s_vals).dictionary_encode(),
        "v": pa.array(v),
        "k": pa.array(k, type=pa.int64()),
        "ni": pa.array(ni.tolist(), type=pa.int64()),
    })
    path = str(tmp_path / "t.parquet")
    pq.write_table(table, path, row_group_size=3000)  # multi-row-group
    got = mpd.read_parquet(path).to_pandas()
    exp = pandas.read_parquet(path)
    assert list(got.columns) == list(exp.columns)
    for c in ("s", "sd"):
        assert_str_equal(got[c].to_numpy(),
                         np.array([NA if (x is None or (isinstance(x, float)
                                                        and np.isnan(x)))
                                   else str(x) for x in exp[c]]), c)
    np.testing.assert_array_equal(got["v"].to_numpy(), exp["v"].to_numpy())
    np.testing.assert_array_equal(got["k"].to_numpy(), exp["k"].to_numpy())
    assert got["ni"].dtype == np.float64
    np.testing.assert_array_equal(got["ni"].to_numpy(),
                                  exp["ni"].to_numpy())
    # projection
    got2 = mpd.read_parquet(path, columns=["k", "s"]).to_pandas()
    assert list(got2.columns) == ["k", "s"]
    np.testing.assert_array_equal(got2["k"].to_numpy(), exp["k"].to_numpy())
    # the ingested frame computes: groupby by the string key vs pandas
    df = mpd.read_parquet(path, columns=["s", "v"])
    g1 = df.groupby("s").sum().to_pandas()
    g2 = exp[["s", "v"]].groupby("s").sum()
    assert list(g1.index) == list(g2.index)
    np.testing.assert_allclose(g1["v"].to_numpy(), g2["v"].to_numpy(),
                               rtol=RTOL, atol=1e-9)


def test_series_utils_vs_golden(npartitions):
    """Series.unique (appearance order, NaN kept), value_counts (count
    desc, ties by appearance, NaN dropped), nunique, isin — int and string
    columns vs the reference."""
    g = load_golden("ser_utils")
    df = mpd.DataFrame({"k": g["in_k"], "s": dec(g["in_s"])})
    np.testing.assert_array_equal(df["k"].unique(), g["out_k_unique"])
    vc = df["k"].value_counts()
    np.testing.assert_array_equal(vc.index.to_numpy(), g["out_k_vc_idx"])
    np.testing.assert_array_equal(vc.to_numpy(), g["out_k_vc"])
    assert df["k"].nunique() == int(g["out_k_nunique"][0])
    su = df["s"].unique()
    assert_str_equal(su, g["out_s_unique"], "s unique")
    svc = df["s"].value_counts()
    assert_str_equal(svc.index.to_numpy(), g["out_s_vc_idx"], "s vc idx")
    np.testing.assert_array_equal(svc.to_numpy(), g["out_s_vc"])
    assert df["s"].nunique() == int(g["out_s_nunique"][0])
    np.testing.assert_array_equal(
        df["k"].isin([3, -7, 999]).to_pandas().to_numpy().astype(np.int64),
        g["out_k_isin"])
    np.testing.assert_array_equal(
        df["s"].isin(["oak", "Yew", "missing"]).to_pandas().to_numpy()
        .astype(np.int64), g["out_s_isin"])
    np.testing.assert_array_equal(
        df["k"].isin([]).to_pandas().to_numpy().astype(np.int64),
        g["out_empty_isin"])


def test_parquet_write_roundtrip(tmp_path, npartitions):
    """to_parquet -> read_parquet round trip: dictionary columns travel as
    parquet dictionary pages, numerics as plain columns."""
    rng = np.random.default_rng(83)
    n = 15_000
    sv = rng.choice(np.array(["aa", "bb", "cc"]), n).astype(object)
    sv[rng.random(n) < 0.05] = None
    pdf = pandas.DataFrame({"s": sv, "v": rng.random(n),
                            "k": rng.integers(-5, 5, n)})
    df = mpd.DataFrame(pdf)
    path = str(tmp_path / "w.parquet")
    df.to_parquet(path)
    back = mpd.read_parquet(path).to_pandas()
    exp = pandas.read_parquet(path)
    assert list(back.columns) == ["s", "v", "k"]
    np.testing.assert_array_equal(back["v"].to_numpy(),
                                  pdf["v"].to_numpy())
    np.testing.assert_array_equal(back["k"].to_numpy(),
                                  pdf["k"].to_numpy())
    for g, e in zip(back["s"], pdf["s"]):
        if e is None:
            assert isinstance(g, float) and np.isnan(g)
        else:
            assert g == e
    # pandas reads our file identically
    np.testing.assert_array_equal(exp["v"].to_numpy(), pdf["v"].to_numpy())


def test_drop_nunique_series_sort(npartitions):
    rng = np.random.default_rng(84)
    n = 20_000
    pdf = pandas.DataFrame({"k": rng.integers(0, 7, n),
                            "v": rng.random(n),
                            "w": rng.integers(0, 1000, n)})
    df = mpd.DataFrame(pdf)
    d = df.drop(columns=["v"]).to_pandas()
    assert list(d.columns) == ["k", "w"]
    nu = df.nunique()
    pnu = pdf.nunique()
    np.testing.assert_array_equal(np.asarray(nu[["k", "w"]]),
                                  pnu[["k", "w"]].to_numpy())
    ss = df["w"].sort_values().to_pandas()
    ps = pdf["w"].sort_values(kind="stable")
    np.testing.assert_array_equal(ss.to_numpy(), ps.to_numpy())
    np.testing.assert_array_equal(ss.index.to_numpy(), ps.index.to_numpy())


def test_left_merge_string_vs_golden(npartitions):
    """Left merge on STRING keys with a string payload column: unmatched
    lefts get NaN payload (dict code −1)."""
    g = load_golden("mgl_str")
    left = mpd.DataFrame({"s": dec(g["in_ls"]),
                          "a": np.zeros(len(g["in_ls"]))})
    right = mpd.DataFrame({"s": dec(g["in_rs"]), "t": dec(g["in_rt"])})
    out = left.merge(right, on="s", how="left").to_pandas()
    assert_str_equal(out["s"].to_numpy(), g["out_s"], "left key")
    assert_str_equal(out["t"].to_numpy(), g["out_t"], "left payload")
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_idx"])


def test_groupby_first_last_vs_golden(npartitions):
    """groupby.first/last: value at min/max original position among each
    group's non-NaN rows — float (all-NaN group -> NaN), int (dtype
    preserved) and STRING value columns, vs the reference."""
    g = load_golden("gbfl_cases")
    df = mpd.DataFrame({"k": g["in_k"], "v": g["in_v"], "w": g["in_w"],
                        "s": dec(g["in_s"])})
    for agg in ("first", "last"):
        out = getattr(df.groupby("k"), agg)().to_pandas()
        np.testing.assert_array_equal(out.index.to_numpy(),
                                      g[f"out_{agg}_keys"])
        np.testing.assert_allclose(out["v"].to_numpy(), g[f"out_{agg}_v"],
                                   rtol=0, equal_nan=True, err_msg=agg)
        assert out["w"].dtype == np.int64
        np.testing.assert_array_equal(out["w"].to_numpy(),
                                      g[f"out_{agg}_w"])
        assert_str_equal(out["s"].to_numpy(), g[f"out_{agg}_s"],
                         f"{agg} s")


def test_groupby_nunique_vs_golden(npartitions):
    """groupby.nunique: distinct non-NaN values per group via sorted
    run-boundary masks; int/float/string values; all-NaN groups 0."""
    g = load_golden("gbnu_cases")
    df = mpd.DataFrame({"k": g["in_k"], "v": g["in_v"], "w": g["in_w"],
                        "s": dec(g["in_s"])})
    out = df.groupby("k").nunique().to_pandas()
    np.testing.assert_array_equal(out.index.to_numpy(), g["out_keys"])
    for cn in ("v", "w", "s"):
        assert out[cn].dtype == np.int64
        np.testing.assert_array_equal(out[cn].to_numpy(), g[f"out_{cn}"],
                                      err_msg=cn)
    out2 = df[["s", "v"]].groupby("s").nunique().to_pandas()
    assert_str_equal(out2.index.to_numpy(), g["out_sk_keys"], "sk keys")
    np.testing.assert_array_equal(out2["v"].to_numpy(), g["out_sk_v"])


def test_string_value_groupby_min_max_count(gpu_ready):
    """groupby min/max/count over STRING value columns (round-2 lift):
    sorted dictionaries make code order == lex order, so the numeric
    code path aggregates and decodes (NaN strings skipped like pandas)."""
    rng = np.random.default_rng(42)
    n = 20_000
    k = rng.integers(0, 150, n).astype(np.int64)
    words = np.array([f"w{i:04d}" for i in range(400)], dtype=object)
    sv = words[rng.integers(0, 400, n)]
    v = rng.random(n)
    # NaN-free strings for min/max: pandas ITSELF raises on object min
    # with mixed str/NaN ("agg function failed [how->min,dtype->object]");
    # count gets the NaN-bearing column below
    pdf = pandas.DataFrame({"k": k, "s": sv, "v": v})
    df = mpd.DataFrame(pdf)
    svn = sv.copy()
    svn[rng.random(n) < 0.1] = np.nan
    pdfn = pandas.DataFrame({"k": k, "s": svn, "v": v})
    dfn = mpd.DataFrame(pdfn)
    got = dfn.groupby("k").count().to_pandas()
    exp = pdfn.groupby("k").count()
    for c in exp.columns:
        np.testing.assert_array_equal(got[c].to_numpy(),
                                      exp[c].to_numpy(),
                                      err_msg=f"count-nan/{c}")
    for op in ("min", "max", "count"):
        got = getattr(df.groupby("k"), op)().to_pandas()
        exp = getattr(pdf.groupby("k"), op)()
        assert list(got.columns) == list(exp.columns)
        np.testing.assert_array_equal(got.index.to_numpy(),
                                      exp.index.to_numpy())
        for c in exp.columns:
            ge, ee = got[c].to_numpy(), exp[c].to_numpy()
            if c == "s" and op != "count":
                same = (pandas.isna(ge) & pandas.isna(ee)) | (ge == ee)
                assert same.all(), f"{op}/s mismatch"
            else:
                np.testing.assert_allclose(ge.astype(np.float64),
                                           ee.astype(np.float64),
                                           rtol=1e-12, err_msg=f"{op}/{c}")


def test_str_accessor_vs_pandas(npartitions):
    """Series.str on device-backed dictionary columns (host LUT + gather)."""
    rng = np.random.default_rng(9)
    n = 40_000
    words = np.array(["Apple", "beta", "Ba", "apple", "CAT", "ca t"],
                     dtype=object)
    sv = words[rng.integers(0, len(words), n)]
    sv[rng.random(n) < 0.1] = np.nan
    t = pandas.Series(sv, name="s")
    df = mpd.DataFrame(pandas.DataFrame({"s": t}))
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
    # masks compose with row selection
    sel = df[df["s"].str.contains("a", na=False)].to_pandas()
    exp_sel = pandas.DataFrame({"s": t})[t.str.contains("a", na=False)]
    same = (pandas.isna(sel["s"].to_numpy()) &
            pandas.isna(exp_sel["s"]).to_numpy()) | \
        (sel["s"].to_numpy() == exp_sel["s"].to_numpy())
    assert same.all()


def test_read_csv_vs_pandas(tmp_path, npartitions):
    """read_csv columnar ingestion: pyarrow.csv -> device columns; output
    matches pandas.read_csv for int64/float64/string columns with nulls."""
    rng = np.random.default_rng(73)
    n = 20_000
    sv = rng.choice(np.array(["red", "green", "blue", "x y"]),
                    n).astype(object)
    sv[rng.random(n) < 0.07] = None
    v = rng.random(n).round(6)
    v[rng.random(n) < 0.05] = np.nan
    k = rng.integers(-1000, 1000, n)
    pdf = pandas.DataFrame({"k": k, "v": v, "s": sv})
    path = str(tmp_path / "t.csv")
    pdf.to_csv(path, index=False)
    exp = pandas.read_csv(path)
    got = mpd.read_csv(path).to_pandas()
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got["k"].to_numpy(), exp["k"].to_numpy())
    np.testing.assert_allclose(got["v"].to_numpy(), exp["v"].to_numpy(),
                               rtol=0, atol=1e-12, equal_nan=True)
    ge, ee = got["s"].to_numpy(), exp["s"].to_numpy()
    same = (pandas.isna(ge) & pandas.isna(ee)) | (ge == ee)
    assert same.all()
    # groupby straight off the ingested frame
    g2 = mpd.read_csv(path).groupby("k").count().to_pandas()
    e2 = exp.groupby("k").count()
    np.testing.assert_array_equal(g2["v"].to_numpy(), e2["v"].to_numpy())


def test_to_csv_roundtrip(tmp_path, npartitions):
    """to_csv -> read_csv round trip (pyarrow C++ writer/reader), incl.
    strings with NaN and datetimes."""
    rng = np.random.default_rng(131)
    n = 5000
    pdf = pandas.DataFrame({
        "a": rng.integers(-100, 100, n),
        "v": np.round(rng.standard_normal(n), 6),
        "s": rng.choice(["aa", "bb", None], n)})
    df = mpd.DataFrame(pdf)
    p = str(tmp_path / "t.csv")
    df.to_csv(p)
    back = mpd.read_csv(p).to_pandas()
    np.testing.assert_array_equal(back["a"].to_numpy(),
                                  pdf["a"].to_numpy())
    np.testing.assert_allclose(back["v"].to_numpy(), pdf["v"].to_numpy(),
                               rtol=0, atol=1e-12)
    g, e = back["s"].to_numpy(), pdf["s"].to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()


def test_astype_dict(npartitions):
    rng = np.random.default_rng(132)
    pdf = pandas.DataFrame({"a": rng.integers(0, 50, 3000),
                            "v": np.round(rng.standard_normal(3000), 3),
                            "w": rng.integers(-9, 9, 3000)})
    df = mpd.DataFrame(pdf)
    got = df.astype({"a": np.float64, "v": np.int64}).to_pandas()
    exp = pdf.astype({"a": np.float64, "v": np.int64})
    assert list(got.dtypes) == list(exp.dtypes)
    np.testing.assert_allclose(got.to_numpy(), exp.to_numpy(), rtol=0)


def test_to_datetime_vs_pandas(npartitions):
    """to_datetime: host-dictionary parse + one device LUT gather;
    NaT for string NaN / coerced failures; downstream dt fields."""
    rng = np.random.default_rng(134)
    n = 30_000
    days = rng.integers(0, 2000, 40)
    cats = [(pandas.Timestamp("2019-01-01")
             + pandas.Timedelta(days=int(d))).strftime("%Y-%m-%d")
            for d in days] + [None]
    sv = rng.choice(np.array(cats, dtype=object), n)
    pdf = pandas.DataFrame({"s": sv})
    df = mpd.DataFrame(pdf)
    got = mpd.to_datetime(df["s"]).to_pandas()
    exp = pandas.to_datetime(pdf["s"])
    assert got.dtype == exp.dtype
    np.testing.assert_array_equal(got.to_numpy(), exp.to_numpy())
    g = mpd.to_datetime(df["s"]).dt.dayofweek.to_pandas()
    e = pandas.to_datetime(pdf["s"]).dt.dayofweek
    np.testing.assert_allclose(g.to_numpy().astype(float),
                               e.to_numpy().astype(float), rtol=0,
                               equal_nan=True)
    # sorting the parsed column: NaT last
    d2 = mpd.DataFrame(query_compiler=mpd.to_datetime(
        df["s"])._query_compiler)
    gs = d2.sort_values("s").to_pandas()["s"]
    es = pandas.DataFrame({"s": exp}).sort_values(
        "s", kind="stable")["s"]
    np.testing.assert_array_equal(gs.to_numpy(), es.to_numpy())


def test_parquet_csv_datetime_roundtrip(tmp_path, npartitions):
    """Parquet and CSV readers carry timestamp columns to the tagged
    int64-ns device form (NaT included); to_parquet writes them back."""
    rng = np.random.default_rng(140)
    n = 4000
    t = pandas.Series(pandas.to_datetime("2020-06-01")
                      + pandas.to_timedelta(rng.integers(0, 10**6, n),
                                            unit="s"))
    t[rng.random(n) < 0.1] = pandas.NaT
    pdf = pandas.DataFrame({"t": t, "v": rng.standard_normal(n)})
    p = str(tmp_path / "d.parquet")
    pdf.to_parquet(p)
    got = mpd.read_parquet(p)
    assert got.dtypes["t"] == np.dtype("datetime64[ns]")
    back = got.to_pandas()
    np.testing.assert_array_equal(back["t"].to_numpy(),
                                  pdf["t"].to_numpy())
    # write side
    p2 = str(tmp_path / "d2.parquet")
    got.to_parquet(p2)
    again = pandas.read_parquet(p2)
    np.testing.assert_array_equal(again["t"].to_numpy(),
                                  pdf["t"].to_numpy())
    # CSV: pyarrow infers ISO timestamps
    p3 = str(tmp_path / "d.csv")
    pdf.to_csv(p3, index=False)
    gcsv = mpd.read_csv(p3)
    assert gcsv.dtypes["t"] == np.dtype("datetime64[ns]")
    np.testing.assert_array_equal(gcsv.to_pandas()["t"].to_numpy(),
                                  pdf["t"].to_numpy())
    # dt accessor straight off the ingested column
    g = gcsv["t"].dt.hour.to_pandas()
    e = pdf["t"].dt.hour
    np.testing.assert_allclose(g.to_numpy().astype(float),
                               e.to_numpy().astype(float), rtol=0,
                               equal_nan=True)


def test_str_extras_vs_pandas(npartitions):
    """Regex contains/match/fullmatch, replace, strip/title/capitalize/
    zfill, string shift, string astype, isin(NaN)."""
    rng = np.random.default_rng(141)
    n = 30_000
    s = rng.choice(["  Alpha ", "beta42", "Gamma", "7.5", "x9y", None], n)
    num = rng.choice(["1", "2.5", "-3"], n)
    pdf = pandas.DataFrame({"s": s, "num": num,
                            "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    for pat in (r"a\d+", r"^[Gx]"):
        g = df["s"].str.contains(pat, regex=True, na=False).to_pandas()
        e = pdf["s"].str.contains(pat, regex=True, na=False)
        np.testing.assert_array_equal(g.to_numpy().astype(bool),
                                      e.to_numpy(), err_msg=pat)
    g = df["s"].str.fullmatch(r"\w+", na=False).to_pandas()
    e = pdf["s"].str.fullmatch(r"\w+", na=False)
    np.testing.assert_array_equal(g.to_numpy().astype(bool), e.to_numpy())
    for op in ("strip", "lstrip", "rstrip", "title", "capitalize"):
        g = getattr(df["s"].str, op)().to_pandas().to_numpy()
        e = getattr(pdf["s"].str, op)().to_numpy()
        same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
        assert same.all(), op
    g = df["num"].str.zfill(5).to_pandas().to_numpy()
    e = pdf["num"].str.zfill(5).to_numpy()
    np.testing.assert_array_equal(g, e)
    g = df["s"].str.replace("a", "_", regex=False).to_pandas().to_numpy()
    e = pdf["s"].str.replace("a", "_", regex=False).to_numpy()
    same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
    assert same.all()
    g = df[["s", "v"]].shift(-2).to_pandas()
    e = pdf[["s", "v"]].shift(-2)
    gs, es = g["s"].to_numpy(), e["s"].to_numpy()
    same = (pandas.isna(gs) & pandas.isna(es)) | (gs == es)
    assert same.all()
    g = df["num"].astype(np.float64).to_pandas()
    np.testing.assert_allclose(g.to_numpy(),
                               pdf["num"].astype(np.float64).to_numpy(),
                               rtol=0)
    g = df["num"].astype(np.float64).to_pandas()
    m = df["v"].where(df["v"] > 1).isin([0.5, np.nan]).to_pandas()
    em = pdf["v"].where(pdf["v"] > 1).isin([0.5, np.nan])
    np.testing.assert_array_equal(m.to_numpy().astype(bool),
                                  em.to_numpy())


def test_where_string_fill_vs_pandas(npartitions):
    """where/mask with a string fill over dictionary columns: dictionary
    union + device code blend."""
    rng = np.random.default_rng(143)
    n = 40_000
    pdf = pandas.DataFrame({"s": rng.choice(["aa", "bb", "cc", None], n),
                            "v": rng.standard_normal(n)})
    df = mpd.DataFrame(pdf)
    m, pm = df["v"] > 0, pdf["v"] > 0
    for fill in ("bb", "new"):
        g = df[["s"]].where(m, fill).to_pandas()
        e = pdf[["s"]].where(pm, fill)
        gs, es = g["s"].to_numpy(), e["s"].to_numpy()
        same = (pandas.isna(gs) & pandas.isna(es)) | (gs == es)
        assert same.all(), fill
    # downstream groupby over the filled column
    d2 = mpd.DataFrame(query_compiler=df[["s"]].where(
        m, "zzz")._query_compiler)
    d2["v"] = df["v"]
    got = d2.groupby("s").count().to_pandas()
    p2 = pdf[["s"]].where(pm, "zzz")
    p2["v"] = pdf["v"]
    exp = p2.groupby("s").count()
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  exp.index.to_numpy())
    np.testing.assert_array_equal(got["v"].to_numpy(),
                                  exp["v"].to_numpy())


def test_get_dummies_vs_pandas(npartitions):
    rng = np.random.default_rng(148)
    n = 40_000
    pdf = pandas.DataFrame({"s": rng.choice(
        ["aa", "bb", "cc", "dd", None], n)})
    df = mpd.DataFrame(pdf)
    got = mpd.get_dummies(df["s"]).to_pandas()
    exp = pandas.get_dummies(pdf["s"])
    assert list(got.columns) == list(exp.columns)
    np.testing.assert_array_equal(got.to_numpy().astype(bool),
                                  exp.to_numpy().astype(bool))
    # dummies sum per row == notna
    s = mpd.get_dummies(df["s"]).sum(axis=1).to_pandas()
    np.testing.assert_array_equal(s.to_numpy().astype(int),
                                  pdf["s"].notna().to_numpy().astype(int))
