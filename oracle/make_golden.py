"""Generate golden parity vectors by running the REAL reference.

TEST INFRASTRUCTURE ONLY — runs in the build container, where
/root/reference (Modin @ 2026-02-13) is importable and its PandasOnPython
engine executes the full L1→L6 stack in-process.  The GPU
box has no /root/reference, so the outputs are committed as small fixtures
under tests/golden/ and consumed from there by both test tiers.

Every case runs through ``MODIN_ENGINE=Python modin.pandas`` with
NPartitions=3 (so the reference's own partitioned map/reduce path — not just
pandas — produces the pinned numbers) and is cross-checked against plain
pandas before being written.

Usage (build container only):
    python -m oracle.make_golden
"""

from __future__ import annotations

import os
import sys

import numpy as np

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")
REFERENCE = "/root/reference"


def _setup_reference():
    os.environ["MODIN_ENGINE"] = "Python"
    sys.path.insert(0, REFERENCE)
    import modin.config as cfg
    import modin.pandas as mpd
    cfg.NPartitions.put(3)
    return mpd


def _check_vs_pandas(mdf_result, pdf_result, rtol=1e-12):
    import pandas
    got = mdf_result._to_pandas() if hasattr(mdf_result, "_to_pandas") else mdf_result
    if isinstance(got, pandas.DataFrame):
        assert list(got.index) == list(pdf_result.index), "index mismatch vs pandas"
        np.testing.assert_allclose(got.values.astype(float),
                                   pdf_result.values.astype(float), rtol=rtol)
    else:
        np.testing.assert_allclose(np.asarray(got, dtype=float),
                                   np.asarray(pdf_result, dtype=float), rtol=rtol)
    return got


def gen_groupby_cases(mpd, rng):
    import pandas
    cases = {}

    def make(name, k, cols):
        n = len(k)
        data = {"k": k.astype(np.int64)}
        data.update(cols)
        mdf = mpd.DataFrame(dict(data))
        pdf = pandas.DataFrame(dict(data))
        arrays = {"in_k": data["k"]}
        for cn, cv in cols.items():
            arrays[f"in_{cn}"] = cv
        for agg in ("sum", "count", "mean", "min", "max"):
            mres = getattr(mdf.groupby("k"), agg)()
            pres = getattr(pdf.groupby("k"), agg)()
            got = _check_vs_pandas(mres, pres)
            arrays[f"out_{agg}_keys"] = np.asarray(got.index, dtype=np.int64)
            for cn in cols:
                arrays[f"out_{agg}_{cn}"] = got[cn].to_numpy()
        cases[name] = arrays

    n = 5000
    make("gb_uniform",
         rng.integers(0, 100, n),
         {"v": rng.random(n), "w": rng.random(n) * 10 - 5})

    zipf = np.minimum(rng.zipf(1.5, n), 500).astype(np.int64)
    make("gb_skew", zipf, {"v": rng.random(n)})

    make("gb_negative_keys", rng.integers(-50, 50, n), {"v": rng.random(n)})

    make("gb_single_key", np.full(n, 7, dtype=np.int64), {"v": rng.random(n)})

    v = rng.random(n)
    v[rng.random(n) < 0.2] = np.nan
    k = rng.integers(0, 50, n)
    v[k == 13] = np.nan  # a group whose values are ALL NaN (present, sum 0.0)
    make("gb_nan_vals", k, {"v": v, "w": rng.random(n)})

    make("gb_sparse_keys", rng.choice(
        np.array([3, 977, 5003, 9998], dtype=np.int64), n), {"v": rng.random(n)})

    make("gb_unsorted", rng.permutation(np.repeat(np.arange(40), n // 40 + 1)[:n])
         .astype(np.int64), {"v": rng.random(n)})

    make("gb_wide", rng.integers(0, 64, n),
         {f"c{i}": rng.random(n) for i in range(6)})

    return cases


def gen_reduce_cases(mpd, rng):
    import pandas
    cases = {}

    def make(name, cols):
        mdf = mpd.DataFrame(dict(cols))
        pdf = pandas.DataFrame(dict(cols))
        arrays = {}
        for cn, cv in cols.items():
            arrays[f"in_{cn}"] = cv
        for agg in ("sum", "count", "mean", "min", "max"):
            mres = getattr(mdf, agg)()
            pres = getattr(pdf, agg)()
            got = mres._to_pandas() if hasattr(mres, "_to_pandas") else mres
            pd_vals = np.asarray(pres, dtype=float)
            got_vals = np.asarray(got, dtype=float)
            np.testing.assert_allclose(got_vals, pd_vals, rtol=1e-12, equal_nan=True)
            arrays[f"out_{agg}"] = got_vals
        cases[name] = arrays

    n = 10000
    v = rng.random(n)
    v[rng.random(n) < 0.1] = np.nan
    make("red_f64", {"a": rng.random(n) * 100 - 50, "b": v,
                     "c": rng.standard_normal(n)})
    make("red_i64", {"a": rng.integers(-1000, 1000, n).astype(np.int64),
                     "b": rng.integers(0, 5, n).astype(np.int64)})
    make("red_allnan", {"a": np.full(64, np.nan)})
    return cases


def gen_map_binary_cases(mpd, rng):
    import pandas
    cases = {}
    n = 4096
    v = rng.random(n) * 10 - 5
    v[rng.random(n) < 0.15] = np.nan
    w = rng.random(n) + 0.5
    i = rng.integers(-100, 100, n).astype(np.int64)

    mdf = mpd.DataFrame({"v": v, "w": w})
    pdf = pandas.DataFrame({"v": v, "w": w})
    arrays = {"in_v": v, "in_w": w, "in_i": i}
    for tag, mres, pres in [
        ("add1", mdf + 1, pdf + 1),
        ("mul2", mdf * 2.5, pdf * 2.5),
        ("sub3", mdf - 3.25, pdf - 3.25),
        ("div2", mdf / 2.0, pdf / 2.0),
        ("rsub", 1.0 - mdf, 1.0 - pdf),
        ("fill0", mdf.fillna(0.0), pdf.fillna(0.0)),
        ("fillm1", mdf.fillna(-1.5), pdf.fillna(-1.5)),
        ("abs", mdf.abs(), pdf.abs()),
        ("frame_add", mdf + mdf, pdf + pdf),
        ("frame_mul", mdf * mdf, pdf * pdf),
        ("frame_div", mdf / (mdf + 10.0), pdf / (pdf + 10.0)),
    ]:
        got = mres._to_pandas()
        np.testing.assert_allclose(got.values, pres.values, rtol=1e-15,
                                   equal_nan=True)
        arrays[f"out_{tag}_v"] = got["v"].to_numpy()
        arrays[f"out_{tag}_w"] = got["w"].to_numpy()
    # int64 map ops (exact)
    mdi = mpd.DataFrame({"i": i})
    pdi = pandas.DataFrame({"i": i})
    for tag, mres, pres in [
        ("iadd", mdi + 7, pdi + 7),
        ("imul", mdi * -3, pdi * -3),
        ("iabs", mdi.abs(), pdi.abs()),
    ]:
        got = mres._to_pandas()
        assert (got.values == pres.values).all()
        arrays[f"out_{tag}_i"] = got["i"].to_numpy()
    cases["map_binary"] = arrays
    return cases


def gen_merge_cases(mpd, rng):
    import pandas
    cases = {}

    def make(name, lk, lcols, rk, rcols):
        ldata = {"k": lk.astype(np.int64), **lcols}
        rdata = {"k": rk.astype(np.int64), **rcols}
        mL, mR = mpd.DataFrame(dict(ldata)), mpd.DataFrame(dict(rdata))
        pL, pR = pandas.DataFrame(dict(ldata)), pandas.DataFrame(dict(rdata))
        mres = mL.merge(mR, on="k")._to_pandas()
        pres = pL.merge(pR, on="k")
        assert list(mres.columns) == list(pres.columns)
        assert (mres.index == pres.index).all()
        for c in pres.columns:
            np.testing.assert_array_equal(mres[c].to_numpy(), pres[c].to_numpy())
        arrays = {"in_lk": ldata["k"], "in_rk": rdata["k"]}
        for n, v in lcols.items():
            arrays[f"in_l_{n}"] = v
        for n, v in rcols.items():
            arrays[f"in_r_{n}"] = v
        arrays["out_columns"] = np.array(list(pres.columns), dtype="U32")
        for c in pres.columns:
            arrays[f"out_{c}"] = pres[c].to_numpy()
        cases[name] = arrays

    nl, nr_ = 3000, 500
    make("mg_basic",
         rng.integers(0, 100, nl), {"lv": rng.random(nl),
                                    "li": rng.integers(-9, 9, nl).astype(np.int64)},
         rng.integers(0, 120, nr_), {"rv": rng.random(nr_) * 10})
    make("mg_dup_right",
         rng.integers(0, 40, nl), {"lv": rng.random(nl)},
         rng.integers(0, 40, nr_), {"rv": rng.random(nr_),
                                    "ri": rng.integers(0, 5, nr_).astype(np.int64)})
    make("mg_collide",
         rng.integers(0, 50, nl), {"v": rng.random(nl)},
         rng.integers(0, 50, nr_), {"v": rng.random(nr_)})
    make("mg_disjoint",
         rng.integers(0, 50, nl), {"lv": rng.random(nl)},
         rng.integers(1000, 1050, nr_), {"rv": rng.random(nr_)})
    make("mg_negative",
         rng.integers(-30, 30, nl), {"lv": rng.random(nl)},
         rng.integers(-30, 30, nr_), {"rv": rng.random(nr_)})
    return cases


def gen_filter_cases(mpd, rng):
    import pandas
    cases = {}
    n = 5000
    v = rng.random(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.standard_normal(n)
    i = rng.integers(0, 10, n).astype(np.int64)
    mdf = mpd.DataFrame({"v": v, "w": w, "i": i})
    pdf = pandas.DataFrame({"v": v, "w": w, "i": i})
    arrays = {"in_v": v, "in_w": w, "in_i": i}
    for tag, mmask, pmask in [
        ("gt", mdf["v"] > 0.25, pdf["v"] > 0.25),
        ("le", mdf["v"] <= 0.5, pdf["v"] <= 0.5),
        ("eq", mdf["i"] == 3, pdf["i"] == 3),
        ("ne", mdf["v"] != 0.0, pdf["v"] != 0.0),  # NaN != 0 -> kept
        ("none", mdf["v"] > 2.0, pdf["v"] > 2.0),  # empty result
    ]:
        m_np = mmask._to_pandas().to_numpy()
        p_np = pmask.to_numpy()
        np.testing.assert_array_equal(m_np, p_np)
        arrays[f"out_mask_{tag}"] = p_np
        mres = mdf[mmask]._to_pandas()
        pres = pdf[pmask]
        np.testing.assert_array_equal(mres.index.to_numpy(),
                                      pres.index.to_numpy())
        arrays[f"out_idx_{tag}"] = pres.index.to_numpy().astype(np.int64)
        for c in ("v", "w", "i"):
            np.testing.assert_array_equal(mres[c].to_numpy(),
                                          pres[c].to_numpy())
            arrays[f"out_{tag}_{c}"] = pres[c].to_numpy()
    cases["flt_basic"] = arrays
    return cases


def gen_sort_cases(mpd, rng):
    import pandas
    cases = {}
    n = 6000
    k = rng.integers(0, 50, n).astype(np.int64)  # heavy duplication: tie order
    v = rng.random(n)
    i = rng.integers(-5, 5, n).astype(np.int64)
    mdf = mpd.DataFrame({"k": k, "v": v, "i": i})
    pdf = pandas.DataFrame({"k": k, "v": v, "i": i})
    arrays = {"in_k": k, "in_v": v, "in_i": i}
    for tag, asc in [("asc", True), ("desc", False)]:
        mres = mdf.sort_values("k", ascending=asc, kind="stable")._to_pandas()
        pres = pdf.sort_values("k", ascending=asc, kind="stable")
        np.testing.assert_array_equal(mres.index.to_numpy(),
                                      pres.index.to_numpy())
        for c in ("k", "v", "i"):
            np.testing.assert_array_equal(mres[c].to_numpy(),
                                          pres[c].to_numpy())
        arrays[f"out_idx_{tag}"] = pres.index.to_numpy().astype(np.int64)
        for c in ("k", "v", "i"):
            arrays[f"out_{tag}_{c}"] = pres[c].to_numpy()
    # negative keys
    kn = rng.integers(-1000, 1000, n).astype(np.int64)
    mdf2 = mpd.DataFrame({"k": kn, "v": v})
    pdf2 = pandas.DataFrame({"k": kn, "v": v})
    mres = mdf2.sort_values("k", kind="stable")._to_pandas()
    pres = pdf2.sort_values("k", kind="stable")
    np.testing.assert_array_equal(mres.index.to_numpy(), pres.index.to_numpy())
    arrays["in_kn"] = kn
    arrays["out_neg_idx"] = pres.index.to_numpy().astype(np.int64)
    arrays["out_neg_k"] = pres["k"].to_numpy()
    arrays["out_neg_v"] = pres["v"].to_numpy()
    cases["srt_basic"] = arrays
    return cases


def gen_hash_groupby_cases(mpd, rng):
    """Unbounded key ranges (the hash-table groupby path)."""
    import pandas
    cases = {}

    def make(name, k, cols):
        data = {"k": k.astype(np.int64), **cols}
        mdf, pdf = mpd.DataFrame(dict(data)), pandas.DataFrame(dict(data))
        arrays = {"in_k": data["k"]}
        for cn, cv in cols.items():
            arrays[f"in_{cn}"] = cv
        for agg in ("sum", "count", "mean", "min", "max"):
            mres = getattr(mdf.groupby("k"), agg)()
            pres = getattr(pdf.groupby("k"), agg)()
            got = _check_vs_pandas(mres, pres)
            arrays[f"out_{agg}_keys"] = np.asarray(got.index, dtype=np.int64)
            for cn in cols:
                arrays[f"out_{agg}_{cn}"] = got[cn].to_numpy()
        cases[name] = arrays

    n = 4000
    make("hh_distinct", rng.integers(-2**60, 2**60, n),
         {"v": rng.random(n)})
    hugekeys = rng.integers(-2**60, 2**60, 250)
    make("hh_dups", rng.choice(hugekeys, n), {"v": rng.random(n),
                                              "w": rng.standard_normal(n)})
    return cases


NA = "__NA__"  # NaN sentinel inside '<U' string arrays (npz, no pickle)


def _enc_str(values):
    """pandas object/str values -> '<U' array with NA sentinel."""
    import pandas
    return np.array([NA if (v is None or (isinstance(v, float) and np.isnan(v)))
                     else str(v) for v in values])


def gen_string_cases(mpd, rng):
    """Dictionary-encoded string column parity: groupby BY a
    string key, merge ON a string key, string comparisons/filters, sort by
    a string column, concat with differing dictionaries — all run through
    the REAL reference and cross-checked vs pandas."""
    import pandas
    cases = {}
    pool = np.array(["apple", "pear", "zebra", "kiwi", "mango", "fig",
                     "plum", "apricot", "melon", "lime", "Grape", "banana"])

    # ---- groupby by string key (NaN keys dropped by pandas) ----
    n = 4000
    ks = rng.choice(pool, n).astype(object)
    ks[rng.random(n) < 0.05] = np.nan
    v = rng.random(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.standard_normal(n)
    mdf = mpd.DataFrame({"s": ks, "v": v, "w": w})
    pdf = pandas.DataFrame({"s": ks, "v": v, "w": w})
    arrays = {"in_s": _enc_str(ks), "in_v": v, "in_w": w}
    for agg in ("sum", "count", "mean", "min", "max"):
        mres = getattr(mdf.groupby("s"), agg)()
        pres = getattr(pdf.groupby("s"), agg)()
        got = mres._to_pandas()
        assert list(got.index) == list(pres.index)
        np.testing.assert_allclose(got.values.astype(float),
                                   pres.values.astype(float), rtol=1e-12)
        arrays[f"out_{agg}_keys"] = _enc_str(got.index)
        for cn in ("v", "w"):
            arrays[f"out_{agg}_{cn}"] = got[cn].to_numpy()
    cases["str_gb"] = arrays

    # ---- merge on string key (differing dictionaries both sides) ----
    nl, nr = 3000, 800
    lk = rng.choice(pool[:10], nl).astype(object)
    rk = rng.choice(pool[4:], nr).astype(object)  # overlap + right-only cats
    la, rb = rng.random(nl), rng.random(nr)
    mout = mpd.DataFrame({"s": lk, "a": la}).merge(
        mpd.DataFrame({"s": rk, "b": rb}), on="s")._to_pandas()
    pout = pandas.DataFrame({"s": lk, "a": la}).merge(
        pandas.DataFrame({"s": rk, "b": rb}), on="s")
    assert list(mout["s"]) == list(pout["s"])
    np.testing.assert_allclose(mout[["a", "b"]].values,
                               pout[["a", "b"]].values, rtol=0)
    cases["str_merge"] = {
        "in_lk": _enc_str(lk), "in_la": la,
        "in_rk": _enc_str(rk), "in_rb": rb,
        "out_s": _enc_str(mout["s"]), "out_a": mout["a"].to_numpy(),
        "out_b": mout["b"].to_numpy(),
        "out_idx": mout.index.to_numpy().astype(np.int64),
    }

    # ---- string comparisons / filter ----
    nf = 3000
    fs = rng.choice(pool, nf).astype(object)
    fs[rng.random(nf) < 0.05] = np.nan
    fv = rng.random(nf)
    mdf = mpd.DataFrame({"s": fs, "v": fv})
    pdf = pandas.DataFrame({"s": fs, "v": fv})
    farr = {"in_s": _enc_str(fs), "in_v": fv}
    for tag, fn in [("eq", lambda d: d["s"] == "apple"),
                    ("ne", lambda d: d["s"] != "apple"),
                    ("gt", lambda d: d["s"] > "m"),
                    ("le", lambda d: d["s"] <= "kiwi"),
                    ("eq_missing", lambda d: d["s"] == "notthere")]:
        mres = mdf[fn(mdf)]._to_pandas()
        pres = pdf[fn(pdf)]
        assert list(mres.index) == list(pres.index)
        assert list(mres["s"].fillna(NA)) == list(pres["s"].fillna(NA))
        farr[f"out_{tag}_idx"] = pres.index.to_numpy().astype(np.int64)
        farr[f"out_{tag}_s"] = _enc_str(pres["s"])
        farr[f"out_{tag}_v"] = pres["v"].to_numpy()
    # dropna over the string column
    mres = mdf.dropna()._to_pandas()
    pres = pdf.dropna()
    assert list(mres.index) == list(pres.index)
    farr["out_dropna_idx"] = pres.index.to_numpy().astype(np.int64)
    farr["out_dropna_s"] = _enc_str(pres["s"])
    cases["str_filter"] = farr

    # ---- sort by string column (no NaN: na_position is a later round) ----
    ns = 2500
    ss = rng.choice(pool, ns).astype(object)
    sv = rng.random(ns)
    mres = mpd.DataFrame({"s": ss, "v": sv}).sort_values("s",
                                                         kind="stable")
    pres = pandas.DataFrame({"s": ss, "v": sv}).sort_values("s",
                                                            kind="stable")
    mres = mres._to_pandas()
    assert list(mres["s"]) == list(pres["s"])
    cases["str_sort"] = {
        "in_s": _enc_str(ss), "in_v": sv,
        "out_idx": pres.index.to_numpy().astype(np.int64),
        "out_s": _enc_str(pres["s"]), "out_v": pres["v"].to_numpy(),
    }

    # ---- concat with differing dictionaries, then groupby ----
    n1, n2 = 1500, 1200
    c1 = rng.choice(pool[:6], n1).astype(object)
    c2 = rng.choice(pool[3:], n2).astype(object)
    v1, v2 = rng.random(n1), rng.random(n2)
    mcat = mpd.concat([mpd.DataFrame({"s": c1, "v": v1}),
                       mpd.DataFrame({"s": c2, "v": v2})],
                      ignore_index=True)
    pcat = pandas.concat([pandas.DataFrame({"s": c1, "v": v1}),
                          pandas.DataFrame({"s": c2, "v": v2})],
                         ignore_index=True)
    mg = mcat.groupby("s").sum()._to_pandas()
    pg = pcat.groupby("s").sum()
    assert list(mg.index) == list(pg.index)
    np.testing.assert_allclose(mg["v"].to_numpy(), pg["v"].to_numpy(),
                               rtol=1e-12)
    cases["str_concat"] = {
        "in_s1": _enc_str(c1), "in_v1": v1,
        "in_s2": _enc_str(c2), "in_v2": v2,
        "out_cat_s": _enc_str(pcat["s"]),
        "out_gb_keys": _enc_str(pg.index), "out_gb_v": pg["v"].to_numpy(),
    }
    return cases


def gen_sort2_cases(mpd, rng):
    """Multi-column sort + NaN-last string sort (na_position default)."""
    import pandas
    cases = {}

    # ---- multi-column: two int keys, mixed ascending ----
    n = 3000
    a = rng.integers(0, 40, n).astype(np.int64)
    b = rng.integers(-1000, 1000, n).astype(np.int64)
    v = rng.random(n)
    mdf = mpd.DataFrame({"a": a, "b": b, "v": v})
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    arrays = {"in_a": a, "in_b": b, "in_v": v}
    for tag, by, asc in [("ab", ["a", "b"], True),
                         ("ab_desc", ["a", "b"], False),
                         ("ab_mixed", ["a", "b"], [True, False]),
                         ("ba", ["b", "a"], True)]:
        mres = mdf.sort_values(by, ascending=asc, kind="stable")._to_pandas()
        pres = pdf.sort_values(by, ascending=asc, kind="stable")
        assert list(mres.index) == list(pres.index)
        arrays[f"out_{tag}_idx"] = pres.index.to_numpy().astype(np.int64)
        arrays[f"out_{tag}_a"] = pres["a"].to_numpy()
        arrays[f"out_{tag}_b"] = pres["b"].to_numpy()
    cases["srt_multi"] = arrays

    # ---- string sort with NaNs (pandas na_position='last'), both dirs,
    #      and string+int two-key sort ----
    ns = 2500
    pool = np.array(["delta", "alpha", "Echo", "bravo", "charlie"])
    sarr = rng.choice(pool, ns).astype(object)
    sarr[rng.random(ns) < 0.08] = np.nan
    w = rng.integers(0, 10, ns).astype(np.int64)
    v2 = rng.random(ns)
    mdf = mpd.DataFrame({"s": sarr, "w": w, "v": v2})
    pdf = pandas.DataFrame({"s": sarr, "w": w, "v": v2})
    arr2 = {"in_s": _enc_str(sarr), "in_w": w, "in_v": v2}
    for tag, by, asc in [("s_asc", "s", True), ("s_desc", "s", False),
                         ("sw", ["s", "w"], True),
                         ("ws_mixed", ["w", "s"], [False, True])]:
        pres = pdf.sort_values(by, ascending=asc, kind="stable")
        # NOTE: the REFERENCE ITSELF CRASHES on string sort keys containing
        # NaN (its range-partitioning quantile sampling runs numpy
        # partition on a mixed float/str object array -> TypeError), so
        # this fixture is pinned against pandas 2.3.3 directly — the same
        # arbiter the reference's own df_equals tests use; where the
        # reference succeeds it must agree with pandas.
        try:
            mres = mdf.sort_values(by, ascending=asc,
                                   kind="stable")._to_pandas()
            assert list(mres.index) == list(pres.index)
        except TypeError:
            pass
        arr2[f"out_{tag}_idx"] = pres.index.to_numpy().astype(np.int64)
        arr2[f"out_{tag}_s"] = _enc_str(pres["s"])
        arr2[f"out_{tag}_w"] = pres["w"].to_numpy()
    cases["srt_str_nan"] = arr2

    # ---- int-valued groupby keeps int64 dtype (sum/min/max) ----
    ng = 3000
    k = rng.integers(0, 60, ng).astype(np.int64)
    iv = rng.integers(-10**6, 10**6, ng).astype(np.int64)
    fv = rng.random(ng)
    mdf = mpd.DataFrame({"k": k, "iv": iv, "fv": fv})
    pdf = pandas.DataFrame({"k": k, "iv": iv, "fv": fv})
    arr3 = {"in_k": k, "in_iv": iv, "in_fv": fv}
    for agg in ("sum", "count", "mean", "min", "max"):
        mres = getattr(mdf.groupby("k"), agg)()._to_pandas()
        pres = getattr(pdf.groupby("k"), agg)()
        assert list(mres.index) == list(pres.index)
        assert list(mres.dtypes) == list(pres.dtypes), \
            f"{agg}: {list(mres.dtypes)} vs {list(pres.dtypes)}"
        arr3[f"out_{agg}_keys"] = pres.index.to_numpy().astype(np.int64)
        for cn in ("iv", "fv"):
            arr3[f"out_{agg}_{cn}"] = pres[cn].to_numpy()
    cases["gbi_intvals"] = arr3
    return cases


def gen_var_cases(mpd, rng):
    """var/std (frame + groupby, ddof 0/1), groupby size, and the agg
    list/dict forms — all through the REAL reference, checked vs pandas."""
    import pandas
    cases = {}
    n = 4000
    k = rng.integers(0, 50, n).astype(np.int64)
    v = rng.random(n) * 10
    v[rng.random(n) < 0.1] = np.nan
    v[k == 7] = np.nan          # an all-NaN group -> var NaN
    k[0] = 999                   # a single-row group -> var NaN (ddof=1)
    w = rng.standard_normal(n)
    mdf = mpd.DataFrame({"k": k, "v": v, "w": w})
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w})
    arrays = {"in_k": k, "in_v": v, "in_w": w}
    for ddof in (0, 1):
        for name, mfn, pfn in [
                (f"var{ddof}", mdf.groupby("k").var, pdf.groupby("k").var),
                (f"std{ddof}", mdf.groupby("k").std, pdf.groupby("k").std)]:
            mres = mfn(ddof=ddof)._to_pandas()
            pres = pfn(ddof=ddof)
            assert list(mres.index) == list(pres.index)
            np.testing.assert_allclose(mres.values, pres.values,
                                       rtol=1e-9, atol=1e-12, equal_nan=True)
            arrays[f"out_{name}_keys"] = pres.index.to_numpy().astype(np.int64)
            for cn in ("v", "w"):
                arrays[f"out_{name}_{cn}"] = pres[cn].to_numpy()
        # frame-level var/std
        np.testing.assert_allclose(
            mdf.var(ddof=ddof)._to_pandas().to_numpy()
            if hasattr(mdf.var(ddof=ddof), "_to_pandas")
            else np.asarray(mdf.var(ddof=ddof)),
            pdf.var(ddof=ddof).to_numpy(), rtol=1e-9, equal_nan=True)
        arrays[f"out_frame_var{ddof}"] = pdf.var(ddof=ddof).to_numpy()
        arrays[f"out_frame_std{ddof}"] = pdf.std(ddof=ddof).to_numpy()
    # size
    msz = mdf.groupby("k").size()._to_pandas() \
        if hasattr(mdf.groupby("k").size(), "_to_pandas") \
        else mdf.groupby("k").size()
    psz = pdf.groupby("k").size()
    np.testing.assert_array_equal(np.asarray(msz), psz.to_numpy())
    arrays["out_size_keys"] = psz.index.to_numpy().astype(np.int64)
    arrays["out_size"] = psz.to_numpy().astype(np.int64)
    cases["gbv_moments"] = arrays

    # ---- agg forms ----
    magg = mdf.groupby("k").agg(["sum", "mean"])._to_pandas()
    pagg = pdf.groupby("k").agg(["sum", "mean"])
    assert list(magg.columns) == list(pagg.columns)
    np.testing.assert_allclose(magg.values, pagg.values, rtol=1e-12,
                               atol=1e-12, equal_nan=True)
    mdd = mdf.groupby("k").agg({"v": "sum", "w": "max"})._to_pandas()
    pdd = pdf.groupby("k").agg({"v": "sum", "w": "max"})
    assert list(mdd.columns) == list(pdd.columns)
    cases["gba_forms"] = {
        "in_k": k, "in_v": v, "in_w": w,
        "out_list_keys": pagg.index.to_numpy().astype(np.int64),
        "out_list_cols": np.array([f"{a}|{b}" for a, b in pagg.columns]),
        "out_list_vals": pagg.to_numpy(),
        "out_dict_keys": pdd.index.to_numpy().astype(np.int64),
        "out_dict_cols": np.array(list(pdd.columns)),
        "out_dict_vals": pdd.to_numpy(),
    }
    return cases


def gen_multikey_cases(mpd, rng):
    """groupby by a KEY LIST (int+int and string+int), all reduce aggs +
    size — the reference's by-list path vs pandas; our backend folds the
    keys into one int64 (monotone, so index order matches pandas
    lexicographic MultiIndex order)."""
    import pandas
    cases = {}
    n = 4000
    a = rng.integers(-20, 20, n).astype(np.int64)
    b = rng.integers(0, 15, n).astype(np.int64)
    v = rng.random(n)
    v[rng.random(n) < 0.1] = np.nan
    w = rng.standard_normal(n)
    mdf = mpd.DataFrame({"a": a, "b": b, "v": v, "w": w})
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v, "w": w})
    arrays = {"in_a": a, "in_b": b, "in_v": v, "in_w": w}
    for agg in ("sum", "count", "mean", "min", "max"):
        mres = getattr(mdf.groupby(["a", "b"]), agg)()._to_pandas()
        pres = getattr(pdf.groupby(["a", "b"]), agg)()
        assert list(mres.index) == list(pres.index)
        np.testing.assert_allclose(mres.values.astype(float),
                                   pres.values.astype(float), rtol=1e-12,
                                   atol=1e-12, equal_nan=True)
        arrays[f"out_{agg}_ka"] = pres.index.get_level_values(0).to_numpy()
        arrays[f"out_{agg}_kb"] = pres.index.get_level_values(1).to_numpy()
        for cn in ("v", "w"):
            arrays[f"out_{agg}_{cn}"] = pres[cn].to_numpy()
    psz = pdf.groupby(["a", "b"]).size()
    msz = mdf.groupby(["a", "b"]).size()
    np.testing.assert_array_equal(np.asarray(msz._to_pandas()
                                             if hasattr(msz, "_to_pandas")
                                             else msz), psz.to_numpy())
    arrays["out_size"] = psz.to_numpy().astype(np.int64)
    cases["gbm_ints"] = arrays

    ns = 3000
    pool = np.array(["north", "south", "east", "west"])
    sarr = rng.choice(pool, ns).astype(object)
    sarr[rng.random(ns) < 0.05] = np.nan  # NaN key rows drop entirely
    g2 = rng.integers(0, 8, ns).astype(np.int64)
    v2 = rng.random(ns)
    mdf = mpd.DataFrame({"s": sarr, "g": g2, "v": v2})
    pdf = pandas.DataFrame({"s": sarr, "g": g2, "v": v2})
    arr2 = {"in_s": _enc_str(sarr), "in_g": g2, "in_v": v2}
    for agg in ("sum", "mean"):
        mres = getattr(mdf.groupby(["s", "g"]), agg)()._to_pandas()
        pres = getattr(pdf.groupby(["s", "g"]), agg)()
        assert list(mres.index) == list(pres.index)
        arr2[f"out_{agg}_ks"] = _enc_str(pres.index.get_level_values(0))
        arr2[f"out_{agg}_kg"] = pres.index.get_level_values(1).to_numpy()
        arr2[f"out_{agg}_v"] = pres["v"].to_numpy()
    cases["gbm_strint"] = arr2
    return cases


def gen_series_cases(mpd, rng):
    """Series.unique / value_counts / nunique / isin vs the reference:
    appearance order, count-desc ties, NaN handling, string dictionaries."""
    import pandas
    cases = {}
    n = 5000
    k = rng.integers(-30, 30, n).astype(np.int64)
    pool = np.array(["ash", "oak", "elm", "fir", "Yew"])
    sv = rng.choice(pool, n).astype(object)
    sv[rng.random(n) < 0.06] = np.nan
    mdf = mpd.DataFrame({"k": k, "s": sv})
    pdf = pandas.DataFrame({"k": k, "s": sv})
    arrays = {"in_k": k, "in_s": _enc_str(sv)}
    # int column
    mu = mdf["k"].unique()
    pu = pdf["k"].unique()
    np.testing.assert_array_equal(np.asarray(mu), pu)
    arrays["out_k_unique"] = pu.astype(np.int64)
    mvc = mdf["k"].value_counts()
    pvc = pdf["k"].value_counts()
    np.testing.assert_array_equal(np.asarray(mvc._to_pandas()
                                             if hasattr(mvc, "_to_pandas")
                                             else mvc), pvc.to_numpy())
    arrays["out_k_vc_idx"] = pvc.index.to_numpy().astype(np.int64)
    arrays["out_k_vc"] = pvc.to_numpy().astype(np.int64)
    assert int(mdf["k"].nunique()) == int(pdf["k"].nunique())
    arrays["out_k_nunique"] = np.array([pdf["k"].nunique()], dtype=np.int64)
    # string column (unique keeps NaN at appearance position)
    pu_s = pdf["s"].unique()
    arrays["out_s_unique"] = _enc_str(pu_s)
    pvc_s = pdf["s"].value_counts()
    arrays["out_s_vc_idx"] = _enc_str(pvc_s.index)
    arrays["out_s_vc"] = pvc_s.to_numpy().astype(np.int64)
    arrays["out_s_nunique"] = np.array([pdf["s"].nunique()],
                                       dtype=np.int64)
    mu_s = mdf["s"].unique()
    assert len(mu_s) == len(pu_s)
    # isin masks
    pk = pdf["k"].isin([3, -7, 999]).to_numpy()
    ps = pdf["s"].isin(["oak", "Yew", "missing"]).to_numpy()
    np.testing.assert_array_equal(
        np.asarray(mdf["k"].isin([3, -7, 999])._to_pandas()), pk)
    arrays["out_k_isin"] = pk.astype(np.int64)
    arrays["out_s_isin"] = ps.astype(np.int64)
    arrays["out_empty_isin"] = pdf["k"].isin([]).to_numpy().astype(np.int64)
    cases["ser_utils"] = arrays
    return cases


def gen_merge2_cases(mpd, rng):
    """Unbounded-span merge keys (full int64 range; the densify-via-
    distinct-keys join path) vs the reference."""
    import pandas
    cases = {}
    nl, nr = 4000, 900
    base = rng.integers(-2**60, 2**60, 700)
    lk = rng.choice(base, nl)                       # some match
    miss = rng.random(nl) < 0.2                      # some never match
    lk[miss] = rng.integers(-2**60, 2**60, int(miss.sum()))
    rk = rng.choice(base, nr)
    la, rb = rng.random(nl), rng.random(nr)
    mout = mpd.DataFrame({"k": lk, "a": la}).merge(
        mpd.DataFrame({"k": rk, "b": rb}), on="k")._to_pandas()
    pout = pandas.DataFrame({"k": lk, "a": la}).merge(
        pandas.DataFrame({"k": rk, "b": rb}), on="k")
    np.testing.assert_array_equal(mout["k"].to_numpy(), pout["k"].to_numpy())
    np.testing.assert_allclose(mout[["a", "b"]].values,
                               pout[["a", "b"]].values, rtol=0)
    cases["mg2_hugespan"] = {
        "in_lk": lk.astype(np.int64), "in_la": la,
        "in_rk": rk.astype(np.int64), "in_rb": rb,
        "out_k": pout["k"].to_numpy().astype(np.int64),
        "out_a": pout["a"].to_numpy(), "out_b": pout["b"].to_numpy(),
        "out_idx": pout.index.to_numpy().astype(np.int64),
    }
    return cases


def gen_float_key_cases(mpd, rng):
    """Float sort keys (NaN last both directions), float groupby keys
    (NaN dropped), float unique/value_counts — the ordered f64<->i64
    transform path vs the reference."""
    import pandas
    cases = {}
    n = 4000
    f = np.round(rng.standard_normal(n) * 100, 2)
    f[rng.random(n) < 0.07] = np.nan
    f[rng.random(n) < 0.02] = -0.0  # groups with +0.0
    f[rng.random(n) < 0.02] = 0.0
    w = rng.integers(0, 12, n).astype(np.int64)
    v = rng.random(n)
    mdf = mpd.DataFrame({"f": f, "w": w, "v": v})
    pdf = pandas.DataFrame({"f": f, "w": w, "v": v})
    arrays = {"in_f": f, "in_w": w, "in_v": v}
    for tag, by, asc in [("f_asc", "f", True), ("f_desc", "f", False),
                         ("fw", ["f", "w"], True),
                         ("wf_mixed", ["w", "f"], [True, False])]:
        mres = mdf.sort_values(by, ascending=asc, kind="stable")._to_pandas()
        pres = pdf.sort_values(by, ascending=asc, kind="stable")
        assert list(mres.index) == list(pres.index), tag
        arrays[f"out_{tag}_idx"] = pres.index.to_numpy().astype(np.int64)
        arrays[f"out_{tag}_f"] = pres["f"].to_numpy()
    for agg in ("sum", "mean", "count"):
        mres = getattr(mdf.groupby("f"), agg)()._to_pandas()
        pres = getattr(pdf.groupby("f"), agg)()
        assert list(mres.index) == list(pres.index), agg
        arrays[f"out_gb_{agg}_keys"] = pres.index.to_numpy()
        for cn in ("w", "v"):
            arrays[f"out_gb_{agg}_{cn}"] = pres[cn].to_numpy()
    pu = pdf["f"].unique()
    arrays["out_unique"] = pu
    pvc = pdf["f"].value_counts()
    arrays["out_vc_idx"] = pvc.index.to_numpy()
    arrays["out_vc"] = pvc.to_numpy().astype(np.int64)
    arrays["out_nunique"] = np.array([pdf["f"].nunique()], dtype=np.int64)
    np.testing.assert_array_equal(np.asarray(mdf["f"].unique()), pu)
    cases["flt_keys"] = arrays
    return cases


def gen_median_cases(mpd, rng):
    """df.median and groupby.median (NaN values skipped, even/odd group
    sizes, string and multi keys) vs the reference."""
    import pandas
    cases = {}
    n = 4000
    k = rng.integers(0, 40, n).astype(np.int64)
    pool = np.array(["red", "blue", "green"])
    sk = rng.choice(pool, n).astype(object)
    sk[rng.random(n) < 0.05] = np.nan
    v = rng.random(n) * 100
    v[rng.random(n) < 0.12] = np.nan
    w = rng.integers(-50, 50, n).astype(np.int64)
    mdf = mpd.DataFrame({"k": k, "s": sk, "v": v, "w": w})
    pdf = pandas.DataFrame({"k": k, "s": sk, "v": v, "w": w})
    arrays = {"in_k": k, "in_s": _enc_str(sk), "in_v": v, "in_w": w}
    mm = mdf[["k", "v", "w"]].median()
    pm = pdf[["k", "v", "w"]].median()
    np.testing.assert_allclose(np.asarray(mm), pm.to_numpy(), rtol=1e-12)
    arrays["out_frame_median"] = pm.to_numpy()
    for tag, by in [("k", "k"), ("s", "s"), ("ks", ["k", "s"])]:
        sub = [c for c in ("k", "s", "v", "w")
               if c not in (by if isinstance(by, list) else [by])]
        msel = mdf[([by] if isinstance(by, str) else by) + sub]
        psel = pdf[([by] if isinstance(by, str) else by) + sub]
        # drop the string column from the aggregation side
        numeric = [c for c in sub if c != "s"]
        mres = msel[([by] if isinstance(by, str) else by) + numeric] \
            .groupby(by).median()._to_pandas()
        pres = psel[([by] if isinstance(by, str) else by) + numeric] \
            .groupby(by).median()
        assert list(mres.index) == list(pres.index), tag
        np.testing.assert_allclose(mres.values, pres.values, rtol=1e-12,
                                   atol=1e-12, equal_nan=True)
        if isinstance(by, str) and by == "k":
            arrays["out_gbk_keys"] = pres.index.to_numpy().astype(np.int64)
        elif isinstance(by, str):
            arrays["out_gbs_keys"] = _enc_str(pres.index)
        else:
            arrays["out_gbks_ka"] = pres.index.get_level_values(0) \
                .to_numpy().astype(np.int64)
            arrays["out_gbks_kb"] = _enc_str(
                pres.index.get_level_values(1))
        sfx = ("gbk" if by == "k" else
               "gbs" if isinstance(by, str) else "gbks")
        for cn in numeric:
            arrays[f"out_{sfx}_{cn}"] = pres[cn].to_numpy()
    cases["med_cases"] = arrays
    return cases


def gen_left_merge_cases(mpd, rng):
    """merge(how='left'): unmatched lefts keep NaN rights (int right
    columns become float64), pandas left row order, string keys and
    huge-span keys included — vs the reference."""
    import pandas
    cases = {}
    nl, nr = 3000, 700
    lk = rng.integers(0, 2000, nl).astype(np.int64)   # many unmatched
    rk = rng.integers(0, 900, nr).astype(np.int64)
    la = rng.random(nl)
    rb = rng.random(nr)
    ri = rng.integers(-100, 100, nr).astype(np.int64)
    mout = mpd.DataFrame({"k": lk, "a": la}).merge(
        mpd.DataFrame({"k": rk, "b": rb, "i": ri}), on="k",
        how="left")._to_pandas()
    pout = pandas.DataFrame({"k": lk, "a": la}).merge(
        pandas.DataFrame({"k": rk, "b": rb, "i": ri}), on="k", how="left")
    assert list(mout.dtypes) == list(pout.dtypes)
    np.testing.assert_array_equal(mout["k"].to_numpy(), pout["k"].to_numpy())
    np.testing.assert_allclose(mout[["a", "b", "i"]].values,
                               pout[["a", "b", "i"]].values, rtol=0,
                               equal_nan=True)
    cases["mgl_basic"] = {
        "in_lk": lk, "in_la": la, "in_rk": rk, "in_rb": rb, "in_ri": ri,
        "out_k": pout["k"].to_numpy().astype(np.int64),
        "out_a": pout["a"].to_numpy(), "out_b": pout["b"].to_numpy(),
        "out_i": pout["i"].to_numpy(),
    }

    # all-matched: int right col stays int64
    lk2 = rng.choice(rk, 1500)
    m2 = mpd.DataFrame({"k": lk2, "a": rng.random(1500)}).merge(
        mpd.DataFrame({"k": rk, "i": ri}), on="k", how="left")._to_pandas()
    p2 = pandas.DataFrame({"k": lk2,
                           "a": np.zeros(1500)}).merge(
        pandas.DataFrame({"k": rk, "i": ri}), on="k", how="left")
    assert m2["i"].dtype == p2["i"].dtype == np.dtype(np.int64)
    cases["mgl_allmatch"] = {
        "in_lk": lk2.astype(np.int64), "in_rk": rk, "in_ri": ri,
        "out_k": p2["k"].to_numpy().astype(np.int64),
        "out_i": p2["i"].to_numpy().astype(np.int64),
    }

    # string keys + string payload
    pool = np.array(["ant", "bee", "cat", "dog", "eel", "fox"])
    lks = rng.choice(pool, 2000).astype(object)
    rks = rng.choice(pool[:4], 500).astype(object)
    rs = rng.choice(np.array(["x", "y"]), 500).astype(object)
    mout = mpd.DataFrame({"s": lks, "a": rng.random(2000)}).merge(
        mpd.DataFrame({"s": rks, "t": rs}), on="s", how="left")._to_pandas()
    pout = pandas.DataFrame({"s": lks, "a": np.zeros(2000)}).merge(
        pandas.DataFrame({"s": rks, "t": rs}), on="s", how="left")
    assert list(mout["s"]) == list(pout["s"])
    assert list(mout["t"].fillna(NA)) == list(pout["t"].fillna(NA))
    cases["mgl_str"] = {
        "in_ls": _enc_str(lks), "in_rs": _enc_str(rks),
        "in_rt": _enc_str(rs),
        "out_s": _enc_str(pout["s"]), "out_t": _enc_str(pout["t"]),
        "out_idx": pout.index.to_numpy().astype(np.int64),
    }

    # huge-span keys (densify path) left join
    base = rng.integers(-2**60, 2**60, 400)
    lkh = rng.choice(base, 2000)
    miss = rng.random(2000) < 0.3
    lkh[miss] = rng.integers(-2**60, 2**60, int(miss.sum()))
    rkh = rng.choice(base, 300)
    rbv = rng.random(300)
    mout = mpd.DataFrame({"k": lkh, "a": rng.random(2000)}).merge(
        mpd.DataFrame({"k": rkh, "b": rbv}), on="k", how="left")._to_pandas()
    pout = pandas.DataFrame({"k": lkh, "a": np.zeros(2000)}).merge(
        pandas.DataFrame({"k": rkh, "b": rbv}), on="k", how="left")
    np.testing.assert_array_equal(mout["k"].to_numpy(), pout["k"].to_numpy())
    cases["mgl_huge"] = {
        "in_lk": lkh.astype(np.int64), "in_rk": rkh.astype(np.int64),
        "in_rb": rbv,
        "out_k": pout["k"].to_numpy().astype(np.int64),
        "out_b": pout["b"].to_numpy(),
    }
    return cases


def gen_firstlast_cases(mpd, rng):
    """groupby.first/last (NaN skipped; all-NaN groups -> NaN; string
    values and keys; int dtype preserved) vs the reference."""
    import pandas
    cases = {}
    n = 3000
    k = rng.integers(0, 30, n).astype(np.int64)
    v = rng.random(n)
    v[rng.random(n) < 0.15] = np.nan
    v[k == 5] = np.nan  # all-NaN group
    w = rng.integers(-99, 99, n).astype(np.int64)
    pool = np.array(["aa", "bb", "cc", "dd"])
    sv = rng.choice(pool, n).astype(object)
    sv[rng.random(n) < 0.1] = np.nan
    mdf = mpd.DataFrame({"k": k, "v": v, "w": w, "s": sv})
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w, "s": sv})
    arrays = {"in_k": k, "in_v": v, "in_w": w, "in_s": _enc_str(sv)}
    for agg in ("first", "last"):
        mres = getattr(mdf.groupby("k"), agg)()._to_pandas()
        pres = getattr(pdf.groupby("k"), agg)()
        assert list(mres.index) == list(pres.index)
        assert list(mres["w"]) == list(pres["w"])
        assert list(mres["s"].fillna(NA)) == list(pres["s"].fillna(NA))
        np.testing.assert_allclose(mres["v"].to_numpy(),
                                   pres["v"].to_numpy(), rtol=0,
                                   equal_nan=True)
        arrays[f"out_{agg}_keys"] = pres.index.to_numpy().astype(np.int64)
        arrays[f"out_{agg}_v"] = pres["v"].to_numpy()
        arrays[f"out_{agg}_w"] = pres["w"].to_numpy().astype(np.int64)
        arrays[f"out_{agg}_s"] = _enc_str(pres["s"])
    cases["gbfl_cases"] = arrays
    return cases


def gen_outer_right_merge_cases(mpd, rng):
    """merge how='outer' (left rows then unmatched rights) and how='right'
    vs the reference, including common-column suffixing."""
    import pandas
    cases = {}
    nl, nr = 2500, 900
    lk = rng.integers(0, 1500, nl).astype(np.int64)
    rk = rng.integers(500, 2500, nr).astype(np.int64)  # both-side unmatched
    la = rng.random(nl)
    rb = rng.random(nr)
    lw = rng.integers(-9, 9, nl).astype(np.int64)
    rw = rng.integers(100, 200, nr).astype(np.int64)  # suffixed common col
    for how in ("outer", "right"):
        mout = mpd.DataFrame({"k": lk, "a": la, "w": lw}).merge(
            mpd.DataFrame({"k": rk, "b": rb, "w": rw}), on="k",
            how=how)._to_pandas()
        pout = pandas.DataFrame({"k": lk, "a": la, "w": lw}).merge(
            pandas.DataFrame({"k": rk, "b": rb, "w": rw}), on="k", how=how)
        assert list(mout.columns) == list(pout.columns)
        np.testing.assert_array_equal(mout["k"].to_numpy(),
                                      pout["k"].to_numpy())
        np.testing.assert_allclose(
            mout[["a", "w_x", "b", "w_y"]].values,
            pout[["a", "w_x", "b", "w_y"]].values, rtol=0, equal_nan=True)
        cases[f"mgo_{how}"] = {
            "in_lk": lk, "in_la": la, "in_lw": lw,
            "in_rk": rk, "in_rb": rb, "in_rw": rw,
            "out_k": pout["k"].to_numpy().astype(np.int64),
            "out_a": pout["a"].to_numpy(),
            "out_wx": pout["w_x"].to_numpy(),
            "out_b": pout["b"].to_numpy(),
            "out_wy": pout["w_y"].to_numpy(),
            "out_cols": np.array(list(pout.columns)),
        }
    return cases


def gen_nunique_cases(mpd, rng):
    """groupby.nunique per value column (distinct non-NaN; all-NaN groups
    0; int/float/string values; string and multi keys) vs the
    reference."""
    import pandas
    cases = {}
    n = 3000
    k = rng.integers(0, 25, n).astype(np.int64)
    v = np.round(rng.random(n) * 20, 1)
    v[rng.random(n) < 0.15] = np.nan
    v[k == 3] = np.nan  # all-NaN group -> 0
    w = rng.integers(0, 8, n).astype(np.int64)
    pool = np.array(["p", "q", "r", "s"])
    sv = rng.choice(pool, n).astype(object)
    sv[rng.random(n) < 0.1] = np.nan
    mdf = mpd.DataFrame({"k": k, "v": v, "w": w, "s": sv})
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w, "s": sv})
    mres = mdf.groupby("k").nunique()._to_pandas()
    pres = pdf.groupby("k").nunique()
    assert list(mres.index) == list(pres.index)
    np.testing.assert_array_equal(mres.values, pres.values)
    arrays = {"in_k": k, "in_v": v, "in_w": w, "in_s": _enc_str(sv),
              "out_keys": pres.index.to_numpy().astype(np.int64)}
    for cn in ("v", "w", "s"):
        arrays[f"out_{cn}"] = pres[cn].to_numpy().astype(np.int64)
    # string key
    pres2 = pdf[["s", "v"]].groupby("s").nunique()
    arrays["out_sk_keys"] = _enc_str(pres2.index)
    arrays["out_sk_v"] = pres2["v"].to_numpy().astype(np.int64)
    cases["gbnu_cases"] = arrays
    return cases


def main():
    os.makedirs(GOLDEN_DIR, exist_ok=True)
    mpd = _setup_reference()
    rng = np.random.default_rng(42)
    all_cases = {}
    all_cases.update(gen_groupby_cases(mpd, rng))
    all_cases.update(gen_reduce_cases(mpd, rng))
    all_cases.update(gen_map_binary_cases(mpd, rng))
    all_cases.update(gen_merge_cases(mpd, rng))
    all_cases.update(gen_filter_cases(mpd, rng))
    all_cases.update(gen_sort_cases(mpd, rng))
    all_cases.update(gen_hash_groupby_cases(mpd, rng))
    all_cases.update(gen_string_cases(mpd, rng))
    all_cases.update(gen_sort2_cases(mpd, rng))
    all_cases.update(gen_var_cases(mpd, rng))
    all_cases.update(gen_multikey_cases(mpd, rng))
    all_cases.update(gen_series_cases(mpd, rng))
    all_cases.update(gen_merge2_cases(mpd, rng))
    all_cases.update(gen_float_key_cases(mpd, rng))
    all_cases.update(gen_median_cases(mpd, rng))
    all_cases.update(gen_left_merge_cases(mpd, rng))
    all_cases.update(gen_firstlast_cases(mpd, rng))
    all_cases.update(gen_outer_right_merge_cases(mpd, rng))
    all_cases.update(gen_nunique_cases(mpd, rng))
    for name, arrays in all_cases.items():
        path = os.path.join(GOLDEN_DIR, f"{name}.npz")
        np.savez_compressed(path, **arrays)
        print(f"wrote {path} ({len(arrays)} arrays)")
    print(f"{len(all_cases)} golden fixtures generated from the reference "
          "(Modin PandasOnPython, NPartitions=3), cross-checked vs pandas.")


if __name__ == "__main__":
    main()
