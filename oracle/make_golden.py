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
    for name, arrays in all_cases.items():
        path = os.path.join(GOLDEN_DIR, f"{name}.npz")
        np.savez_compressed(path, **arrays)
        print(f"wrote {path} ({len(arrays)} arrays)")
    print(f"{len(all_cases)} golden fixtures generated from the reference "
          "(Modin PandasOnPython, NPartitions=3), cross-checked vs pandas.")


if __name__ == "__main__":
    main()
