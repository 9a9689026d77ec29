"""GPU tier: randomized cross-operation pipelines vs pandas.

Each case builds a random frame (int64/float64/string columns, NaNs
planted) and runs a random composition of the backend's operations,
comparing every intermediate that materializes against pandas computing
the same chain.  Seeds are FIXED — failures reproduce exactly.  pandas is
the arbiter the reference's own df_equals tests use.
"""

import os

import numpy as np
import pandas
import pytest

import modin_amd.pandas as mpd

pytestmark = pytest.mark.gpu

POOL = np.array(["ash", "birch", "cedar", "dub", "elm", "Fir", "ginkgo"])


@pytest.fixture(autouse=True)
def _ready(gpu_ready):
    yield


def make_frame(rng, n):
    k = rng.integers(0, rng.integers(5, 500), n).astype(np.int64)
    v = rng.standard_normal(n) * rng.integers(1, 50)
    v[rng.random(n) < rng.random() * 0.2] = np.nan
    w = rng.integers(-1000, 1000, n).astype(np.int64)
    s = rng.choice(POOL[: rng.integers(2, len(POOL))], n).astype(object)
    s[rng.random(n) < rng.random() * 0.1] = np.nan
    t = (pandas.Timestamp("2000-01-01").value
         + rng.integers(0, 10**18, n)).astype("datetime64[ns]")
    t[rng.random(n) < rng.random() * 0.1] = np.datetime64("NaT")
    pdf = pandas.DataFrame({"k": k, "v": v, "w": w, "s": s, "t": t})
    return pdf


def check(df, pdf, msg):
    got = df.to_pandas()
    assert list(got.columns) == list(pdf.columns), msg
    np.testing.assert_array_equal(got.index.to_numpy(),
                                  pdf.index.to_numpy(), err_msg=msg)
    for c in pdf.columns:
        g, e = got[c].to_numpy(), pdf[c].to_numpy()
        if e.dtype == object:
            for i, (a, b) in enumerate(zip(g, e)):
                bn = isinstance(b, float) and np.isnan(b)
                gn = isinstance(a, float) and np.isnan(a)
                assert gn == bn and (gn or a == b), f"{msg}/{c}[{i}]"
        elif np.issubdtype(e.dtype, np.datetime64):
            np.testing.assert_array_equal(g, e, err_msg=f"{msg}/{c}")
        else:
            np.testing.assert_allclose(g, e, rtol=1e-12, atol=1e-9,
                                       equal_nan=True,
                                       err_msg=f"{msg}/{c}")


@pytest.mark.parametrize("seed", range(int(os.environ.get(
    "HF_FUZZ_N", "20"))))
def test_fuzz_pipeline(seed):
    import modin_amd.config as _cfg
    rng = np.random.default_rng(1000 + seed)
    n = int(rng.integers(500, 40_000))
    old_np = _cfg.NPartitions.get()
    _cfg.NPartitions.put(int(rng.integers(1, 5)))
    try:
        _fuzz_pipeline_body(rng, n, seed)
    finally:
        _cfg.NPartitions.put(old_np)


def _fuzz_pipeline_body(rng, n, seed):
    pdf = make_frame(rng, n)
    df = mpd.DataFrame(pdf)

    steps = rng.integers(2, 5)
    for si in range(steps):
        op = rng.choice(["filter", "sort", "head", "dropna", "arith",
                         "round", "where", "dedup", "iloc", "strmask",
                         "assign", "mapdict", "query"])
        msg = f"seed {seed} step {si} op {op}"
        if op == "filter":
            thr = float(np.round(rng.standard_normal() * 10, 2))
            df = df[df["v"] > thr]
            pdf = pdf[pdf["v"] > thr]
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "sort":
            by = [["k"], ["s"], ["k", "w"], ["v"], ["s", "k"]][
                rng.integers(0, 5)]
            by = [b for b in by if b in pdf.columns] or ["v"]
            if "s" in by and pdf["s"].isna().any() and len(by) > 1 \
                    and by[0] != "s":
                by = ["k", "w"]
            asc = bool(rng.integers(0, 2))
            df = df.sort_values(by, ascending=asc)
            pdf = pdf.sort_values(by, ascending=asc, kind="stable")
            check(df, pdf, msg)
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "head":
            m = int(rng.integers(1, max(len(pdf), 2)))
            df, pdf = df.head(m), pdf.head(m)
        elif op == "dropna":
            df = df.dropna()
            pdf = pdf.dropna()
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "round":
            d = int(rng.integers(-1, 3))
            sub2 = [c for c in ("k", "v", "w") if c in pdf.columns]
            df = df[sub2].round(d)
            pdf = pdf[sub2].round(d)
        elif op == "where":
            thr = float(np.round(rng.standard_normal() * 5, 2))
            sub2 = [c for c in ("k", "v", "w") if c in pdf.columns]
            df = df[sub2].where(df["v"] > thr)
            pdf = pdf[sub2].where(pdf["v"] > thr)
        elif op == "iloc":
            if rng.integers(0, 2):
                a = int(rng.integers(0, max(len(pdf) - 1, 1)))
                b = int(rng.integers(a, len(pdf) + 1))
                df, pdf = df.iloc[a:b], pdf.iloc[a:b]
            else:
                sel = rng.integers(0, max(len(pdf), 1),
                                   int(rng.integers(1, 2000))).tolist()
                df, pdf = df.iloc[sel], pdf.iloc[sel]
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "strmask":
            if "s" not in pdf.columns:
                continue
            pat = str(rng.choice(["a", "e", "ir", "d"]))
            df = df[df["s"].str.contains(pat, na=False)]
            pdf = pdf[pdf["s"].str.contains(pat, na=False)]
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "assign":
            if not all(c in pdf.columns for c in ("v", "w")):
                continue
            c = float(np.round(rng.standard_normal() * 3, 3))
            pdf = pdf.copy()
            df["q"] = df["v"] * c + df["w"]
            pdf["q"] = pdf["v"] * c + pdf["w"]
            sc = int(rng.integers(-9, 9))
            df["qc"] = sc
            pdf["qc"] = sc
        elif op == "mapdict":
            if "w" not in pdf.columns or pdf["w"].dtype != np.int64:
                continue
            pdf = pdf.copy()
            keys = rng.integers(-1000, 1000, 8)
            rep = {int(a): int(b)
                   for a, b in zip(keys, rng.integers(-5, 5, 8))}
            df["w"] = df["w"].replace(rep)
            pdf["w"] = pdf["w"].replace(rep)
        elif op == "query":
            if not all(c in pdf.columns for c in ("k", "v", "w")):
                continue
            thr = int(rng.integers(-500, 500))
            expr = [f"w > {thr}", f"k < 200 and w != {thr}",
                    f"w > k or v < 0.5",
                    f"not (w > {thr} or v > 1.0)"][rng.integers(0, 4)]
            df = df.query(expr)
            pdf = pdf.query(expr)
            check(df, pdf, msg)
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        elif op == "dedup":
            subs = [["k"], ["k", "s"], ["s", "w"], None][rng.integers(0, 4)]
            if subs is not None:
                subs = [c for c in subs if c in pdf.columns] or None
            df = df.drop_duplicates(subs)
            pdf = pdf.drop_duplicates(subset=subs)
            check(df, pdf, msg)
            df = mpd.DataFrame(df.to_pandas().reset_index(drop=True))
            pdf = pdf.reset_index(drop=True)
        else:
            c = float(np.round(rng.standard_normal(), 3))
            sub = [col for col in ("v", "w") if col in pdf.columns]
            for col in sub:
                got = (df[col] + c).to_pandas()
                exp = pdf[col] + c
                np.testing.assert_allclose(got.to_numpy(),
                                           exp.to_numpy(), rtol=1e-12,
                                           atol=1e-12, equal_nan=True,
                                           err_msg=msg)
        check(df, pdf, msg)
        if len(pdf) == 0:
            break

    # dt fields if the datetime column survived the pipeline
    if len(pdf) and "t" in pdf.columns:
        for f in ("year", "month", "dayofweek"):
            g = np.asarray(getattr(df["t"].dt, f).to_pandas())
            e = getattr(pdf["t"].dt, f).to_numpy()
            np.testing.assert_array_equal(g, e,
                                          err_msg=f"seed {seed} dt.{f}")

    # closing transform on whatever survived (original-row-order family)
    if len(pdf) and all(c in pdf.columns for c in ("s", "v", "w")):
        by = ["k", "s"][rng.integers(0, 2)]
        tr = ["cumsum", "cumcount", "shift", "rank", "tsum",
              "tmean", "cumprod"][rng.integers(0, 7)]
        sub = [by, "v", "w"]
        if tr == "cumprod":
            # bound the running product: once it overflows f64, inf/0/NaN
            # propagation differs between a sequential product (pandas)
            # and the tree-combined scan (documented deviation) — keep
            # |values| <= 1 so the product stays finite
            gb_g = mpd.DataFrame(
                pandas.concat([pdf[[by]],
                               pdf[["v", "w"]].clip(-1.0, 1.0)],
                              axis=1)).groupby(by)
            gb_p = pandas.concat([pdf[[by]],
                                  pdf[["v", "w"]].clip(-1.0, 1.0)],
                                 axis=1).groupby(by)
        else:
            gb_g = df[sub].groupby(by)
            gb_p = pdf[sub].groupby(by)
        if tr == "cumcount":
            g = gb_g.cumcount().to_pandas().to_numpy().astype(float)
            e = gb_p.cumcount().to_numpy().astype(float)
            np.testing.assert_allclose(g, e, rtol=0, equal_nan=True,
                                       err_msg=f"seed {seed} cumcount")
        else:
            if tr in ("tsum", "tmean"):
                gout = gb_g.transform(tr[1:]).to_pandas()
                pout = gb_p.transform(tr[1:])
            elif tr == "shift":
                pr = int(rng.integers(-2, 3)) or 1
                gout = gb_g.shift(pr).to_pandas()
                pout = gb_p.shift(pr)
            else:
                gout = getattr(gb_g, tr)().to_pandas()
                pout = getattr(gb_p, tr)()
            for c in pout.columns:
                np.testing.assert_allclose(
                    gout[c].to_numpy().astype(float),
                    pout[c].to_numpy().astype(float), rtol=1e-12,
                    atol=1e-9, equal_nan=True,
                    err_msg=f"seed {seed} {by}/{tr}/{c}")

    # closing aggregation on whatever survived
    if len(pdf) and all(c in pdf.columns for c in ("s", "v", "w")):
        by = ["k", "s"][rng.integers(0, 2)]
        agg = ["sum", "mean", "count", "min", "max", "var", "median",
               "first", "last", "prod", "sem"][rng.integers(0, 11)]
        sub = [by, "v", "w"]  # numeric values only (string agg is loud)
        gout = getattr(df[sub].groupby(by), agg)().to_pandas()
        pout = getattr(pdf[sub].groupby(by), agg)()
        assert len(gout) == len(pout), f"seed {seed} closing {by}/{agg}"
        for c in pout.columns:
            g, e = gout[c].to_numpy(), pout[c].to_numpy()
            if e.dtype == object:
                continue
            # sem is sqrt-of-variance: the one-pass moment formula's
            # cancellation noise (~1e-15 in var) amplifies to ~1e-7
            at = 1e-4 if agg == "sem" else 1e-9
            np.testing.assert_allclose(
                g.astype(float), e.astype(float), rtol=1e-9, atol=at,
                equal_nan=True, err_msg=f"seed {seed} {by}/{agg}/{c}")


@pytest.mark.parametrize("seed", range(int(os.environ.get(
    "HF_MERGE_FUZZ_N", "15"))))
def test_fuzz_merge(seed):
    """Randomized merge matrix vs pandas: key dtype (small/wide int,
    float+NaN, string, datetime, two-key), how, overlapping payload
    names, empty-overlap cases — sorted-multiset comparison."""
    rng = np.random.default_rng(3000 + seed)
    nl = int(rng.integers(200, 20_000))
    nr = int(rng.integers(100, 8_000))
    kind = rng.choice(["int", "wide", "float", "str", "dt", "multi"])
    if kind in ("str", "dt"):
        # low-cardinality keys: bound the expected join size (the
        # pandas EXPECTATION dominates runtime otherwise)
        nl, nr = min(nl, 3000), min(nr, 1200)
    elif kind == "float":
        nl = min(nl, 8000)
    how = str(rng.choice({
        "int": ["inner", "left", "right", "outer"],
        "wide": ["inner", "left", "right", "outer"],
        "float": ["inner", "left", "right", "outer"],
        "str": ["inner", "left"],
        "dt": ["inner"],
        "multi": ["inner", "left", "right"],
    }[kind]))

    card = int(rng.integers(max(20, (nl * nr) // 2_000_000 + 1), 400))

    def keys(n):
        if kind == "int":
            return rng.integers(0, card, n)
        if kind == "wide":
            return rng.integers(-10**14, 10**14, n)
        if kind == "float":
            pool = np.r_[rng.standard_normal(50), np.nan]
            return rng.choice(pool, n)
        if kind == "str":
            return rng.choice(np.array(
                ["a", "b", "c", "d", None], dtype=object), n)
        if kind == "dt":
            return (pandas.Timestamp("2020-01-01").value
                    + rng.integers(0, 50, n) * 86_400 * 10**9
                    ).astype("datetime64[ns]")
        return None

    if kind == "multi":
        lpdf = pandas.DataFrame({
            "a": rng.integers(0, 40, nl), "b": rng.integers(0, 7, nl),
            "x": rng.standard_normal(nl), "c": rng.integers(0, 5, nl)})
        rpdf = pandas.DataFrame({
            "a": rng.integers(0, 40, nr), "b": rng.integers(0, 7, nr),
            "y": rng.standard_normal(nr), "c": rng.integers(5, 9, nr)})
        on = ["a", "b"]
        sort_cols = ["a", "b", "x", "y", "c_x", "c_y"]
    else:
        lpdf = pandas.DataFrame({"k": keys(nl),
                                 "x": rng.standard_normal(nl),
                                 "c": rng.integers(0, 5, nl)})
        rpdf = pandas.DataFrame({"k": keys(nr),
                                 "y": rng.standard_normal(nr),
                                 "c": rng.integers(5, 9, nr)})
        on = "k"
        sort_cols = ["k", "x", "y", "c_x", "c_y"]
    got = mpd.DataFrame(lpdf).merge(mpd.DataFrame(rpdf), on=on,
                                    how=how).to_pandas()
    exp = lpdf.merge(rpdf, on=on, how=how)
    msg = f"seed {seed} {kind}/{how}"
    assert list(got.columns) == list(exp.columns), msg
    assert len(got) == len(exp), msg
    if not len(exp):
        return
    gs = got.sort_values(sort_cols, na_position="last").reset_index(
        drop=True)
    es = exp.sort_values(sort_cols, na_position="last").reset_index(
        drop=True)
    for c in exp.columns:
        g, e = gs[c].to_numpy(), es[c].to_numpy()
        if e.dtype == object:
            same = (pandas.isna(g) & pandas.isna(e)) | (g == e)
            assert same.all(), f"{msg}/{c}"
        elif np.issubdtype(e.dtype, np.datetime64):
            np.testing.assert_array_equal(g, e, err_msg=f"{msg}/{c}")
        else:
            np.testing.assert_allclose(g.astype(float), e.astype(float),
                                       rtol=0, equal_nan=True,
                                       err_msg=f"{msg}/{c}")
