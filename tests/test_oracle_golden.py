"""CPU tier: pin the oracle against the reference-generated golden vectors.

The vectors were produced by oracle/make_golden.py running the REAL reference
(Modin PandasOnPython, NPartitions=3) in the build container and are the
parity anchor on the GPU box where /root/reference does not exist.
"""

import numpy as np
import pytest

import oracle
from tests.conftest import golden_cases, load_golden

GB_CASES = golden_cases("gb_")
RED_CASES = golden_cases("red_")


def _in_cols(g):
    return {k[3:]: v for k, v in g.items() if k.startswith("in_") and k != "in_k"}


@pytest.mark.parametrize("case", GB_CASES)
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_groupby_vs_golden(case, agg):
    g = load_golden(case)
    keys, out = oracle.groupby_agg(g["in_k"], _in_cols(g), agg)
    np.testing.assert_array_equal(keys, g[f"out_{agg}_keys"])
    for name in _in_cols(g):
        expect = g[f"out_{agg}_{name}"]
        if agg == "count":
            np.testing.assert_array_equal(out[name], expect.astype(np.int64))
        else:
            np.testing.assert_allclose(out[name], expect, rtol=1e-12, equal_nan=True)


@pytest.mark.parametrize("case", GB_CASES)
def test_partitioned_groupby_matches_global(case):
    """The reference's two-phase (map per partition / reduce) path must agree
    with the one-shot oracle up to fp reassociation."""
    g = load_golden(case)
    cols = _in_cols(g)
    k1, o1 = oracle.groupby_agg(g["in_k"], cols, "sum")
    k2, o2 = oracle.partitioned_groupby_agg(g["in_k"], cols, "sum", num_splits=4)
    np.testing.assert_array_equal(k1, k2)
    for name in cols:
        np.testing.assert_allclose(o1[name], o2[name], rtol=1e-12)


@pytest.mark.parametrize("case", RED_CASES)
@pytest.mark.parametrize("agg", ["sum", "count", "mean", "min", "max"])
def test_reduce_vs_golden(case, agg):
    g = load_golden(case)
    names = [k[3:] for k in g if k.startswith("in_")]
    got = np.array([float(oracle.reduce_op(agg, g[f"in_{n}"])) for n in names])
    np.testing.assert_allclose(got, g[f"out_{agg}"], rtol=1e-12, equal_nan=True)


def test_map_binary_vs_golden():
    g = load_golden("map_binary")
    v, w, i = g["in_v"], g["in_w"], g["in_i"]
    checks = {
        "add1": lambda x: oracle.map_op("add", x, 1),
        "mul2": lambda x: oracle.map_op("mul", x, 2.5),
        "sub3": lambda x: oracle.map_op("sub", x, 3.25),
        "div2": lambda x: oracle.map_op("div", x, 2.0),
        "rsub": lambda x: oracle.map_op("rsub", x, 1.0),
        "fill0": lambda x: oracle.map_op("fillna", x, 0.0),
        "fillm1": lambda x: oracle.map_op("fillna", x, -1.5),
        "abs": lambda x: oracle.map_op("abs", x),
        "frame_add": lambda x: oracle.binary_op("add", x, x),
        "frame_mul": lambda x: oracle.binary_op("mul", x, x),
        "frame_div": lambda x: oracle.binary_op("div", x, oracle.map_op("add", x, 10.0)),
    }
    for tag, fn in checks.items():
        np.testing.assert_array_equal(fn(v), g[f"out_{tag}_v"], err_msg=tag)
        np.testing.assert_array_equal(fn(w), g[f"out_{tag}_w"], err_msg=tag)
    np.testing.assert_array_equal(oracle.map_op("add", i, 7), g["out_iadd_i"])
    np.testing.assert_array_equal(oracle.map_op("mul", i, -3), g["out_imul_i"])
    np.testing.assert_array_equal(oracle.map_op("abs", i), g["out_iabs_i"])


def test_split_row_counts_matches_reference_rule():
    """compute_chunksize (storage_formats/pandas/utils.py:28): ceil division,
    floored at MinRowPartitionSize."""
    # chunk = max(ceil(n/num_splits), min_size)
    assert oracle.split_row_counts(100, 4, 32) == [32, 32, 32, 4]
    assert oracle.split_row_counts(100, 3, 32) == [34, 34, 32]
    assert oracle.split_row_counts(100, 3, 1) == [34, 34, 32]
    assert oracle.split_row_counts(10, 4, 32) == [10]
    assert oracle.split_row_counts(0, 4, 32) == [0]
    assert sum(oracle.split_row_counts(10**6, 8, 32)) == 10**6
    assert len(oracle.split_row_counts(10**6, 8, 32)) == 8


def _merge_inputs(g):
    lcols = {k[len("in_l_"):]: v for k, v in g.items() if k.startswith("in_l_")}
    rcols = {k[len("in_r_"):]: v for k, v in g.items() if k.startswith("in_r_")}
    return g["in_lk"], lcols, g["in_rk"], rcols


def oracle_merge_columns(lk, lcols, rk, rcols):
    """Assemble the full pandas-shaped merge output from the oracle join
    (suffix rules of pandas merge: colliding names get _x/_y)."""
    keys, lidx, out_l, out_r = oracle.inner_join(lk, lcols, rk, rcols)
    common = set(lcols) & set(rcols)
    out = {"k": keys}
    for n in lcols:
        out[n + "_x" if n in common else n] = out_l[n]
    for n in rcols:
        out[n + "_y" if n in common else n] = out_r[n]
    return out


@pytest.mark.parametrize("case", golden_cases("mg_"))
def test_merge_vs_golden(case):
    g = load_golden(case)
    lk, lcols, rk, rcols = _merge_inputs(g)
    out = oracle_merge_columns(lk, lcols, rk, rcols)
    expect_cols = [str(c) for c in g["out_columns"]]
    assert set(out) == set(expect_cols)
    for c in expect_cols:
        np.testing.assert_array_equal(out[c], g[f"out_{c}"], err_msg=c)


def test_filter_vs_golden():
    g = load_golden("flt_basic")
    cols = {"v": g["in_v"], "w": g["in_w"], "i": g["in_i"]}
    masks = {
        "gt": oracle.compare_op("gt", g["in_v"], 0.25),
        "le": oracle.compare_op("le", g["in_v"], 0.5),
        "eq": oracle.compare_op("eq", g["in_i"], 3),
        "ne": oracle.compare_op("ne", g["in_v"], 0.0),
        "none": oracle.compare_op("gt", g["in_v"], 2.0),
    }
    for tag, mask in masks.items():
        np.testing.assert_array_equal(mask, g[f"out_mask_{tag}"], err_msg=tag)
        pos, out = oracle.filter_rows(mask, cols)
        np.testing.assert_array_equal(pos, g[f"out_idx_{tag}"], err_msg=tag)
        for c in cols:
            np.testing.assert_array_equal(out[c], g[f"out_{tag}_{c}"],
                                          err_msg=f"{tag}/{c}")


def test_sort_vs_golden():
    g = load_golden("srt_basic")
    for tag, asc in [("asc", True), ("desc", False)]:
        perm = oracle.sort_perm(g["in_k"], ascending=asc)
        np.testing.assert_array_equal(perm, g[f"out_idx_{tag}"], err_msg=tag)
        for c in ("k", "v", "i"):
            np.testing.assert_array_equal(g[f"in_{c}"][perm],
                                          g[f"out_{tag}_{c}"],
                                          err_msg=f"{tag}/{c}")
    perm = oracle.sort_perm(g["in_kn"])
    np.testing.assert_array_equal(perm, g["out_neg_idx"])
    np.testing.assert_array_equal(g["in_kn"][perm], g["out_neg_k"])


def test_groupby_empty():
    keys, out = oracle.groupby_agg(np.empty(0, np.int64), {"v": np.empty(0)}, "sum")
    assert keys.size == 0 and out["v"].size == 0


@pytest.mark.parametrize("seed", range(6))
def test_oracle_vs_pandas_property(seed):
    """Property pin beyond the committed goldens: the oracle restatement
    matches pandas 2.3.3 itself (the arbiter the reference's df_equals
    tests use — SURVEY §8c) on randomized inputs with NaNs."""
    import pandas
    rng = np.random.default_rng(500 + seed)
    n = int(rng.integers(1_000, 50_000))
    keys = rng.integers(-50, 50, n).astype(np.int64)
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.1] = np.nan
    pdf = pandas.DataFrame({"k": keys, "v": vals})
    for agg in ("sum", "count", "mean", "min", "max"):
        ok, ov = oracle.groupby_agg(keys, {"v": vals}, agg)
        exp = getattr(pdf.groupby("k")["v"], agg)()
        np.testing.assert_array_equal(ok, exp.index.to_numpy(), err_msg=agg)
        np.testing.assert_allclose(
            np.asarray(ov["v"], dtype=np.float64),
            exp.to_numpy().astype(np.float64), rtol=1e-12, atol=1e-12,
            equal_nan=True, err_msg=agg)
    # stable sort permutation == pandas kind='stable' row order
    perm = oracle.sort_perm(keys, ascending=True)
    exp_idx = pdf.sort_values("k", kind="stable").index.to_numpy()
    np.testing.assert_array_equal(perm, exp_idx)
    # reductions
    for op in ("sum", "count", "min", "max", "mean"):
        got = oracle.reduce_op(op, vals)
        exp = getattr(pandas.Series(vals), op)()
        np.testing.assert_allclose(float(got), float(exp), rtol=1e-12,
                                   err_msg=op)


@pytest.mark.parametrize("seed", range(4))
def test_oracle_join_filter_vs_pandas_property(seed):
    """Oracle inner join + filter vs pandas on randomized inputs."""
    import pandas
    rng = np.random.default_rng(700 + seed)
    nl, nr = int(rng.integers(500, 8000)), int(rng.integers(200, 3000))
    lk = rng.integers(0, 400, nl).astype(np.int64)
    rk = rng.integers(0, 400, nr).astype(np.int64)
    lv = rng.standard_normal(nl)
    rv = rng.standard_normal(nr)
    ok, _lidx, olv, orv = oracle.inner_join(lk, {"a": lv}, rk, {"b": rv})
    pl = pandas.DataFrame({"k": lk, "a": lv})
    pr = pandas.DataFrame({"k": rk, "b": rv})
    exp = pl.merge(pr, on="k")
    assert len(ok) == len(exp)
    np.testing.assert_allclose(np.sort(np.asarray(olv["a"])),
                               np.sort(exp["a"].to_numpy()), rtol=1e-15)
    np.testing.assert_allclose(np.sort(np.asarray(orv["b"])),
                               np.sort(exp["b"].to_numpy()), rtol=1e-15)
    # filter mask semantics (NaN compares False except NE)
    v = rng.standard_normal(2000)
    v[rng.random(2000) < 0.2] = np.nan
    thr = float(rng.standard_normal())
    mask = oracle.compare_op("gt", v, thr)
    np.testing.assert_array_equal(
        mask.astype(bool), pandas.Series(v).gt(thr).to_numpy())
    _pos, fcols = oracle.filter_rows(mask, {"v": v})
    kept = fcols["v"]
    np.testing.assert_allclose(
        np.asarray(kept), v[pandas.Series(v).gt(thr).to_numpy()],
        rtol=0, equal_nan=True)


@pytest.mark.parametrize("seed", range(10))
def test_oracle_fuzz_vs_pandas(seed):
    """Randomized oracle fuzz: groupby (all aggs, random key spans and
    NaN densities, incl. single-group and all-NaN-column cases),
    partitioned==global, sort asc/desc ties, joins with empty sides.
    The bench's pre-timing parity gate trusts these functions."""
    import pandas
    rng = np.random.default_rng(900 + seed)
    n = int(rng.integers(100, 30_000))
    span = int(rng.integers(1, 1000))
    keys = rng.integers(-span, span + 1, n).astype(np.int64)
    vals = rng.standard_normal(n)
    nan_frac = float(rng.random()) * 0.9
    vals[rng.random(n) < nan_frac] = np.nan
    pdf = pandas.DataFrame({"k": keys, "v": vals})
    for agg in ("sum", "count", "mean", "min", "max"):
        gk, gv = oracle.groupby_agg(keys, {"v": vals}, agg)
        exp = getattr(pdf.groupby("k")["v"], agg)()
        np.testing.assert_array_equal(gk, exp.index.to_numpy(),
                                      err_msg=f"{seed}/{agg}")
        np.testing.assert_allclose(
            np.asarray(gv["v"], dtype=np.float64),
            exp.to_numpy().astype(np.float64), rtol=1e-12, atol=1e-12,
            equal_nan=True, err_msg=f"{seed}/{agg}")
        # partitioned map-reduce == global (the multi-partition bench path)
        splits = int(rng.integers(1, 9))
        pk, pv = oracle.partitioned_groupby_agg(keys, {"v": vals}, agg,
                                                splits)
        np.testing.assert_array_equal(pk, gk)
        np.testing.assert_allclose(np.asarray(pv["v"], dtype=np.float64),
                                   np.asarray(gv["v"], dtype=np.float64),
                                   rtol=1e-12, atol=1e-12, equal_nan=True,
                                   err_msg=f"{seed}/{agg}/part{splits}")
    for asc in (True, False):
        perm = oracle.sort_perm(keys, ascending=asc)
        exp_idx = pdf.sort_values("k", ascending=asc,
                                  kind="stable").index.to_numpy()
        np.testing.assert_array_equal(perm, exp_idx, err_msg=str(asc))
    # join with a possibly-empty right
    nr = int(rng.integers(0, 500))
    rk = rng.integers(-span, span + 1, nr).astype(np.int64)
    rv = rng.standard_normal(nr)
    jk, _jl, _ja, _jb = oracle.inner_join(keys, {"a": vals}, rk, {"b": rv})
    exp_len = len(pdf.merge(pandas.DataFrame({"k": rk, "b": rv}), on="k"))
    assert len(jk) == exp_len
