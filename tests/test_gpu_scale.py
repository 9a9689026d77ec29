"""GPU tier: size-independent PROPERTY checks at near-BASELINE sizes.

Small-size parity lives in test_gpu_parity.py; these runs push the NEWER
compositions (cut, sample, melt, corr, duplicated keep variants) to
2e8-1e9 rows with device-generated frames (no host data) and verify
properties the oracle RNG mirrors make exact: bin-count conservation,
draw-set membership against the oracle stream, column-sum conservation
under reshape, and closed-form correlations.
"""

import numpy as np
import pandas
import pytest

import modin_amd.pandas as mpd
from modin_amd.core import lib
from modin_amd.core.dataframe import HipDataframe
from modin_amd.core.partition import DeviceBlock, HipDataframePartition
from modin_amd.query_compiler import HipQueryCompiler

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _ready(gpu_ready):
    yield


def device_frame(n, seed, keys=1000):
    k = lib.fill_randint(n, seed, 0, keys)
    v = lib.fill_randf64(n, seed + 7)
    block = DeviceBlock({"k": k, "v": v}, n)
    frame = HipDataframe([HipDataframePartition(block)],
                         pandas.RangeIndex(n), ["k", "v"], [n],
                         pandas.Series({"k": np.dtype(np.int64),
                                        "v": np.dtype(np.float64)}))
    return mpd.DataFrame(query_compiler=HipQueryCompiler(frame))


def test_cut_conservation_1e9():
    """cut over 1e9 uniform draws: every row lands in exactly one bin
    (counts sum to n), and the per-bin counts equal the exact uniform
    expectation bands (binomial 6-sigma)."""
    n = 1_000_000_000
    df = device_frame(n, 101)
    df["bin"] = mpd.cut(df["v"], [-1e-9, 0.25, 0.5, 0.75, 1.0])
    vc = df["bin"].value_counts()
    counts = np.asarray(vc)
    assert counts.sum() == n  # conservation: U[0,1) never leaves [0,1)
    p = 0.25
    sigma = np.sqrt(n * p * (1 - p))
    assert (np.abs(counts - n * p) < 6 * sigma).all(), counts
    # labels=False codes agree with a direct threshold count
    codes = mpd.cut(df["v"], [-1e-9, 0.5, 1.0], labels=False)
    s = codes._query_compiler.sum().iloc[0]
    upper = int((df["v"] > 0.5)._query_compiler.sum().iloc[0])
    assert int(s) == upper  # code 1 iff v > 0.5 (right-closed bins)


def test_sample_oracle_membership_1e9():
    """sample(1e6 of 1e9): the selected row positions are EXACTLY the
    argsort head of the oracle's mirrored uniform stream."""
    import oracle
    n = 400_000_000
    take = 1_000_000
    k = lib.fill_randint(n, 77, 0, 1 << 40)
    block = DeviceBlock({"k": k}, n)
    frame = HipDataframe([HipDataframePartition(block)],
                         pandas.RangeIndex(n), ["k"], [n],
                         pandas.Series({"k": np.dtype(np.int64)}))
    df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))
    got = df.sample(n=take, random_state=5)
    pos = np.sort(np.asarray(got.index))
    # oracle mirror, chunked to bound host RAM
    best = []
    for a in range(0, n, 100_000_000):
        m = min(100_000_000, n - a)
        u = oracle.rand_f64(5, m, a)
        idx = np.argpartition(u, take)[:take]
        best.append(np.stack([u[idx], (idx + a)]))
    allb = np.concatenate(best, axis=1)
    order = np.argsort(allb[0], kind="stable")[:take]
    exp = np.sort(allb[1][order].astype(np.int64))
    np.testing.assert_array_equal(pos, exp)


def test_melt_conservation_2e8():
    """melt of a 2e8x2 frame: 4e8 rows out, per-variable sums equal the
    source column sums exactly (f64 reduce is deterministic on the same
    data in the same order)."""
    n = 200_000_000
    df = device_frame(n, 55)
    df["w"] = df["v"] * 2.0
    m = df.melt(id_vars="k", value_vars=["v", "w"])
    assert len(m) == 2 * n
    sv = float(df["v"]._query_compiler.sum().iloc[0])
    sw = float(df["w"]._query_compiler.sum().iloc[0])
    both = m[m["variable"].str.contains("v", na=False)]
    sm_v = float(both["value"]._query_compiler.sum().iloc[0])
    assert abs(sm_v - sv) < 1e-6 * max(abs(sv), 1.0)
    sm_all = float(m["value"]._query_compiler.sum().iloc[0])
    assert abs(sm_all - (sv + sw)) < 1e-6 * max(abs(sv + sw), 1.0)


def test_corr_closed_form_2e8():
    """corr at 2e8 rows: corr(v, a*v+b) = 1 and corr against an
    independent seed ~ 0 (|r| < 6/sqrt(n))."""
    n = 200_000_000
    v = lib.fill_randf64(n, 900)
    u = lib.fill_randf64(n, 901)
    w = lib.binary(lib.BIN_ADD,
                   lib.map_scalar(lib.MAP_MUL, v, 3.0),
                   lib.map_scalar(lib.MAP_MUL, u, 0.0))
    block = DeviceBlock({"v": v, "u": u, "w": w}, n)
    frame = HipDataframe([HipDataframePartition(block)],
                         pandas.RangeIndex(n), ["v", "u", "w"], [n],
                         pandas.Series({c: np.dtype(np.float64)
                                        for c in ("v", "u", "w")}))
    df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))
    c = df.corr()
    assert abs(c.loc["v", "w"] - 1.0) < 1e-9
    assert abs(c.loc["v", "u"]) < 6.0 / np.sqrt(n)


def test_duplicated_last_consistency_2e8():
    """duplicated keep variants at 2e8 rows, 1e6 keys: per-key exactly
    one kept row for 'first' and 'last'; keep=False drops every key
    that appears more than once (device counts cross-check)."""
    n = 200_000_000
    keys = 1_000_000
    df = device_frame(n, 333, keys=keys)
    nuniq = len(df[["k"]].drop_duplicates("k"))
    nlast = len(df[["k"]].drop_duplicates("k", keep="last"))
    assert nuniq == nlast == keys  # every key hit at this density
    f = df["k"]._query_compiler.duplicated("k")
    l_ = df["k"]._query_compiler.duplicated("k", keep="last")
    sf = int(f.sum().iloc[0])
    sl = int(l_.sum().iloc[0])
    assert sf == sl == n - keys
