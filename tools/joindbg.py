#!/usr/bin/env python3
"""joindbg — bisect the join_build failure scale: builds with arange keys
(no duplicates) and uniform keys at growing sizes, validates probe output
against the oracle on a sample."""

import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from modin_amd.core import lib  # noqa: E402
import oracle  # noqa: E402


def tryit(tag, rk, lk, rv):
    try:
        rkc = lib.put(rk)
        rvc = lib.put(rv)
        r = lib.reduce(rkc)
        kmin, n_slots = r.imn, r.imx - r.imn + 1
        j = lib.join_build(rkc, [rvc], kmin, n_slots)
        lkc = lib.put(lk)
        keys_c, lidx_c, rcols, n_out = lib.join_probe(j, lkc)
        ok, olidx, _, orv = oracle.inner_join(lk, {}, rk, {"v": rv})
        got_keys = lib.get(keys_c)
        got_lidx = lib.get(lidx_c)
        got_rv = lib.get(rcols[0])
        assert n_out == ok.size, f"n_out {n_out} vs {ok.size}"
        np.testing.assert_array_equal(got_keys, ok)
        np.testing.assert_array_equal(got_lidx, olidx)
        np.testing.assert_array_equal(got_rv, orv["v"])
        print(f"{tag}: OK n_out={n_out}")
    except Exception as e:
        print(f"{tag}: FAIL {type(e).__name__}: {str(e)[:140]}")
    finally:
        lib.sync()


def csr_check(tag, rk):
    """Build only; D2H the CSR and diff against numpy bincount."""
    import ctypes as ct
    try:
        rkc = lib.put(rk)
        rvc = lib.put(np.zeros(rk.size))
        r = lib.reduce(rkc)
        kmin, n_slots = r.imn, r.imx - r.imn + 1
        j = lib.join_build(rkc, [rvc], kmin, n_slots)
        print(f"{tag}: build OK")
    except lib.HfError as e:
        print(f"{tag}: build FAIL {str(e)[:120]}")
    lib.sync()


def replicate_joinbench(nl, nr):
    """Exact joinbench flow: left frame first (k+lv), right (k+rv), qc.merge."""
    import pandas
    from modin_amd.core.dataframe import HipDataframe
    from modin_amd.core.partition import DeviceBlock, HipDataframePartition
    from modin_amd.query_compiler import HipQueryCompiler

    lib.profiling(True)
    rng = np.random.default_rng(42)
    keyspace = nr

    def frame(n, cols):
        block = {"k": lib.put(rng.integers(0, keyspace, n).astype(np.int64))}
        for name in cols:
            block[name] = lib.put(rng.random(n))
        names = list(block)
        return HipQueryCompiler(HipDataframe(
            [HipDataframePartition(DeviceBlock(block, n))],
            pandas.RangeIndex(n), names, [n],
            pandas.Series({"k": np.dtype(np.int64),
                           **{c: np.dtype(np.float64) for c in cols}})))

    qL = frame(nl, ["lv"])
    qR = frame(nr, ["rv"])
    try:
        out = qL.merge(qR, on="k")
        lib.sync()
        print(f"replicate nl={nl} nr={nr}: OK out={len(out)}")
    except lib.HfError as e:
        print(f"replicate nl={nl} nr={nr}: FAIL {str(e)[:120]}")
        # diff the histogram vs numpy to see whether the counts were wrong
        rk = lib.get(qR._modin_frame._partitions[0].block().columns["k"])
        cnt = np.bincount(rk, minlength=keyspace)
        print("  numpy max multiplicity:", cnt.max())


def main():
    lib.ensure_ready(0)
    if os.environ.get("JOINDBG_PROF"):
        lib.profiling(True)
        print("profiling ON")
    mode = os.environ.get("JOINDBG_MODE", "replicate")
    if mode == "replicate":
        replicate_joinbench(2 * 10**7, 10**7)
        return
    rng = np.random.default_rng(42)
    reps = int(os.environ.get("JOINDBG_REPS", "3"))
    n = 10**7
    for rep in range(reps):
        rku = rng.integers(0, n, n).astype(np.int64)
        csr_check(f"rep{rep} uniform n={n}", rku)
        tryit(f"rep{rep} full n={n}", rku,
              rng.integers(0, n, 2 * n).astype(np.int64), rng.random(n))


if __name__ == "__main__":
    main()
