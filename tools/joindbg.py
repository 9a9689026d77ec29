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


def main():
    lib.ensure_ready(0)
    if os.environ.get("JOINDBG_PROF"):
        lib.profiling(True)
        print("profiling ON")
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
