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


def main():
    lib.ensure_ready(0)
    rng = np.random.default_rng(0)
    for n in (10**4, 10**5, 10**6, 10**7):
        nl = n * 2
        rk = np.arange(n, dtype=np.int64)
        tryit(f"arange n={n}", rk, rng.integers(0, n, nl).astype(np.int64),
              rng.random(n))
        rku = rng.integers(0, n, n).astype(np.int64)
        tryit(f"uniform n={n}", rku, rng.integers(0, n, nl).astype(np.int64),
              rng.random(n))


if __name__ == "__main__":
    main()
