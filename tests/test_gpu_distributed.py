"""GPU tier: the multi-rank path end-to-end on ONE GPU (world 2, gloo
backend with device compute — the single-box stand-in for one-process-per-GPU
over RCCL; the collectives' device form is covered by the same code paths,
with gloo's host bounce replacing the xGMI transport).

Covers both distributed groupby routes of partition_manager.groupby_reduce:
  * dense-range: per-rank table accumulate + table all-reduce + compact,
  * range shuffle: sampled splitters + all-to-all row exchange + per-rank
    local aggregation + result all-gather (the unbounded-key route).
Every rank must return the IDENTICAL replicated result equal to pandas on
the union of the shards.
"""

import multiprocessing as mp
import os

import numpy as np
import pandas
import pytest

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, fail_q):
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import modin_amd.distributed as dist_mod
        assert dist_mod.init_from_env(backend="gloo", gpu=True)
        import modin_amd.config as config
        import modin_amd.pandas as mpd

        rng = np.random.default_rng(41)  # same stream on all ranks
        n = 60_000
        scenarios = {
            # dense-range route (key span < MaxGroupbySlots)
            "dense": rng.integers(-50, 1000, n).astype(np.int64),
            # shuffle route (full int64 span; heavy dup run straddles shards)
            "huge": rng.integers(-2**62, 2**62, n).astype(np.int64),
        }
        scenarios["huge"][rng.random(n) < 0.2] = 77
        vals = rng.random(n)
        vals[rng.random(n) < 0.1] = np.nan
        ivals = rng.integers(-100, 100, n).astype(np.int64)
        lo = rank * n // world
        hi = (rank + 1) * n // world
        for name, keys in scenarios.items():
            pdf = pandas.DataFrame({"k": keys, "v": vals, "w": ivals})
            df = mpd.DataFrame(pdf.iloc[lo:hi])
            for agg in ("sum", "count", "mean"):
                out = getattr(df.groupby("k"), agg)().to_pandas()
                expect = getattr(pdf.groupby("k"), agg)()
                np.testing.assert_array_equal(
                    out.index.to_numpy(), expect.index.to_numpy(),
                    err_msg=f"{name}/{agg} keys")
                for c in ("v", "w"):
                    np.testing.assert_allclose(
                        out[c].to_numpy(), expect[c].to_numpy(),
                        rtol=1e-12, atol=1e-9, equal_nan=True,
                        err_msg=f"{name}/{agg}/{c}")
        # ---- distributed merge: every rank probes its left shard against
        # the all-gathered right table; rank outputs are the pandas-merge
        # slices of the corresponding left shards ----
        rngm = np.random.default_rng(43)
        nl, nr = 30_000, 8_000
        lk = rngm.integers(0, 5_000, nl).astype(np.int64)
        lv = rngm.random(nl)
        rk = rngm.integers(0, 5_000, nr).astype(np.int64)
        rv = rngm.random(nr)
        lpdf = pandas.DataFrame({"k": lk, "a": lv})
        rpdf = pandas.DataFrame({"k": rk, "b": rv})
        llo, lhi = rank * nl // world, (rank + 1) * nl // world
        rlo, rhi = rank * nr // world, (rank + 1) * nr // world
        ldf = mpd.DataFrame(lpdf.iloc[llo:lhi].reset_index(drop=True))
        rdf = mpd.DataFrame(rpdf.iloc[rlo:rhi].reset_index(drop=True))
        got = ldf.merge(rdf, on="k").to_pandas()
        expect = lpdf.iloc[llo:lhi].reset_index(drop=True).merge(rpdf, on="k")
        for c in ("k", "a", "b"):
            np.testing.assert_allclose(got[c].to_numpy(),
                                       expect[c].to_numpy(), rtol=0,
                                       err_msg=f"merge/{c}")

        # ---- distributed sort_values: range shuffle + local radix sort;
        # the rank-order concat of the shards must equal pandas exactly,
        # index (original global positions) included ----
        rngs = np.random.default_rng(44)
        ns = 40_000
        sk = rngs.integers(-2**62, 2**62, ns).astype(np.int64)
        sk[rngs.random(ns) < 0.15] = -5  # dup run straddling shards
        sv = rngs.random(ns)
        spdf = pandas.DataFrame({"k": sk, "v": sv})
        slo, shi = rank * ns // world, (rank + 1) * ns // world
        sdf = mpd.DataFrame(spdf.iloc[slo:shi].reset_index(drop=True))
        for asc in (True, False):
            out = sdf.sort_values("k", ascending=asc).to_pandas()
            # local index is shard-relative positions offset by the global
            # base, i.e. already the GLOBAL original positions
            # the local index already carries GLOBAL original positions
            gathered = dist_mod.allgather_arrays(
                [out.index.to_numpy(),
                 out["k"].to_numpy(), out["v"].to_numpy()])
            exp = spdf.sort_values("k", ascending=asc, kind="stable")
            np.testing.assert_array_equal(gathered[0],
                                          exp.index.to_numpy(),
                                          err_msg=f"sort asc={asc} idx")
            np.testing.assert_array_equal(gathered[1],
                                          exp["k"].to_numpy(),
                                          err_msg=f"sort asc={asc} k")
            np.testing.assert_array_equal(gathered[2],
                                          exp["v"].to_numpy(),
                                          err_msg=f"sort asc={asc} v")
        # ---- distributed median/quantile/first/last (shuffle-then-local,
        # round-2 lift): every rank returns the identical replicated
        # result equal to pandas on the union ----
        rngq = np.random.default_rng(45)
        nq = 30_000
        qk = rngq.integers(0, 500, nq).astype(np.int64)
        qv = rngq.random(nq)
        qv[rngq.random(nq) < 0.1] = np.nan
        qw = rngq.integers(-50, 50, nq).astype(np.int64)
        qpdf = pandas.DataFrame({"k": qk, "v": qv, "w": qw})
        qlo, qhi = rank * nq // world, (rank + 1) * nq // world
        qdf = mpd.DataFrame(qpdf.iloc[qlo:qhi].reset_index(drop=True))
        for op in ("median", "first", "last"):
            out = getattr(qdf.groupby("k"), op)().to_pandas()
            expect = getattr(qpdf.groupby("k"), op)()
            np.testing.assert_array_equal(out.index.to_numpy(),
                                          expect.index.to_numpy(),
                                          err_msg=f"{op} keys")
            for c in ("v", "w"):
                np.testing.assert_allclose(
                    out[c].to_numpy(dtype=np.float64),
                    expect[c].to_numpy(dtype=np.float64), rtol=1e-12,
                    atol=1e-12, equal_nan=True, err_msg=f"{op}/{c}")

        # ---- distributed nunique + idxmax/idxmin (round-2 lift) ----
        out = qdf.groupby("k").nunique().to_pandas()
        expect = qpdf.groupby("k").nunique()
        for c in ("v", "w"):
            np.testing.assert_array_equal(out[c].to_numpy(),
                                          expect[c].to_numpy(),
                                          err_msg=f"nunique/{c}")
        for mx in (True, False):
            out = (qdf.groupby("k").idxmax() if mx
                   else qdf.groupby("k").idxmin()).to_pandas()
            expect = (qpdf.groupby("k").idxmax() if mx
                      else qpdf.groupby("k").idxmin())
            for c in ("v", "w"):
                np.testing.assert_allclose(
                    out[c].to_numpy(dtype=np.float64),
                    expect[c].to_numpy(dtype=np.float64), rtol=0,
                    equal_nan=True, err_msg=f"idx mx={mx}/{c}")

        # ---- distributed MULTI-KEY tail aggs (combined-key route) ----
        qpdf2 = qpdf.assign(k2=(qpdf["k"] % 7))
        qdf2 = mpd.DataFrame(qpdf2.iloc[qlo:qhi].reset_index(drop=True))
        for op in ("median", "nunique"):
            out = getattr(qdf2.groupby(["k", "k2"]), op)().to_pandas()
            expect = getattr(qpdf2.groupby(["k", "k2"]), op)()
            assert list(out.index) == list(expect.index), f"mk-{op} keys"
            for c in ("v", "w"):
                np.testing.assert_allclose(
                    out[c].to_numpy(dtype=np.float64),
                    expect[c].to_numpy(dtype=np.float64), rtol=1e-12,
                    atol=1e-12, equal_nan=True, err_msg=f"mk-{op}/{c}")

        # ---- distributed transforms (shuffle + local + route-back):
        # row-aligned results on each rank's own shard ----
        tdf_p = qpdf.iloc[qlo:qhi].reset_index(drop=True)
        tdf = mpd.DataFrame(tdf_p)
        for how in ("cumsum", "cumcount", "ngroup", "rank", "shift"):
            gb = tdf.groupby("k")
            egb = qpdf.groupby("k")
            if how == "cumsum":
                got = gb.cumsum().to_pandas()
                exp = qpdf.groupby("k").cumsum().iloc[qlo:qhi]
                for c in ("v", "w"):
                    np.testing.assert_allclose(
                        got[c].to_numpy(dtype=np.float64),
                        exp[c].to_numpy(dtype=np.float64), rtol=1e-12,
                        atol=1e-9, equal_nan=True,
                        err_msg=f"transform cumsum/{c}")
            elif how == "cumcount":
                got = gb.cumcount().to_pandas()
                exp = egb.cumcount().iloc[qlo:qhi]
                np.testing.assert_array_equal(
                    np.asarray(got).reshape(-1),
                    exp.to_numpy(), err_msg="transform cumcount")
            elif how == "ngroup":
                got = gb.ngroup().to_pandas()
                exp = egb.ngroup().iloc[qlo:qhi]
                np.testing.assert_array_equal(
                    np.asarray(got).reshape(-1), exp.to_numpy(),
                    err_msg="transform ngroup")
            elif how == "rank":
                got = gb.rank().to_pandas()
                exp = egb.rank().iloc[qlo:qhi]
                for c in ("v", "w"):
                    np.testing.assert_allclose(
                        got[c].to_numpy(), exp[c].to_numpy(), rtol=1e-12,
                        equal_nan=True, err_msg=f"transform rank/{c}")
            else:
                got = gb.shift(1).to_pandas()
                exp = egb.shift(1).iloc[qlo:qhi]
                for c in ("v", "w"):
                    np.testing.assert_allclose(
                        got[c].to_numpy(dtype=np.float64),
                        exp[c].to_numpy(dtype=np.float64), rtol=0,
                        equal_nan=True, err_msg=f"transform shift/{c}")

        # ---- distributed concat: global pandas row order ----
        cpdf_a = qpdf.iloc[qlo:qhi].reset_index(drop=True)
        cpdf_b = spdf.iloc[slo:shi].reset_index(drop=True)[["k", "v"]]
        ca = mpd.DataFrame(cpdf_a[["k", "v"]])
        cb = mpd.DataFrame(cpdf_b)
        cc = mpd.concat([ca, cb])
        blk = cc._query_compiler._modin_frame._partitions[0].block()
        from modin_amd.core import lib as hl
        gathered = dist_mod.allgather_arrays(
            [hl.get(blk.columns["k"]), hl.get(blk.columns["v"])])
        exp_cat = pandas.concat([qpdf[["k", "v"]],
                                 spdf[["k", "v"]]])
        np.testing.assert_array_equal(gathered[0],
                                      exp_cat["k"].to_numpy(),
                                      err_msg="concat k order")
        np.testing.assert_allclose(gathered[1], exp_cat["v"].to_numpy(),
                                   rtol=0, equal_nan=True,
                                   err_msg="concat v order")
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


@pytest.mark.timeout(300)
def test_world2_groupby_dense_and_shuffle_on_gpu(gpu_ready):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29561
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=280)
    errs = []
    while not q.empty():
        errs.append(q.get())
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def _nccl_worker(rank, world, port, fail_q):
    """Exercise the RCCL branch on hardware.  Two ranks on ONE GPU is
    refused by RCCL ("Duplicate GPU detected" — verified on this image),
    so this runs a WORLD-1 nccl group: the RCCL all-reduce on the dense
    CUDA table, all_to_all_single, the zero-copy CAI send of
    exchange_column and the varlen gather all execute through the real
    nccl code paths (gloo's host bounce skips them)."""
    try:
        os.environ.update({
            "RANK": "0", "WORLD_SIZE": "1", "LOCAL_RANK": "0",
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        })
        import torch
        import torch.distributed as dist
        import modin_amd.distributed as dist_mod
        from modin_amd.core import lib
        dist.init_process_group("nccl", rank=0, world_size=1)
        dist_mod._state.update(active=True, rank=0, world=1,
                               backend="nccl", device="cuda:0")
        dist_mod._state["table_device"] = "cuda:0"
        lib.ensure_ready(0)

        n_slots = 1000
        rng = np.random.default_rng(100)
        host_sums = rng.random(n_slots)
        sums = torch.tensor(host_sums, dtype=torch.float64, device="cuda:0")
        rowcnt = torch.ones(n_slots, dtype=torch.int64, device="cuda:0")

        class T:
            _torch_tensors = (sums, rowcnt, None)
            agg_op = 0

        dist_mod.maybe_allreduce_table(T)   # RCCL allreduce (1-rank)
        np.testing.assert_allclose(T._torch_tensors[0].cpu().numpy(),
                                   host_sums, rtol=0)

        # exchange_column: zero-copy CAI send through all_to_all_single
        vals = np.arange(1000, dtype=np.float64)
        col = lib.put(vals)
        view = dist_mod._as_torch_view(col)
        assert view is not None and view.data_ptr() == col.dptr(), \
            "CAI zero-copy view rejected by this torch build"
        out = dist_mod.exchange_column(col, [1000])
        np.testing.assert_array_equal(lib.get(out), vals)
        ic = lib.put(np.arange(500, dtype=np.int64) * 3)
        out2 = dist_mod.exchange_column(ic, [500])
        np.testing.assert_array_equal(lib.get(out2), np.arange(500) * 3)

        arrs = dist_mod.allgather_arrays([np.arange(7, dtype=np.int64)])
        np.testing.assert_array_equal(arrs[0], np.arange(7))
        assert dist_mod.allgather_lengths(5) == [5]
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"{e}\n{traceback.format_exc()}")


@pytest.mark.timeout(240)
def test_nccl_branch_single_gpu(gpu_ready):
    """The RCCL code path on real hardware (1-rank nccl group — RCCL
    refuses 2 ranks on one device, so the multi-rank topology is covered
    by the gloo world-2 tests and the nccl TRANSPORT by this one)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_nccl_worker, args=(0, 1, 29571, q))]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=200)
    errs = []
    while not q.empty():
        errs.append(q.get())
    for p in procs:
        if p.is_alive():
            p.terminate()
            errs.append("nccl worker hung")
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)
