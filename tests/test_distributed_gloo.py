"""CPU tier: the multi-GPU exchange logic over gloo, world_size=2.

Covers the two collectives of the distributed path
without HIP: (a) the dense groupby-table all-reduce — per-rank shard tables
built by the oracle must merge to the global oracle table; (b) the
TreeReduce partial combine; (c) the key-range min/max.
"""

import multiprocessing as mp
import os

import numpy as np
import pytest

import oracle


def _worker(rank, world, port, fail_q):
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import torch
        import modin_amd.distributed as dist_mod
        assert dist_mod.init_from_env(backend="gloo", gpu=False)

        rng = np.random.default_rng(123)  # same on both ranks
        n = 20_000
        keys = rng.integers(-10, 90, n).astype(np.int64)
        vals = rng.random(n)
        vals[rng.random(n) < 0.1] = np.nan
        counts = oracle.split_row_counts(n, world, 32)
        offs = np.cumsum([0] + counts)
        sl = slice(offs[rank], offs[rank + 1])
        k_loc, v_loc = keys[sl], vals[sl]

        # (c) key-range all-reduce
        lo, hi = dist_mod.allreduce_minmax(
            int(k_loc.min()) if k_loc.size else None,
            int(k_loc.max()) if k_loc.size else None,
        )
        assert lo == int(keys.min()) and hi == int(keys.max())

        # (a) dense-table all-reduce == global oracle
        n_slots = hi - lo + 1
        shifted = k_loc - lo
        valid = ~np.isnan(v_loc)
        sums = np.bincount(shifted[valid], weights=v_loc[valid], minlength=n_slots)
        rowcnt = np.bincount(shifted, minlength=n_slots)
        cnts = np.bincount(shifted[valid], minlength=n_slots)

        class FakeTable:
            _torch_tensors = (
                torch.tensor(sums, dtype=torch.float64),
                torch.tensor(rowcnt, dtype=torch.int64),
                torch.tensor(cnts, dtype=torch.int64),
            )

        dist_mod.maybe_allreduce_table(FakeTable)
        g_sums, g_rowcnt, g_cnts = (t.numpy() for t in FakeTable._torch_tensors)
        ok, osum = oracle.groupby_agg(keys, {"v": vals}, "sum")
        _, ocnt = oracle.groupby_agg(keys, {"v": vals}, "count")
        present = g_rowcnt > 0
        np.testing.assert_array_equal(np.nonzero(present)[0] + lo, ok)
        np.testing.assert_allclose(g_sums[present], osum["v"], rtol=1e-12)
        np.testing.assert_array_equal(g_cnts[present], ocnt["v"])

        # (b) TreeReduce partial combine
        loc = {
            "v": {
                "sum": oracle.reduce_op("sum", v_loc),
                "count": oracle.reduce_op("count", v_loc),
                "mn": oracle.reduce_op("min", v_loc),
                "mx": oracle.reduce_op("max", v_loc),
                "isum": 0, "imn": 0, "imx": 0,
            },
            "k": {
                "sum": float(k_loc.sum()),
                "count": int(k_loc.size),
                "mn": float(k_loc.min()), "mx": float(k_loc.max()),
                "isum": int(k_loc.sum()), "imn": int(k_loc.min()),
                "imx": int(k_loc.max()),
            },
        }
        out = dist_mod.allreduce_partials(loc, ["v", "k"])
        np.testing.assert_allclose(out["v"]["sum"],
                                   oracle.reduce_op("sum", vals), rtol=1e-12)
        assert out["v"]["count"] == oracle.reduce_op("count", vals)
        np.testing.assert_allclose(out["v"]["mn"],
                                   oracle.reduce_op("min", vals), rtol=0)
        np.testing.assert_allclose(out["v"]["mx"],
                                   oracle.reduce_op("max", vals), rtol=0)
        assert out["k"]["isum"] == int(keys.sum())
        assert out["k"]["imn"] == int(keys.min())
        assert out["k"]["imx"] == int(keys.max())
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


def _shuffle_worker(rank, world, port, fail_q):
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import torch
        import modin_amd.distributed as dist_mod
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        rng = np.random.default_rng(7)  # same stream on all ranks
        n = 5000
        keys_all = rng.integers(0, 10**6, (world, n)).astype(np.int64)
        mine = keys_all[rank]
        # hash-partition by destination rank, spans ordered by dest — the
        # host side of the shuffle split (shuffle_partitions device form)
        dest = mine % world
        order = np.argsort(dest, kind="stable")
        counts = np.bincount(dest, minlength=world)
        t = torch.tensor(mine[order])
        out, recv_counts = dist_mod.exchange_splits(t, counts.tolist())
        got = np.sort(out.numpy())
        # expectation: every key (from any rank) whose hash lands on me
        expect = np.sort(np.concatenate(
            [keys_all[src][keys_all[src] % world == rank]
             for src in range(world)]))
        np.testing.assert_array_equal(got, expect)
        assert sum(recv_counts) == expect.size
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


def _range_shuffle_groupby_worker(rank, world, port, fail_q):
    """The full range-partitioning shuffle-groupby recipe
    (partition_manager._groupby_shuffle) executed on the oracle backend:
    real collectives (sample_splitters / exchange_splits /
    allgather_groupby), numpy in place of the device kernels."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import torch
        import modin_amd.distributed as dist_mod
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        rng = np.random.default_rng(31)  # same stream on all ranks
        n = 30_000
        keys_g = rng.integers(-2**62, 2**62, n).astype(np.int64)
        keys_g[rng.random(n) < 0.2] = 77  # heavy duplicate straddling test
        vals_g = rng.random(n)
        vals_g[rng.random(n) < 0.1] = np.nan
        counts = oracle.split_row_counts(n, world, 32)
        offs = np.cumsum([0] + counts)
        sl = slice(offs[rank], offs[rank + 1])
        k_loc, v_loc = keys_g[sl], vals_g[sl]

        # strided sample -> identical splitters on every rank
        S = min(k_loc.size, 4096)
        sample = (k_loc[np.linspace(0, k_loc.size - 1, S).astype(np.int64)]
                  if S else np.empty(0, dtype=np.int64))
        splitters = dist_mod.sample_splitters(sample)
        sp2 = dist_mod.sample_splitters(sample)
        np.testing.assert_array_equal(splitters, sp2)  # deterministic

        dest = oracle.shuffle_dest(k_loc, splitters)
        assert dest.min() >= 0 and dest.max() < world if dest.size else True
        order = np.argsort(dest, kind="stable")
        send_counts = np.bincount(dest, minlength=world).tolist()
        rk, _ = dist_mod.exchange_splits(
            torch.from_numpy(k_loc[order].copy()), send_counts)
        rv, _ = dist_mod.exchange_splits(
            torch.from_numpy(v_loc[order].copy()), send_counts)
        rk, rv = rk.numpy(), rv.numpy()

        def sort_gb(k, v):  # huge-span groupby (bincount oracle would blow up)
            o = np.argsort(k, kind="stable")
            ks, vs = k[o], v[o]
            uk, starts = np.unique(ks, return_index=True)
            sums = np.add.reduceat(np.nan_to_num(vs), starts) \
                if ks.size else np.empty(0)
            cnts = np.add.reduceat((~np.isnan(vs)).astype(np.int64), starts) \
                if ks.size else np.empty(0, dtype=np.int64)
            return uk, sums, cnts

        # a key's rows all land on exactly one rank
        ok_loc, osum_loc, ocnt_loc = sort_gb(rk, rv)
        gk, gs, gc = dist_mod.allgather_groupby(
            ok_loc, [osum_loc], [ocnt_loc])
        ok, osum, ocnt = sort_gb(keys_g, vals_g)
        np.testing.assert_array_equal(gk, ok)  # ascending, disjoint, complete
        np.testing.assert_allclose(gs[0], osum, rtol=1e-12, atol=1e-9)
        np.testing.assert_array_equal(gc[0], ocnt)
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


def _dist_sort_worker(rank, world, port, fail_q):
    """The distributed sort recipe (dataframe._sort_rows_distributed) on the
    numpy backend with real collectives: splitters -> dest -> stable
    per-dest split + positions -> exchange -> local stable sort; the
    rank-order concat must equal the global stable sort, positions
    included."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import torch
        import modin_amd.distributed as dist_mod
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        rng = np.random.default_rng(57)
        n = 20_000
        keys_g = rng.integers(-1000, 1000, n).astype(np.int64)
        vals_g = rng.random(n)
        counts = oracle.split_row_counts(n, world, 32)
        offs = np.cumsum([0] + counts)
        k_loc = keys_g[offs[rank]:offs[rank + 1]]
        v_loc = vals_g[offs[rank]:offs[rank + 1]]
        base = dist_mod.global_row_base(k_loc.size)
        assert base == offs[rank]
        for asc in (True, False):
            S = min(k_loc.size, 4096)
            sample = k_loc[np.linspace(0, k_loc.size - 1, S).astype(np.int64)]
            spl = dist_mod.sample_splitters(sample)
            dest = oracle.shuffle_dest(k_loc, spl)
            if not asc:
                dest = (world - 1) - dest
            order = np.argsort(dest, kind="stable")
            sc = np.bincount(dest, minlength=world).tolist()
            pos = base + np.arange(k_loc.size, dtype=np.int64)
            rk, _ = dist_mod.exchange_splits(
                torch.from_numpy(k_loc[order].copy()), sc)
            rp, _ = dist_mod.exchange_splits(
                torch.from_numpy(pos[order].copy()), sc)
            rk, rp = rk.numpy(), rp.numpy()
            o2 = np.argsort(-rk if not asc else rk, kind="stable")
            gk, gp = dist_mod.allgather_arrays([rk[o2], rp[o2]])
            eo = np.argsort(-keys_g if not asc else keys_g, kind="stable")
            np.testing.assert_array_equal(gk, keys_g[eo])
            np.testing.assert_array_equal(gp, eo)
            np.testing.assert_array_equal(vals_g[gp], vals_g[eo])
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


@pytest.mark.timeout(120)
@pytest.mark.parametrize("world", [2, 3])
def test_gloo_distributed_sort_recipe(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29550 + world
    procs = [ctx.Process(target=_dist_sort_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


@pytest.mark.timeout(120)
@pytest.mark.parametrize("world", [2, 3])
def test_gloo_range_shuffle_groupby(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29540 + world
    procs = [ctx.Process(target=_range_shuffle_groupby_worker,
                         args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


@pytest.mark.timeout(120)
@pytest.mark.parametrize("world", [2, 3])
def test_gloo_shuffle_exchange(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29520 + world
    procs = [ctx.Process(target=_shuffle_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


@pytest.mark.timeout(120)
def test_gloo_world2_exchange_logic():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def _concat_worker(rank, world, port, fail_q):
    """World>1 concat must produce the GLOBAL pandas concat order (frame 0
    across ranks, then frame 1, ...) — the exchange-based concat, exercised
    through the REAL dataframe.py composition code on the numpy lib mock."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import pandas
        import modin_amd.distributed as dist_mod
        from tests import mocklib

        class _RawPatch:
            def setattr(self, obj, name, fn):
                setattr(obj, name, fn)

        mocklib.install(_RawPatch())
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        from modin_amd.core import lib
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import (DeviceBlock,
                                              HipDataframePartition)

        rng = np.random.default_rng(31)  # same stream on all ranks
        nA, nB = 1000, 700
        A_k = rng.integers(0, 50, nA).astype(np.int64)
        A_v = rng.random(nA)
        B_k = rng.integers(0, 50, nB).astype(np.int64)
        B_v = rng.random(nB)

        def shard(arr):
            counts = oracle.split_row_counts(len(arr), world, 1)
            while len(counts) < world:
                counts.append(0)
            offs = np.cumsum([0] + counts)
            return arr[offs[rank]:offs[rank + 1]], offs[rank]

        def frame(k, v, idx0):
            n = len(k)
            block = DeviceBlock({"k": lib.put(k), "v": lib.put(v)}, n)
            return HipDataframe(
                [HipDataframePartition(block)],
                pandas.RangeIndex(idx0, idx0 + n), ["k", "v"], [n],
                pandas.Series({"k": np.dtype(np.int64),
                               "v": np.dtype(np.float64)}))

        (ak, aoff) = shard(A_k)
        (av, _) = shard(A_v)
        (bk, boff) = shard(B_k)
        (bv, _) = shard(B_v)
        fa = frame(ak, av, aoff)
        fb = frame(bk, bv, boff)
        out = fa.concat_rows([fb])

        # gather every rank's shard in rank order == global pandas concat
        got_k = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(out._partitions[0].block().columns["k"])))
        got_v = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(out._partitions[0].block().columns["v"])))
        got_idx = np.concatenate(dist_mod._gather_np_varlen(
            np.asarray(out.index).astype(np.int64)))
        exp = pandas.concat([
            pandas.DataFrame({"k": A_k, "v": A_v}),
            pandas.DataFrame({"k": B_k, "v": B_v}),
        ])
        np.testing.assert_array_equal(got_k, exp["k"].to_numpy())
        np.testing.assert_allclose(got_v, exp["v"].to_numpy(), rtol=0)
        np.testing.assert_array_equal(got_idx, exp.index.to_numpy())
        dist_mod.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")


@pytest.mark.timeout(120)
@pytest.mark.parametrize("world", [2, 3])
def test_gloo_concat_global_order(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29530 + world
    procs = [ctx.Process(target=_concat_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def _melt_worker(rank, world, port, fail_q):
    """World>1 melt + column assignment through the REAL composition code
    on the numpy mock: melt's concat must land in GLOBAL pandas melt
    order (value column 0 over all ranks, then column 1, ...), with the
    dictionary-encoded `variable` column surviving the exchange."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import pandas
        import modin_amd.distributed as dist_mod
        from tests import mocklib

        class _RawPatch:
            def setattr(self, obj, name, fn):
                setattr(obj, name, fn)

        mocklib.install(_RawPatch())
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        import modin_amd.pandas as mpd
        from modin_amd.core import lib
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import (DeviceBlock,
                                              HipDataframePartition)
        from modin_amd.query_compiler import HipQueryCompiler

        rng = np.random.default_rng(77)  # same stream on all ranks
        n = 1200
        gid = rng.integers(0, 9, n).astype(np.int64)
        gx = rng.random(n)
        gy = rng.random(n)
        counts = oracle.split_row_counts(n, world, 1)
        offs = np.cumsum([0] + counts)
        sl = slice(offs[rank], offs[rank + 1])
        nl = counts[rank]
        block = DeviceBlock({"id": lib.put(gid[sl]), "x": lib.put(gx[sl]),
                             "y": lib.put(gy[sl])}, nl)
        frame = HipDataframe(
            [HipDataframePartition(block)],
            pandas.RangeIndex(offs[rank], offs[rank + 1]),
            ["id", "x", "y"], [nl],
            pandas.Series({"id": np.dtype(np.int64),
                           "x": np.dtype(np.float64),
                           "y": np.dtype(np.float64)}))
        df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))

        # column assignment stays rank-local (co-sharded operands)
        df["z"] = df["x"] - df["y"]
        got_z = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(df._query_compiler._modin_frame._partitions[0]
                    .block().columns["z"])))
        np.testing.assert_allclose(got_z, gx - gy, rtol=0)

        out = df[["id", "x", "y"]].melt(id_vars="id")._query_compiler\
            ._modin_frame
        blk = out._partitions[0].block()
        got_id = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(blk.columns["id"])))
        got_var = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(blk.columns["variable"])))
        got_val = np.concatenate(dist_mod._gather_np_varlen(
            lib.get(blk.columns["value"])))
        exp = pandas.DataFrame({"id": gid, "x": gx, "y": gy}).melt(
            id_vars="id")
        np.testing.assert_array_equal(got_id, exp["id"].to_numpy())
        cats = blk.cats["variable"]
        np.testing.assert_array_equal(
            cats.to_numpy(dtype=object)[got_var],
            exp["variable"].to_numpy())
        np.testing.assert_allclose(got_val, exp["value"].to_numpy(),
                                   rtol=0)
        dist_mod.shutdown()
    except Exception:  # noqa: BLE001
        import traceback
        fail_q.put(f"rank {rank}:\n{traceback.format_exc()}")
        raise SystemExit(1)


@pytest.mark.parametrize("world", [2])
def test_gloo_melt_setitem_global_order(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29555
    procs = [ctx.Process(target=_melt_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def _corr_worker(rank, world, port, fail_q):
    """World>1 corr/cov: the six masked moments per pair SUM-all-reduce
    across shards; result must equal pandas on the GLOBAL frame."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import pandas
        import modin_amd.distributed as dist_mod
        from tests import mocklib

        class _RawPatch:
            def setattr(self, obj, name, fn):
                setattr(obj, name, fn)

        mocklib.install(_RawPatch())
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        import modin_amd.pandas as mpd
        from modin_amd.core import lib
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import (DeviceBlock,
                                              HipDataframePartition)
        from modin_amd.query_compiler import HipQueryCompiler

        rng = np.random.default_rng(88)  # same stream on all ranks
        n = 3000
        gx = rng.standard_normal(n)
        gy = 0.6 * gx + rng.standard_normal(n)
        gx[rng.random(n) < 0.1] = np.nan
        gy[rng.random(n) < 0.1] = np.nan
        counts = oracle.split_row_counts(n, world, 1)
        offs = np.cumsum([0] + counts)
        sl = slice(offs[rank], offs[rank + 1])
        nl = counts[rank]
        block = DeviceBlock({"x": lib.put(gx[sl]),
                             "y": lib.put(gy[sl])}, nl)
        frame = HipDataframe(
            [HipDataframePartition(block)],
            pandas.RangeIndex(offs[rank], offs[rank + 1]),
            ["x", "y"], [nl],
            pandas.Series({"x": np.dtype(np.float64),
                           "y": np.dtype(np.float64)}))
        df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))
        pdf = pandas.DataFrame({"x": gx, "y": gy})
        np.testing.assert_allclose(df.corr().to_numpy(),
                                   pdf.corr().to_numpy(), rtol=1e-10)
        np.testing.assert_allclose(df.cov().to_numpy(),
                                   pdf.cov().to_numpy(), rtol=1e-10)
        dist_mod.shutdown()
    except Exception:  # noqa: BLE001
        import traceback
        fail_q.put(f"rank {rank}:\n{traceback.format_exc()}")
        raise SystemExit(1)


@pytest.mark.parametrize("world", [2])
def test_gloo_corr_allreduce(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29557
    procs = [ctx.Process(target=_corr_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def _natsort_worker(rank, world, port, fail_q):
    """World>1 sort_values by a NaT-bearing datetime key: the sentinel
    scheme must survive the range shuffle (NaT rows land LAST on the
    last rank for na_position='last')."""
    try:
        os.environ.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        })
        import pandas
        import modin_amd.distributed as dist_mod
        from tests import mocklib

        class _RawPatch:
            def setattr(self, obj, name, fn):
                setattr(obj, name, fn)

        mocklib.install(_RawPatch())
        assert dist_mod.init_from_env(backend="gloo", gpu=False)
        import modin_amd.pandas as mpd
        from modin_amd.core import lib
        from modin_amd.core.dataframe import HipDataframe
        from modin_amd.core.partition import (DeviceBlock,
                                              HipDataframePartition)
        from modin_amd.query_compiler import HipQueryCompiler

        rng = np.random.default_rng(99)  # same stream on all ranks
        n = 4000
        base = pandas.Timestamp("2019-06-01").value
        tns = base + rng.integers(0, 10**15, n)
        tns[rng.random(n) < 0.12] = np.iinfo(np.int64).min  # NaT
        gv = rng.standard_normal(n)
        counts = oracle.split_row_counts(n, world, 1)
        offs = np.cumsum([0] + counts)
        sl = slice(offs[rank], offs[rank + 1])
        nl = counts[rank]
        block = DeviceBlock({"t": lib.put(tns[sl]), "v": lib.put(gv[sl])},
                            nl)
        frame = HipDataframe(
            [HipDataframePartition(block)],
            pandas.RangeIndex(nl),
            ["t", "v"], [nl],
            pandas.Series({"t": np.dtype("datetime64[ns]"),
                           "v": np.dtype(np.float64)}))
        df = mpd.DataFrame(query_compiler=HipQueryCompiler(frame))
        for asc, nap in ((True, "last"), (False, "last"),
                         (True, "first")):
            out = df.sort_values("t", ascending=asc, na_position=nap)
            of = out._query_compiler._modin_frame
            got_t = np.concatenate(dist_mod._gather_np_varlen(
                lib.get(of._partitions[0].block().columns["t"])))
            got_pos = np.concatenate(dist_mod._gather_np_varlen(
                np.asarray(of.index).astype(np.int64)))
            pdf = pandas.DataFrame(
                {"t": tns.view("datetime64[ns]"), "v": gv})
            exp = pdf.sort_values("t", ascending=asc, na_position=nap,
                                  kind="stable")
            np.testing.assert_array_equal(
                got_t.view("datetime64[ns]"), exp["t"].to_numpy(),
                err_msg=f"{asc}/{nap}")
            np.testing.assert_array_equal(got_pos, exp.index.to_numpy(),
                                          err_msg=f"{asc}/{nap}")
        dist_mod.shutdown()
    except Exception:  # noqa: BLE001
        import traceback
        fail_q.put(f"rank {rank}:\n{traceback.format_exc()}")
        raise SystemExit(1)


@pytest.mark.parametrize("world", [2])
def test_gloo_natsort_global_order(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29559
    procs = [ctx.Process(target=_natsort_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=110)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)
