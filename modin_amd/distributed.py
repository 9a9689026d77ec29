"""Multi-GPU execution: one process per GPU, torch.distributed over RCCL.

The reference has no explicit collectives — data movement is Ray/Dask futures
(SURVEY.md §5.8).  The MI355X-native equivalent: every rank owns the row
shard of the frame it built (SPMD), map/binary/filter run with zero exchange
("weak" scaling), and the two exchange points are

  * TreeReduce: host-side combine of tiny per-rank partials (gloo/nccl
    all-reduce of a few doubles),
  * GroupByReduce: RCCL all-reduce of the dense key-indexed table
    (sums f64 + rowcnt/counts u64) over xGMI — the device form of the
    reference's reduce phase (algebra/groupby.py:211), chosen per SURVEY §8e
    option (i) for bounded key spaces (1e6 keys → 8–24 MB buffers).

torch is plumbing here: device allocation for RCCL-visible buffers and the
process group.  Kernels remain the hipframe HIP kernels, which write into the
torch-allocated table via raw device pointers on the hipframe stream
(hf_sync before the collective, torch.cuda.synchronize after).
"""

from __future__ import annotations

import os

_state = {"active": False, "rank": 0, "world": 1, "device": None, "backend": None}


def is_active() -> bool:
    return _state["active"]


def rank() -> int:
    return _state["rank"]


def world_size() -> int:
    return _state["world"]


def init_from_env(backend=None, gpu: bool = True):
    """Initialise from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return False
    import torch
    import torch.distributed as dist
    r = int(os.environ.get("RANK", "0"))
    local = int(os.environ.get("LOCAL_RANK", str(r)))
    if backend is None:
        backend = os.environ.get("MODIN_AMD_DIST_BACKEND")
    if backend is None:
        backend = "nccl" if (gpu and torch.cuda.is_available()) else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    _state.update(active=True, rank=r, world=world, backend=backend)
    if gpu and torch.cuda.is_available():
        # oversubscription fallback (testing N ranks on fewer GPUs)
        dev = local % torch.cuda.device_count()
        torch.cuda.set_device(dev)
        # small collectives run on the backend's native device; the groupby
        # TABLE always lives on the GPU (hipframe kernels write it) and is
        # CPU-bounced for gloo collectives in maybe_allreduce_table
        _state["device"] = f"cuda:{dev}" if backend == "nccl" else "cpu"
        _state["table_device"] = f"cuda:{dev}"
        os.environ["MODIN_AMD_GPU"] = str(dev)
    else:
        _state["device"] = "cpu"
        _state["table_device"] = "cpu"
    return True


class local_mode:
    """Temporarily mark the distributed state inactive: after a key-range
    shuffle each rank owns disjoint groups, so the LOCAL single-rank
    engine must run without the dense-table all-reduce re-merging them
    (the shuffle-then-local recipe of _groupby_shuffle, applied at the
    frame level for median/quantile/first/last/transforms)."""

    def __enter__(self):
        self._prev = _state["active"]
        _state["active"] = False
        return self

    def __exit__(self, *exc):
        _state["active"] = self._prev
        return False


def shutdown():
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    _state.update(active=False, rank=0, world=1, device=None, backend=None)


def barrier():
    if not is_active():
        return
    import torch.distributed as dist
    dist.barrier()


def allreduce_minmax(kmin, kmax):
    """Global [min, max] of per-rank key ranges (None on empty shards)."""
    import torch
    import torch.distributed as dist
    dev = _state["device"]
    big = 2**62
    t = torch.tensor(
        [kmin if kmin is not None else big, -(kmax if kmax is not None else -big)],
        dtype=torch.int64, device=dev,
    )
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    lo, hi = int(t[0].item()), -int(t[1].item())
    if lo == big:
        return None, None
    return lo, hi


def allreduce_partials(partials: dict, names: list) -> dict:
    """Cross-rank reduce phase of TreeReduce: combine the per-rank 1-row
    partial dicts (sum/count/min/max) — the device grid's p-way combine
    (dataframe.py:2244-2247) extended over ranks with three tiny collectives.
    Empty-shard ranks contribute identity elements."""
    import torch
    import torch.distributed as dist
    dev = _state["device"]
    inf = float("inf")
    big = 2**62
    add, mns, mxs = [], [], []
    for n in names:
        p = partials[n]
        add += [p["sum"], float(p["count"])]
        mns += [p["mn"] if p["count"] else inf,
                float(p["imn"]) if p["count"] else inf]
        mxs += [p["mx"] if p["count"] else -inf,
                float(p["imx"]) if p["count"] else -inf]
    t_add = torch.tensor(add, dtype=torch.float64, device=dev)
    t_mn = torch.tensor(mns, dtype=torch.float64, device=dev)
    t_mx = torch.tensor(mxs, dtype=torch.float64, device=dev)
    t_isum = torch.tensor([partials[n]["isum"] for n in names],
                          dtype=torch.int64, device=dev)
    t_iminmax = torch.tensor(
        [(partials[n]["imn"] if partials[n]["count"] else big) for n in names]
        + [-(partials[n]["imx"] if partials[n]["count"] else -big) for n in names],
        dtype=torch.int64, device=dev)
    dist.all_reduce(t_add)
    dist.all_reduce(t_isum)
    dist.all_reduce(t_mn, op=dist.ReduceOp.MIN)
    dist.all_reduce(t_mx, op=dist.ReduceOp.MAX)
    dist.all_reduce(t_iminmax, op=dist.ReduceOp.MIN)
    out = {}
    for i, n in enumerate(names):
        cnt = int(t_add[2 * i + 1].item())
        out[n] = {
            "sum": float(t_add[2 * i].item()),
            "count": cnt,
            "isum": int(t_isum[i].item()),
            "mn": float(t_mn[2 * i].item()) if cnt else float("nan"),
            "mx": float(t_mx[2 * i].item()) if cnt else float("nan"),
            "imn": int(t_iminmax[i].item()) if cnt else 0,
            "imx": -int(t_iminmax[len(names) + i].item()) if cnt else 0,
        }
    return out


def allreduce_scalars(values):
    """Sum-all-reduce a small list of floats (TreeReduce partial combine)."""
    import torch
    import torch.distributed as dist
    t = torch.tensor(values, dtype=torch.float64, device=_state["device"])
    dist.all_reduce(t)
    return t.tolist()


def alloc_table_torch(nvals: int, n_slots: int, want_counts: bool,
                      init: float = 0.0):
    """RCCL-reducible table buffers on this rank's GPU (value table filled
    with the agg identity: 0 / +inf / -inf).

    Returns (keepalive, sums_ptr, rowcnt_ptr, counts_ptr).
    """
    import torch
    dev = _state.get("table_device") or _state["device"]
    sums = torch.full((nvals * n_slots,), init, dtype=torch.float64,
                      device=dev)
    rowcnt = torch.zeros(n_slots, dtype=torch.int64, device=dev)
    counts = (torch.zeros(nvals * n_slots, dtype=torch.int64, device=dev)
              if want_counts else None)
    if dev != "cpu":
        torch.cuda.synchronize()
    keep = (sums, rowcnt, counts)
    return keep, sums.data_ptr(), rowcnt.data_ptr(), \
        (counts.data_ptr() if counts is not None else 0)


def exchange_splits(t, send_counts):
    """All-to-all row exchange — the communication step of the
    range/hash-partitioning shuffle (SURVEY §8e option (ii): the device form
    of ``PartitionManager.shuffle_partitions``, partition_manager.py:1937,
    whose p-way split becomes a P-rank exchange over xGMI).

    ``t``: 1-D tensor laid out as P contiguous spans, span i destined for
    rank i with ``send_counts[i]`` elements.  Returns (recv_tensor,
    recv_counts).  nccl uses all_to_all_single (pairwise xGMI); gloo (CPU
    test tier) emulates with ordered pairwise send/recv.
    """
    import torch
    import torch.distributed as dist
    P = world_size()
    r = rank()
    send_counts = [int(c) for c in send_counts]
    assert len(send_counts) == P and sum(send_counts) == t.numel()
    cnt = torch.tensor(send_counts, dtype=torch.int64, device=_state["device"])
    all_cnt = [torch.zeros_like(cnt) for _ in range(P)]
    dist.all_gather(all_cnt, cnt)
    recv_counts = [int(all_cnt[src][r].item()) for src in range(P)]
    out = torch.empty(sum(recv_counts), dtype=t.dtype, device=t.device)
    if _state["backend"] == "nccl":
        dist.all_to_all_single(out, t, recv_counts, send_counts)
        torch.cuda.synchronize()
        return out, recv_counts
    # gloo fallback: keep own span, pairwise exchange the rest in ring order
    send_offs = [0]
    for c in send_counts:
        send_offs.append(send_offs[-1] + c)
    recv_offs = [0]
    for c in recv_counts:
        recv_offs.append(recv_offs[-1] + c)
    out[recv_offs[r]:recv_offs[r + 1]] = t[send_offs[r]:send_offs[r + 1]]
    for step in range(1, P):
        to = (r + step) % P
        frm = (r - step) % P
        sbuf = t[send_offs[to]:send_offs[to + 1]].contiguous()
        rbuf = torch.empty(recv_counts[frm], dtype=t.dtype, device=t.device)
        if r % 2 == 0:  # deadlock-free ordering; both sides know the counts
            if sbuf.numel():
                dist.send(sbuf, to)
            if rbuf.numel():
                dist.recv(rbuf, frm)
        else:
            if rbuf.numel():
                dist.recv(rbuf, frm)
            if sbuf.numel():
                dist.send(sbuf, to)
        out[recv_offs[frm]:recv_offs[frm + 1]] = rbuf
    return out, recv_counts


def _allgather_varlen(t):
    """All-gather a 1-D tensor of per-rank-varying length: size exchange +
    pad-to-max all_gather (NCCL has no allgatherv) + trim.  Device tensor
    collectives throughout — no host pickling.
    Returns the list of per-rank tensors on the collective device."""
    import torch
    import torch.distributed as dist
    P = world_size()
    dev = _state["device"]
    t = t.to(dev)
    n = torch.tensor([t.numel()], dtype=torch.int64, device=dev)
    sizes = [torch.zeros_like(n) for _ in range(P)]
    dist.all_gather(sizes, n)
    sizes = [int(s.item()) for s in sizes]
    m = max(sizes + [1])
    pad = torch.zeros(m, dtype=t.dtype, device=dev)
    if t.numel():
        pad[: t.numel()] = t
    outs = [torch.empty(m, dtype=t.dtype, device=dev) for _ in range(P)]
    dist.all_gather(outs, pad)
    return [outs[i][: sizes[i]] for i in range(P)]


def _gather_np_varlen(arr):
    """numpy in, rank-order list of numpy out, via _allgather_varlen."""
    import numpy as np
    import torch
    t = torch.from_numpy(np.ascontiguousarray(arr))
    return [g.cpu().numpy() for g in _allgather_varlen(t)]


def sample_splitters(local_sample):
    """world-1 key splitters from the gathered per-rank samples — the
    device form of the reference's RangePartitioning sampling
    (range-partitioning pre-shuffle: sample each partition, combine, take
    quantiles).  Deterministic and identical on every rank: all_gather is
    rank-ordered and the quantile rule is pure."""
    import numpy as np
    gathered = _gather_np_varlen(np.asarray(local_sample, dtype=np.int64))
    alls = np.sort(np.concatenate([g for g in gathered if len(g)] or
                                  [np.empty(0, dtype=np.int64)]))
    P = world_size()
    if alls.size == 0 or P <= 1:
        return np.empty(0, dtype=np.int64)
    qs = [(i * alls.size) // P for i in range(1, P)]
    return alls[qs]


class _CAIView:
    """Zero-copy view of an hf device column for torch (the send side of the
    RCCL exchange reads hipframe memory directly over xGMI — no staging
    mirror).  The ColumnRef must outlive the view (caller scope does)."""

    def __init__(self, ptr: int, length: int, typestr: str):
        self.__cuda_array_interface__ = {
            "shape": (length,),
            "typestr": typestr,
            "data": (ptr, False),
            "strides": None,
            "version": 2,
        }


def _as_torch_view(col):
    """torch CUDA tensor aliasing an hf column (no copy), or None if this
    torch build rejects the __cuda_array_interface__ handoff."""
    import torch
    from .core import lib
    ts = "<i8" if col.dtype_code == lib.HF_INT64 else "<f8"
    try:
        t = torch.as_tensor(_CAIView(col.dptr(), col.length, ts),
                            device=_state["device"])
        if t.data_ptr() != col.dptr():
            return None  # a copy was made onto the torch allocator — fine too
        return t
    except Exception:
        return None


def exchange_column(col, send_counts):
    """Move one hf column's P destination spans to their ranks
    (exchange_splits under the hood) and return the received rows as a new
    hf column.  nccl: the send buffer is a zero-copy torch view of the hf
    column (RCCL reads it in place over xGMI) and the received rows land in
    one device copy into hf-owned memory; gloo (CPU test tier / single-GPU
    multi-rank validation): bounced via host numpy."""
    import numpy as np
    import torch
    from .core import lib
    tdt = torch.int64 if col.dtype_code == lib.HF_INT64 else torch.float64
    if _state["backend"] == "nccl":
        t = _as_torch_view(col) if col.length else None
        if t is None:
            t = torch.empty(col.length, dtype=tdt, device=_state["device"])
            if col.length:
                lib.memcpy_dd(t.data_ptr(), col.dptr(), 8 * col.length)
        lib.sync()  # hf-stream writes must land before the collective
        out, recv_counts = exchange_splits(t, send_counts)
        recv = lib.alloc(out.numel(), col.dtype_code)
        if out.numel():
            lib.memcpy_dd(recv.dptr(), out.data_ptr(), 8 * out.numel())
            lib.sync()
        return recv
    host = lib.get(col)
    t = torch.from_numpy(np.ascontiguousarray(host))
    out, recv_counts = exchange_splits(t, send_counts)
    return lib.put(out.numpy())


def allgather_groupby(keys_np, sums_np, counts_np):
    """Concatenate the per-rank (disjoint, ascending-range) groupby results
    in rank order so every rank returns the identical replicated frame —
    the same output convention as the dense-table all-reduce path."""
    import numpy as np
    gk = np.concatenate(_gather_np_varlen(np.asarray(keys_np, np.int64)))
    nv = len(sums_np)
    gs = [np.concatenate(_gather_np_varlen(
        np.asarray(sums_np[c], np.float64))) for c in range(nv)]
    gc = ([np.concatenate(_gather_np_varlen(
        np.asarray(counts_np[c], np.int64))) for c in range(nv)]
        if counts_np is not None else None)
    return gk, gs, gc


def allgather_arrays(arrays):
    """All-gather a list of per-rank numpy arrays; returns the rank-order
    concatenation of each (the host form of ncclAllGatherv for the
    broadcast-join right side and the sorted-result merge)."""
    import numpy as np
    return [np.concatenate(_gather_np_varlen(np.ascontiguousarray(a)))
            for a in arrays]


def allgather_lengths(local_n: int) -> list:
    """Every rank's shard length, in rank order (one tiny i64 all_gather)."""
    import torch
    import torch.distributed as dist
    dev = _state["device"]
    t = torch.tensor([int(local_n)], dtype=torch.int64, device=dev)
    outs = [torch.zeros_like(t) for _ in range(world_size())]
    dist.all_gather(outs, t)
    return [int(o.item()) for o in outs]


def global_row_base(local_n: int) -> int:
    """This rank's global row offset: sum of all earlier ranks' shard
    lengths (the SPMD frame is the rank-order concat of the shards)."""
    return sum(allgather_lengths(local_n)[: rank()])


def maybe_allreduce_table(table) -> None:
    """RCCL all-reduce of the dense groupby table (reduce phase across GPUs)."""
    if not is_active():
        return
    import torch
    import torch.distributed as dist
    sums, rowcnt, counts = table._torch_tensors
    if getattr(sums, "is_cuda", False):
        from .core import lib
        lib.sync()  # hipframe-stream accumulation must be visible to RCCL
    agg_op = getattr(table, "agg_op", 0)
    op = (dist.ReduceOp.MIN if agg_op == 1
          else dist.ReduceOp.MAX if agg_op == 2 else dist.ReduceOp.SUM)

    def reduce_(t, o):
        if t.is_cuda and _state["backend"] != "nccl":
            # gloo cannot reduce device tensors: bounce via host (test path)
            h = t.cpu()
            dist.all_reduce(h, op=o)
            t.copy_(h)
        else:
            dist.all_reduce(t, op=o)

    reduce_(sums, op)
    reduce_(rowcnt, dist.ReduceOp.SUM)
    if counts is not None:
        reduce_(counts, dist.ReduceOp.SUM)
    if sums.is_cuda:
        torch.cuda.synchronize()  # collective must land before compaction reads
