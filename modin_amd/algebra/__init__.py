"""Operator templates — the modin_amd form of Modin's dataframe algebra.

Mirrors ``modin/core/dataframe/algebra/``:
``Operator`` (operator.py:25), ``Map.register`` (map.py:28-70),
``TreeReduce.register`` (tree_reduce.py:29-82), ``Binary.register``
(binary.py:293-459, scalar branch :449 → lazy map, frame branch :420 →
n_ary_op), ``GroupByReduce.register`` (groupby.py:33-790) with the
map/reduce pairing table of ``GroupbyReduceImpl``
(storage_formats/pandas/groupby.py:26-113,237-248).

The reference registers pandas callables; here ``register`` takes a device
kernel descriptor (a hipframe op code) and the built caller runs hand-written
gfx950 HIP kernels through the partition layer instead of pandas — same
call shape ``caller(query_compiler, *args) -> query_compiler``.
"""

from __future__ import annotations

import numpy as np
import pandas

from ..core import lib
from ..core.partition import DeviceBlock


class Operator:
    """Builder base (reference: algebra/operator.py:25)."""

    def __init__(self):
        raise ValueError("Operator classes are static builders, not instances")

    @classmethod
    def register(cls, *args, **kwargs):
        raise NotImplementedError


def _map_block(op_code, scalar, f64_only=False):
    """Build a DeviceBlock->DeviceBlock elementwise kernel call."""

    def block_fn(block: DeviceBlock) -> DeviceBlock:
        if block.cats:
            raise lib.HfError(
                f"arithmetic on string column(s) {sorted(block.cats)}: "
                "select numeric columns")
        out = {}
        for name, col in block.columns.items():
            src = lib.cast_f64(col) if (f64_only and col.dtype_code == lib.HF_INT64) \
                else col
            out[name] = lib.map_scalar(op_code, src, scalar)
        return DeviceBlock(out, block.length)

    return block_fn


class Map(Operator):
    """Elementwise per-partition operator (reference: algebra/map.py:28)."""

    @classmethod
    def register(cls, op_code, f64_only=False):
        def caller(query_compiler, scalar, **kwargs):
            return query_compiler.__constructor__(
                query_compiler._modin_frame.map(
                    _map_block(op_code, scalar, f64_only=f64_only),
                    lazy=kwargs.get("lazy", False),
                )
            )

        return caller


class Binary(Operator):
    """Binary operator (reference: algebra/binary.py:293).

    Scalar ``other`` -> lazy Map (binary.py:449); frame ``other`` ->
    n_ary_op zip over co-partitioned frames (binary.py:420,
    dataframe.py:3851).
    """

    @classmethod
    def register(cls, map_op_code, bin_op_code, reverse_map_op_code=None):
        def caller(query_compiler, other, **kwargs):
            frame = query_compiler._modin_frame
            if np.isscalar(other):
                return query_compiler.__constructor__(
                    frame.map(_map_block(map_op_code, other), lazy=True)
                )

            other_frame = other._modin_frame

            def zip_fn(lblock: DeviceBlock, rblock: DeviceBlock) -> DeviceBlock:
                if lblock.cats or rblock.cats:
                    raise lib.HfError(
                        "binary ops on string columns: select numeric "
                        "columns")
                out = {}
                if lblock.width == 1 and rblock.width == 1:
                    # Series op Series: positional pairing, left name wins
                    (lname, lcol), = lblock.columns.items()
                    (rcol,) = rblock.columns.values()
                    out[lname] = lib.binary(bin_op_code, lcol, rcol)
                    return DeviceBlock(out, lblock.length)
                for name, lcol in lblock.columns.items():
                    rcol = rblock.columns.get(name)
                    if rcol is None:
                        raise lib.HfError(
                            f"binary frame op: column {name!r} missing on the "
                            "right (NaN-fill alignment is a later round)")
                    out[name] = lib.binary(bin_op_code, lcol, rcol)
                return DeviceBlock(out, lblock.length)

            return query_compiler.__constructor__(frame.n_ary_op(zip_fn, other_frame))

        return caller


class TreeReduce(Operator):
    """Tree-reduce operator (reference: algebra/tree_reduce.py:29).

    Map phase = the single-pass per-partition reduce kernel
    (sum/count/min/max partials in one HBM scan); reduce phase = host/RCCL
    combine of the 1-row partials (dataframe.py:2244-2247 device form).
    """

    _FINALIZERS = {
        "sum": lambda s, is_int: (s["isum"] if is_int else s["sum"]),
        "count": lambda s, is_int: s["count"],
        "mean": lambda s, is_int: (s["sum"] / s["count"]) if s["count"] else float("nan"),
        "min": lambda s, is_int: (s["imn"] if is_int else s["mn"]) if s["count"] else float("nan"),
        "max": lambda s, is_int: (s["imx"] if is_int else s["mx"]) if s["count"] else float("nan"),
    }

    @classmethod
    def register(cls, agg: str):
        if agg not in cls._FINALIZERS:
            raise lib.HfError(f"TreeReduce agg {agg!r} not implemented")
        fin = cls._FINALIZERS[agg]

        def caller(query_compiler, **kwargs):
            frame = query_compiler._modin_frame
            partials = frame.tree_reduce(frame.columns)
            vals, names = [], []
            for name in frame.columns:
                is_int = frame.dtypes[name] == np.dtype(np.int64)
                names.append(name)
                vals.append(fin(partials[name], is_int))
            dtype = (np.int64 if agg in ("sum", "count", "min", "max")
                     and all(frame.dtypes[n] == np.dtype(np.int64) for n in names)
                     and agg != "mean" else np.float64)
            if agg == "count":
                dtype = np.int64
            return pandas.Series(vals, index=pandas.Index(names), dtype=dtype)

        return caller


class Reduce(TreeReduce):
    """Single-phase reduce (reference: algebra/reduce.py:28) — identical on
    a p×1 device grid."""


class GroupByReduce(Operator):
    """Groupby map-reduce operator (reference: algebra/groupby.py:33).

    The map/reduce fn table mirrors GroupbyReduceImpl
    (storage_formats/pandas/groupby.py:237-248):
      sum   -> map "accumulate", reduce "table-merge", value sums
      count -> same kernels, value counts
      mean  -> sums/counts division on the compacted columns (:87-113 shape)
      min/max -> ds/global f64 min/max tables; empty groups fixed to NaN
    """

    SUPPORTED = ("sum", "count", "mean", "min", "max")

    @classmethod
    def register(cls, agg: str):
        if agg not in cls.SUPPORTED:
            raise lib.HfError(
                f"groupby agg {agg!r} not implemented (round-1 supports "
                "sum/count/mean/min/max)"
            )

        def caller(query_compiler, by: str, dropna: bool = True,
                   **kwargs):
            frame = query_compiler._modin_frame
            result = frame.groupby_reduce(by, agg, dropna=dropna)
            return query_compiler.__constructor__(result)

        return caller
