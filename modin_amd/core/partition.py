"""Device-resident block partition.

Mirrors the abstract partition protocol of
``modin/core/dataframe/pandas/partitioning/partition.py``:
``apply`` (:114), ``add_to_apply_calls`` (:140 — lazy call queue),
``drain_call_queue`` (:174), ``put`` (:277), ``wait``, and the
length/width caches.  The payload is not a pandas.DataFrame but a
``DeviceBlock``: an ordered set of HIP device column buffers (SoA), one
per dataframe column — the MI355X-native partition format.

Laziness: like the reference, ``add_to_apply_calls`` records
``[func, args, kwargs]`` and ``drain_call_queue`` executes them in order
(partition.py:167-172).  Because every kernel is enqueued on one HIP stream,
draining is itself asynchronous on the device; ``wait`` is a stream sync.
"""

from __future__ import annotations

import numpy as np
import pandas

from . import lib


class DeviceBlock:
    """Ordered mapping column-name -> lib.ColumnRef, all of equal length."""

    __slots__ = ("columns", "length")

    def __init__(self, columns: dict, length: int):
        self.columns = columns  # name -> ColumnRef
        self.length = length

    @classmethod
    def from_pandas(cls, df: pandas.DataFrame) -> "DeviceBlock":
        cols = {}
        for name in df.columns:
            arr = df[name].to_numpy()
            if arr.dtype not in (np.dtype(np.int64), np.dtype(np.float64)):
                raise lib.HfError(
                    f"column {name!r} has dtype {arr.dtype}: the HipNative "
                    "backend stores int64/float64 device columns only"
                )
            cols[name] = lib.put(arr)
        return cls(cols, len(df))

    def to_pandas(self, index=None) -> pandas.DataFrame:
        data = {name: lib.get(col) for name, col in self.columns.items()}
        if index is None:
            index = pandas.RangeIndex(self.length)
        return pandas.DataFrame(data, index=index)

    def select(self, names) -> "DeviceBlock":
        return DeviceBlock({n: self.columns[n] for n in names}, self.length)

    @property
    def width(self) -> int:
        return len(self.columns)


class HipDataframePartition:
    """One 2-D-grid cell: a DeviceBlock plus a lazy call queue."""

    def __init__(self, block: DeviceBlock, call_queue=None):
        self._block = block
        self.call_queue = call_queue or []

    # -- reference protocol (partition.py:114) --
    def apply(self, func, *args, **kwargs):
        """Drain the queue, then apply func(DeviceBlock)->DeviceBlock."""
        self.drain_call_queue()
        return HipDataframePartition(func(self._block, *args, **kwargs))

    def add_to_apply_calls(self, func, *args, **kwargs):
        """(partition.py:140) — lazily queue func; returns a NEW partition."""
        return HipDataframePartition(
            self._block, call_queue=self.call_queue + [[func, args, kwargs]]
        )

    def drain_call_queue(self):
        """(partition.py:174) — run queued calls in insertion order."""
        if not self.call_queue:
            return
        block = self._block
        for func, args, kwargs in self.call_queue:
            block = func(block, *args, **kwargs)
        self._block = block
        self.call_queue = []

    def wait(self):
        """Stream-sync the device (the HIP analog of future.wait)."""
        self.drain_call_queue()
        lib.sync()

    @classmethod
    def put(cls, df: pandas.DataFrame) -> "HipDataframePartition":
        """(partition.py:277) — H2D upload of a pandas block."""
        return cls(DeviceBlock.from_pandas(df))

    def get(self) -> pandas.DataFrame:
        self.drain_call_queue()
        return self._block.to_pandas()

    def block(self) -> DeviceBlock:
        self.drain_call_queue()
        return self._block

    def length(self) -> int:
        return self._block.length

    def width(self) -> int:
        return self._block.width
