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


def decode_dict(codes: np.ndarray, cats: pandas.Index) -> np.ndarray:
    """codes (int64, −1 = NaN) -> object array of category values/NaN."""
    out = np.empty(len(codes), dtype=object)
    valid = codes >= 0
    out[valid] = cats.to_numpy(dtype=object)[codes[valid]]
    out[~valid] = np.nan
    return out


def encode_dict(values: pandas.Series):
    """values -> (int64 codes with −1 = NaN, sorted category Index).

    Sorted categories make code order == value order, so device-side sort
    and range comparisons run directly on codes."""
    codes, cats = pandas.factorize(values, sort=True)
    return codes.astype(np.int64), pandas.Index(cats)


class DeviceBlock:
    """Ordered mapping column-name -> lib.ColumnRef, all of equal length.

    String/object columns are DICTIONARY-ENCODED: the device
    column holds int64 codes (−1 = NaN), ``cats[name]`` holds the host-side
    sorted category Index (code order == lexicographic order, so sort /
    range compares work in code space).  The dictionary is built ONCE per
    frame at ingestion (HipDataframe.from_pandas) and shared by all
    partitions; blocks only carry the reference.
    """

    __slots__ = ("columns", "length", "cats")

    def __init__(self, columns: dict, length: int, cats: dict = None):
        self.columns = columns  # name -> ColumnRef
        self.length = length
        self.cats = cats or {}  # name -> pandas.Index (dict-encoded cols)

    @classmethod
    def from_pandas(cls, df: pandas.DataFrame,
                    cats: dict = None) -> "DeviceBlock":
        """``df`` must already be device-typed: int64/float64 columns, with
        string columns pre-encoded to int64 codes (listed in ``cats``)."""
        cols = {}
        for name in df.columns:
            arr = df[name].to_numpy()
            if arr.dtype not in (np.dtype(np.int64), np.dtype(np.float64)):
                raise lib.HfError(
                    f"column {name!r} has dtype {arr.dtype}: the HipNative "
                    "backend stores int64/float64 device columns only"
                )
            cols[name] = lib.put(arr)
        mycats = {n: c for n, c in (cats or {}).items() if n in cols}
        return cls(cols, len(df), mycats)

    def to_pandas(self, index=None) -> pandas.DataFrame:
        data = {}
        for name, col in self.columns.items():
            arr = lib.get(col)
            if name in self.cats:
                arr = decode_dict(arr, self.cats[name])
            data[name] = arr
        if index is None:
            index = pandas.RangeIndex(self.length)
        return pandas.DataFrame(data, index=index)

    def select(self, names) -> "DeviceBlock":
        return DeviceBlock({n: self.columns[n] for n in names}, self.length,
                           {n: self.cats[n] for n in names if n in self.cats})

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
    def put(cls, df: pandas.DataFrame,
            cats: dict = None) -> "HipDataframePartition":
        """(partition.py:277) — H2D upload of a pandas block (string
        columns pre-encoded; ``cats`` holds their dictionaries)."""
        return cls(DeviceBlock.from_pandas(df, cats))

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
