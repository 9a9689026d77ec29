#!/usr/bin/env python3
"""Time sort_values at 1e9 rows (device-born frame) — the r1 figure was
~85-90 ms (1e6-key range, 3 narrow passes); bit-ballot should land ~55-65."""
import sys, time, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import pandas
from modin_amd.core import lib
from modin_amd.core.dataframe import HipDataframe
from modin_amd.core.partition import DeviceBlock, HipDataframePartition
from modin_amd.query_compiler import HipQueryCompiler

n = int(sys.argv[1]) if len(sys.argv) > 1 else 1_000_000_000
span = int(sys.argv[2]) if len(sys.argv) > 2 else 1_000_000
lib.ensure_ready(0)
k = lib.fill_randint(n, 42, 0, span)
v = lib.fill_randf64(n, 43)
frame = HipDataframe(
    [HipDataframePartition(DeviceBlock({"k": k, "v": v}, n))],
    pandas.RangeIndex(n), ["k", "v"], [n],
    pandas.Series({"k": np.dtype(np.int64), "v": np.dtype(np.float64)}))
qc = HipQueryCompiler(frame)
qc.sort_rows_by_column_values("k")  # warm
lib.sync()
for r in range(3):
    t0 = time.perf_counter()
    out = qc.sort_rows_by_column_values("k")
    lib.sync()
    print(f"sort_values {n} rows span={span}: "
          f"{(time.perf_counter()-t0)*1e3:.1f} ms")
# spot parity at the tails
res = out._modin_frame
head = res.to_pandas().head(3)["k"].to_numpy()
assert (np.diff(head) >= 0).all()
print("ok")
