"""modin_amd — an MI355X-native execution backend for Modin's
partition-operator path (BASELINE.json north_star).

Layers (mirroring the reference's architecture, L1->L6):
  modin_amd.pandas           — drop-in API surface for the hot path (L1)
  modin_amd.query_compiler   — HipQueryCompiler (L2), reference method names
  modin_amd.algebra          — Map/TreeReduce/Binary/GroupByReduce templates (L3)
  modin_amd.core.dataframe   — HipDataframe (L4)
  modin_amd.core.partition*  — device-block partitions + manager (L5)
  modin_amd.core.lib         — ctypes boundary into libhipframe.so (L6):
                               hand-written gfx950 HIP kernels; see
                               include/hipframe.h and INTEGRATION.md.
"""

from . import config  # noqa: F401

__version__ = "0.1.0"
