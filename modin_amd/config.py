"""Environment-variable config, mirroring the reference's plugin knobs.

Reference: ``modin/config/envvars.py`` — ``NPartitions`` (:837, default =
CpuCount there; here the natural default is one partition per GPU process),
``MinRowPartitionSize`` (:1149, default 32), ``BenchmarkMode`` (:950, wraps
every partition-manager op in a synchronization barrier —
``partition_manager.py:52-92``), ``GpuCount`` (:818).
"""

from __future__ import annotations

import os


class _Option:
    varname: str = ""
    default = None
    _value = None

    @classmethod
    def get(cls):
        if cls._value is not None:
            return cls._value
        env = os.environ.get(cls.varname)
        if env is not None:
            return cls._cast(env)
        return cls.default

    @classmethod
    def put(cls, value):
        cls._value = value

    @classmethod
    def _cast(cls, s):
        return s


class _IntOption(_Option):
    @classmethod
    def _cast(cls, s):
        return int(s)


class _BoolOption(_Option):
    @classmethod
    def _cast(cls, s):
        return s.lower() in ("1", "true", "yes")


class NPartitions(_IntOption):
    """Row partitions per frame (reference: envvars.py:837)."""
    varname = "MODIN_AMD_NPARTITIONS"
    default = 1


class MinRowPartitionSize(_IntOption):
    """Minimum rows per partition (reference: envvars.py:1149)."""
    varname = "MODIN_AMD_MIN_ROW_PARTITION_SIZE"
    default = 32


class BenchmarkMode(_BoolOption):
    """Synchronize after every partition-manager op (reference: envvars.py:950;
    our barrier is hf_sync == hipStreamSynchronize)."""
    varname = "MODIN_AMD_BENCHMARK_MODE"
    default = False


class GpuCount(_IntOption):
    """GPUs per node (reference: envvars.py:818 — exists there, unused)."""
    varname = "MODIN_AMD_GPU_COUNT"
    default = 1


class MaxGroupbySlots(_IntOption):
    """Dense groupby table slot cap; key ranges beyond this raise (the hash
    fallback path is a later round — SURVEY.md §7 hard part (a))."""
    varname = "MODIN_AMD_MAX_GROUPBY_SLOTS"
    default = 1 << 27
