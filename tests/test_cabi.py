"""CPU tier: the C-ABI library builds, loads and exports every declared
symbol (no compute calls — those need a GPU and fail loudly without one)."""

import re

import pytest

from modin_amd.core import lib


def test_so_loads_and_exports_all_symbols(built_so):
    dll = lib.load()
    for sym in lib.exported_symbols():
        assert getattr(dll, sym, None) is not None, f"missing symbol {sym}"


def test_header_symbols_covered(built_so):
    """Every hf_* function declared in include/hipframe.h is in the binding's
    export list (so the symbol check above is complete)."""
    import os
    hdr = os.path.join(os.path.dirname(lib.so_path()), "..", "..", "include",
                       "hipframe.h")
    with open(hdr) as f:
        text = f.read()
    declared = set(re.findall(r"\b(hf_[a-z0-9_]+)\s*\(", text))
    declared -= {"hf_col_free", }  # appears in comments too; keep set exact
    declared.add("hf_col_free")
    missing = declared - set(lib.exported_symbols())
    assert not missing, f"header declares symbols the binding misses: {missing}"


def test_compute_without_gpu_fails_loudly(built_so):
    """The product path has no CPU fallback: on a GPU-less box hf_init fails
    and compute raises HfError."""
    if lib.device_count() > 0:
        pytest.skip("GPU present — covered by the gpu tier instead")
    import numpy as np
    with pytest.raises(lib.HfError):
        lib.put(np.zeros(4))


def test_unsupported_dtype_message(built_so):
    import pandas
    from modin_amd.core.partition import DeviceBlock
    with pytest.raises(lib.HfError, match="int64/float64"):
        DeviceBlock.from_pandas(pandas.DataFrame({"s": ["a", "b"]}))
