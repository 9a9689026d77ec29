import glob
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

GOLDEN_DIR = os.path.join(REPO, "tests", "golden")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a visible MI355X (run via gpurun)"
    )
    config.addinivalue_line(
        "markers", "slow: multi-GB / multi-minute case (run explicitly)"
    )


@pytest.fixture(scope="session")
def built_so():
    """Ensure libhipframe.so exists (hipcc cross-compiles on CPU boxes)."""
    so = os.path.join(REPO, "modin_amd", "csrc", "libhipframe.so")
    if not os.path.exists(so):
        subprocess.run(["make", "-C", os.path.dirname(so)], check=True)
    return so


@pytest.fixture(scope="session")
def gpu_ready(built_so):
    """hf_init on GPU 0 — skips (not passes) when no GPU is visible, but on a
    GPU box a failed init is a hard failure (no silent fallback)."""
    from modin_amd.core import lib
    if lib.device_count() == 0:
        pytest.skip("no HIP device visible")
    lib.ensure_ready(0)
    return True


def golden_cases(prefix):
    return sorted(
        os.path.basename(p)[: -len(".npz")]
        for p in glob.glob(os.path.join(GOLDEN_DIR, f"{prefix}*.npz"))
    )


def load_golden(name):
    return dict(np.load(os.path.join(GOLDEN_DIR, f"{name}.npz")))
