import ctypes
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (run with -m gpu on a GPU box)"
    )


def pytest_collection_modifyitems(config, items):
    # Skip gpu tests automatically when no GPU is present and the user did not
    # explicitly select them.
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def _build_oracle():
    lib = REPO / "oracle" / "liboracle.so"
    subprocess.run(
        ["make", "-s", "-C", str(REPO / "oracle")],
        check=True,
        capture_output=True,
        text=True,
    )
    return lib


@pytest.fixture(scope="session")
def oracle():
    """ctypes handle to the CPU oracle (test infrastructure only)."""
    lib_path = _build_oracle()
    lib = ctypes.CDLL(str(lib_path))
    return lib
