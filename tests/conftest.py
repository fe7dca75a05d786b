import os
import subprocess
import sys
import pathlib

import pytest

REPO = pathlib.Path(__file__).resolve().parent.parent


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def _have_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


def pytest_collection_modifyitems(config, items):
    if _have_gpu():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def oracle_lib():
    """Builds (if needed) and loads the CPU oracle shared library."""
    so = REPO / "oracle" / "liboracle.so"
    if not so.exists():
        subprocess.run(["make", "-C", str(REPO / "oracle"), "liboracle.so"],
                       check=True)
    sys.path.insert(0, str(REPO))
    from tests.oracle_binding import OracleLib
    return OracleLib(str(so))
