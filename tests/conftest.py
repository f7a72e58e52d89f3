import os
import subprocess
import sys

# Deterministic import order: arrow's C++ runtime loads BEFORE torch/HIP.
# (A rare collection-time segfault was observed once with the reverse order;
# pinning the order here removes the variability.)
try:
    import pyarrow  # noqa: F401
    import pyarrow.parquet  # noqa: F401
except ImportError:
    pass

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def _ensure_oracle_built():
    lib = os.path.join(REPO, "oracle", "liboracle.so")
    src = os.path.join(REPO, "oracle", "oracle.c")
    if not os.path.exists(lib) or os.path.getmtime(lib) < os.path.getmtime(src):
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                       capture_output=True)


@pytest.fixture(scope="session")
def oracle():
    _ensure_oracle_built()
    from oracle import Oracle
    return Oracle()
