import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def oracle():
    """ctypes handle to the CPU oracle (test infrastructure only)."""
    from tests._oracle import Oracle
    return Oracle()


@pytest.fixture(scope="session", autouse=True)
def _build_oracle():
    so = os.path.join(REPO, "oracle", "liboracle_t9.so")
    if not os.path.exists(so):
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle")],
                       check=True, capture_output=True)
