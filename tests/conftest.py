import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
NATIVE = os.path.join(REPO, "tensor_fusion_amd", "_native")

sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers", "slow: long-running CPU benchmark test")


@pytest.fixture(scope="session")
def native_built():
    """Build the CPU-side native artifacts once per test session."""

    subprocess.run([sys.executable, os.path.join(REPO, "build_native.py")],
                   check=True, capture_output=True)
    return NATIVE
