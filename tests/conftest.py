import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a ROCm GPU (run on MI355X via gpurun)")


def ensure_oracle_lib() -> str:
    """Build oracle/liboracle.so if missing (gcc, seconds)."""
    path = os.path.join(REPO, "oracle", "liboracle.so")
    if not os.path.exists(path):
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle"),
                        "liboracle.so"], check=True)
    return path


@pytest.fixture(scope="session")
def oracle_lib_path():
    return ensure_oracle_lib()
