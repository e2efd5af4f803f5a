import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def oracle_lib():
    """Build the CPU oracle if missing, return the capi module."""
    so = os.path.join(REPO, "oracle", "libhbls_oracle.so")
    if not os.path.exists(so):
        subprocess.check_call(["make", "-C", os.path.join(REPO, "oracle")])
    from oracle import capi
    return capi
