import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

ORACLE = os.path.join(REPO, "oracle", "bin", "oracle_tool")
GOLDEN = os.path.join(REPO, "tests", "golden")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (runs via gpurun)")


@pytest.fixture(scope="session")
def oracle_bin():
    """Build the CPU oracle if needed (works on both the CPU container and GPU box)."""
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   stdout=subprocess.DEVNULL)
    assert os.path.exists(ORACLE)
    return ORACLE


@pytest.fixture(scope="session")
def product_lib():
    """Build the HIP library if missing (hipcc cross-compiles without a GPU)."""
    so = os.path.join(REPO, "cassandra_amd", "libcassandra_gpucompact.so")
    if not os.path.exists(so):
        subprocess.run([
            "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC", "-shared",
            os.path.join(REPO, "cassandra_amd", "csrc", "gpucompact.cpp"), "-o", so,
        ], check=True)
    return so


def oracle_run(*args, check=True):
    return subprocess.run([ORACLE, *[str(a) for a in args]], check=check,
                          capture_output=True, text=True)
