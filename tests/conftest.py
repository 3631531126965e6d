import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

LIB_DIR = os.path.join(REPO, "library")
LIB_SO = os.path.join(LIB_DIR, "build", "libvgpu-control.so")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def built_library():
    """Build the C library + test binaries once per session."""
    subprocess.run(["make", "-s", "all"], cwd=LIB_DIR, check=True)
    return LIB_SO


@pytest.fixture(scope="session")
def built_core():
    """Build only the CPU-side core binaries (probe + nogpu tests)."""
    subprocess.run(["make", "-s", "core"], cwd=LIB_DIR, check=True)
    return os.path.join(LIB_DIR, "build")
