import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X) and ROCm runtime")


@pytest.fixture(scope="session")
def repo_root():
    return REPO


@pytest.fixture(scope="session")
def native_built():
    """Build native helpers once per test session (host-only parts always work;
    HIP parts cross-compile without a GPU)."""
    from sofa_amd.native.build import build_all

    build_all()
    return True
