import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a HIP device (MI355X)")


@pytest.fixture(scope="session")
def core():
    import gats_amd

    return gats_amd.core()


@pytest.fixture(scope="session")
def gpu(core):
    if core.gpu_device_count() == 0:
        pytest.skip("no HIP device")
    return core
