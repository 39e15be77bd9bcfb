import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def core():
    from elbencho_amd import load_core

    return load_core()


@pytest.fixture(scope="session")
def gpu_available(core):
    return core.gpu_device_count() > 0
