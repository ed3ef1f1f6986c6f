import os
import sys

import pytest

# Make the in-tree package importable without installation.
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def fake_lib():
    from k8s_dra_driver_amd.hal import FakeDeviceLib

    lib = FakeDeviceLib()
    lib.open()
    yield lib
    lib.close()
