import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import nezha_amd as nz  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


def pytest_collection_modifyitems(config, items):
    if nz.hip_device_count() > 0:
        return
    skip = pytest.mark.skip(reason="no HIP GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def cpu_ctx():
    return nz.Ctx(nz.CPU)


@pytest.fixture(scope="session")
def hip_ctx():
    assert nz.hip_device_count() > 0
    return nz.Ctx(nz.HIP, 0)
