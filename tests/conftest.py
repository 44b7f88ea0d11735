import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def fixture_8x():
    from kubegpu_amd.discovery import fixtures

    return fixtures.fixture_8x_mi355x()


@pytest.fixture
def fixture_2hive():
    from kubegpu_amd.discovery import fixtures

    return fixtures.fixture_2hive_8gpu()


@pytest.fixture
def fixture_no_xgmi():
    from kubegpu_amd.discovery import fixtures

    return fixtures.fixture_4x_no_xgmi()
