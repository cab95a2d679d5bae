import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    # nothing filtered here; the driver selects with -m "gpu" / -m "not gpu"
    pass


@pytest.fixture(scope="session")
def golden_dir():
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")
