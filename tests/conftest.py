import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X box (run via gpurun)")


@pytest.fixture
def synthetic_host(tmp_path):
    from tests.fixtures import SyntheticHost
    h = SyntheticHost(tmp_path)
    yield h
    h.cleanup()
