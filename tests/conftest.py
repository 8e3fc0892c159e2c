import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )
    config.addinivalue_line("markers", "slow: long-running tests")


@pytest.fixture(scope="session")
def oracle_path():
    path = os.path.join(os.path.dirname(__file__), "..", "traces", "mi355x_throughputs.json")
    assert os.path.exists(path)
    return path


@pytest.fixture(scope="session")
def throughputs(oracle_path):
    from shockwave_amd.core.throughputs import read_throughputs

    return read_throughputs(oracle_path)
