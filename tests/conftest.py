import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X (run via gpurun / driver)")


@pytest.fixture(scope="session")
def golden_dir():
    d = os.path.join(REPO, "tests", "golden")
    assert os.path.isdir(d), "run python -m oracle.golden_gen first"
    return d
