import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")


@pytest.fixture
def rng():
    import numpy as np

    return np.random.default_rng(12345)
