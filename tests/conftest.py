import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# Small-tensor CPU torch ops thrash with one thread per core on many-core
# boxes (a 2s fit becomes minutes); cap threads for the whole test session.
os.environ.setdefault("OMP_NUM_THREADS", "8")
import torch  # noqa: E402

torch.set_num_threads(min(8, os.cpu_count() or 8))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")


@pytest.fixture
def rng():
    import numpy as np

    return np.random.default_rng(12345)
