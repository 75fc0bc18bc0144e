import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X (ROCm) GPU and the HIP extension"
    )


@pytest.fixture(autouse=True)
def _deterministic_seed():
    import numpy as np
    import torch

    torch.manual_seed(1)
    np.random.seed(1)
    yield
