import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")
    config.addinivalue_line(
        "markers", "slow: long-running test")


@pytest.fixture
def seeded_rng():
    random.seed(0)
    np.random.seed(0)
    torch.manual_seed(0)
    return np.random.RandomState(0)


@pytest.fixture
def gpu_device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU available")
    return torch.device("cuda", 0)
