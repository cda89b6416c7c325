import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X / ROCm GPU")
    config.addinivalue_line("markers", "slow: long-running (full training) test")


@pytest.fixture(autouse=True)
def fix_random():
    random.seed(0)
    np.random.seed(0)
    torch.manual_seed(0)
    yield


@pytest.fixture
def gpu_device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU available")
    return torch.device("cuda:0")
