import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")
    config.addinivalue_line("markers", "slow: long-running CPU test")


@pytest.fixture(autouse=True)
def _seed():
    random.seed(0)
    np.random.seed(0)
    torch.manual_seed(0)
