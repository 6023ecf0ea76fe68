import random

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line("markers", "distributed: multi-process test (gloo on CPU)")
    config.addinivalue_line("markers", "slow: long-running CPU test")


@pytest.fixture(autouse=True)
def fixed_seed():
    torch.manual_seed(123)
    random.seed(123)
    yield
