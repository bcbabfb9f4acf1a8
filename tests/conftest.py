import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run with -m gpu on a GPU box)")


@pytest.fixture(autouse=True)
def _seed():
    random.seed(0)
    np.random.seed(0)
    torch.manual_seed(0)
