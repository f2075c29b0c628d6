import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X / ROCm GPU")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests on CPU-only boxes so a plain `pytest` run is
    clean (no -m filter needed); the driver still runs `-m gpu` on MI355X."""
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="needs ROCm GPU (torch.cuda unavailable)")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def _seed_everything():
    torch.manual_seed(0)
    np.random.seed(0)
    random.seed(0)
    yield


@pytest.fixture
def device():
    if torch.cuda.is_available():
        return torch.device("cuda:0")
    return torch.device("cpu")
