import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: test requires an MI355X GPU (run via gpurun)')


@pytest.fixture(autouse=True)
def _seed_everything():
    """Seeded RNG per test for reproducibility (reference conftest.py:61-127)."""
    random.seed(0)
    np.random.seed(0)
    torch.manual_seed(0)
    yield


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason='no GPU in this container')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip_gpu)
