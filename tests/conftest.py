import os
import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on a gpurun box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def with_seed():
    """Reference pattern: tests/python/unittest/common.py — with_seed().
    Seed is logged via MXA_TEST_SEED for replay."""
    seed = int(os.environ.get("MXA_TEST_SEED", "0")) or \
        random.randint(0, 2 ** 31)
    np.random.seed(seed)
    torch.manual_seed(seed)
    random.seed(seed)
    yield seed
