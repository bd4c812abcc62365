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
    """Reference pattern: tests/python/unittest/common.py — with_seed(),
    but with a FIXED default seed: a random per-test seed made the
    numeric-gradient checks flake roughly once per hundred suite runs
    (tolerances are calibrated, the tail isn't).  Set MXA_TEST_SEED=random
    for fuzzing, or to an integer to replay one."""
    env = os.environ.get("MXA_TEST_SEED", "")
    if env == "random":
        seed = random.randint(0, 2 ** 31)
    else:
        seed = int(env) if env else 20240901
    np.random.seed(seed)
    torch.manual_seed(seed)
    random.seed(seed)
    yield seed
