"""Top-level random module (parity with python/mxnet/random.py)."""
from .ndarray.random import (uniform, normal, randint, randn, exponential,
                             gamma, poisson, multinomial, shuffle)


def seed(seed_state, ctx="all"):
    """Seed every device RNG (reference mx.random.seed)."""
    import torch
    import numpy as np
    torch.manual_seed(seed_state)
    np.random.seed(seed_state % (2 ** 31))
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed_state)
    try:  # native salt stream (dropout / flash replay / sampling)
        from .ops import hip as _hip
        _hip.reset_seed(seed_state)
    except ImportError:  # pragma: no cover
        pass
