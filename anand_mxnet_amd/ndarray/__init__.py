"""The ``nd`` namespace: NDArray + every registered operator as a function
(parity with python/mxnet/ndarray/ — where op functions are code-generated
from the C++ registry at import, here they are injected from ops.registry).
"""
import sys as _sys

from .ndarray import (NDArray, array, zeros, ones, empty, full, arange,
                      from_torch, waitall, concat_nd)
from . import random  # noqa: F401
from . import sparse  # noqa: F401 (RowSparseNDArray / row_sparse_array)
from . import linalg  # noqa: F401 (la_op.cc surface)
from . import utils  # noqa: F401 (save/load helpers)

from .. import ops as _ops


def _inject():
    mod = _sys.modules[__name__]
    for name, fn in _ops.OPS.items():
        if not hasattr(mod, name):
            setattr(mod, name, fn)


_inject()


def save(fname, data):
    """NDArray.Save — byte-compatible .params writer (see ..serializer)."""
    from ..serializer import save_ndarrays
    save_ndarrays(fname, data)


def load(fname):
    from ..serializer import load_ndarrays
    return load_ndarrays(fname)


def moveaxis(x, source, destination):
    import torch
    return NDArray(torch.movedim(x._t, source, destination))


def stack(*args, axis=0):
    return _ops.OPS["stack"](*args, axis=axis)


def concat(*args, dim=1, axis=None):
    if axis is not None:
        dim = axis
    return _ops.OPS["concat"](*args, dim=dim)


class _ContribNS:
    """mx.nd.contrib.* namespace (reference: ndarray/contrib.py —
    contrib ops are registered with a `_contrib_` prefix and re-exposed
    bare under nd.contrib)."""

    def __getattr__(self, item):
        fn = globals().get(f"_contrib_{item}")
        if fn is None:
            import sys
            mod = sys.modules[__name__]
            fn = getattr(mod, f"_contrib_{item}", None)
        if fn is None:
            raise AttributeError(f"nd.contrib has no op {item!r}")
        return fn


contrib = _ContribNS()
