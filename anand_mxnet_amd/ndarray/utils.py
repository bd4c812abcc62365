"""mx.nd.utils — save/load + array-creation helpers.

Reference surface: python/mxnet/ndarray/utils.py (save/load/
load_frombuffer delegating to NDArray::Save/Load; zeros/ones/array
dispatching on storage type).
"""
import io as _io

from .ndarray import NDArray, array, zeros, ones  # noqa: F401
from ..serializer import save_ndarrays, load_ndarrays


def save(fname, data):
    """Write NDArrays (list or dict) in the byte-compatible .params
    format (serializer.py; reference ndarray.cc NDArray::Save)."""
    save_ndarrays(fname, data)


def load(fname):
    return load_ndarrays(fname)


def load_frombuffer(buf):
    """Load NDArrays from an in-memory .params byte buffer."""
    import tempfile
    import os
    with tempfile.NamedTemporaryFile(delete=False) as f:
        f.write(buf)
        path = f.name
    try:
        return load_ndarrays(path)
    finally:
        os.unlink(path)
