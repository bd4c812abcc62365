"""Base utilities: errors, dtype mapping, native-extension loading.

MI355X-native re-implementation of the role played by the reference's
``python/mxnet/base.py`` (ctypes ``_LIB``/``check_call``).  Here the native
core is a torch C++/HIP extension (``anand_mxnet_amd._C``) loaded lazily; the
Python<->native boundary is pybind11/torch instead of a flat ctypes C ABI.
"""
from __future__ import annotations

import os
import numpy as np
import torch

__all__ = ["MXNetError", "np_dtype_to_torch", "torch_dtype_to_np", "dtype_from_any",
           "lib", "have_native", "require_native"]


class MXNetError(RuntimeError):
    """Framework error type (parity with mxnet.base.MXNetError)."""


_NP2TORCH = {
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.float16): torch.float16,
    np.dtype(np.int8): torch.int8,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.bool_): torch.bool,
}
_TORCH2NP = {v: k for k, v in _NP2TORCH.items()}
# bfloat16 has no numpy dtype; handled by the string 'bfloat16'.
_STR2TORCH = {
    "float32": torch.float32, "float64": torch.float64, "float16": torch.float16,
    "bfloat16": torch.bfloat16, "int8": torch.int8, "uint8": torch.uint8,
    "int32": torch.int32, "int64": torch.int64, "bool": torch.bool,
}


def np_dtype_to_torch(dt) -> torch.dtype:
    return _NP2TORCH[np.dtype(dt)]


def torch_dtype_to_np(dt: torch.dtype):
    if dt == torch.bfloat16:
        return "bfloat16"
    return _TORCH2NP[dt]


def dtype_from_any(dt) -> torch.dtype:
    """Accept numpy dtype / python type / string / torch dtype -> torch dtype."""
    if dt is None:
        return torch.float32
    if isinstance(dt, torch.dtype):
        return dt
    if isinstance(dt, str):
        try:
            return _STR2TORCH[dt]
        except KeyError:
            raise MXNetError(f"unknown dtype string {dt!r}")
    return _NP2TORCH[np.dtype(dt)]


_lib = None
_lib_err = None


def lib():
    """Load and return the native extension module, or None if unavailable.

    The extension is built in-tree (anand_mxnet_amd/_C*.so) by
    ``python setup.py build_ext --inplace`` / ``__graft_entry__.build()``.
    """
    global _lib, _lib_err
    if _lib is not None or _lib_err is not None:
        return _lib
    try:
        from anand_mxnet_amd import _C  # type: ignore
        _lib = _C
    except ImportError as e:  # pragma: no cover - exercised only without build
        _lib_err = e
        _lib = None
    return _lib


def have_native() -> bool:
    return lib() is not None


def require_native():
    """GPU compute paths MUST run our HIP kernels: fail loudly if missing."""
    m = lib()
    if m is None:
        raise MXNetError(
            "anand_mxnet_amd native extension (_C) is not built but a GPU "
            "compute path needs it. Run `python setup.py build_ext --inplace` "
            f"(import error: {_lib_err})")
    return m


def env_flag(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v not in ("0", "false", "False", "")


def env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v is not None else default
