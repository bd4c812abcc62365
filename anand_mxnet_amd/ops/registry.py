"""Operator registry and NDArray<->tensor wrapping.

Plays the role of the reference's NNVM op registry + the per-op Python
codegen (python/mxnet/ndarray/register.py): every registered op becomes a
function in the ``anand_mxnet_amd.nd`` namespace.  Implementations operate on
torch tensors; GPU hot paths dispatch to hand-written HIP/CDNA4 kernels via
``gpu_impl`` (anand_mxnet_amd.ops.hip), CPU paths are torch compositions
(the CPU implementation is the numerics oracle, as in the reference's
check_consistency test pattern).
"""
from __future__ import annotations

import functools
import torch

from .. import autograd as _ag
from ..base import MXNetError

# name -> nd-level function (wrapped)
OPS = {}
# name -> raw tensor-level function
RAW = {}
# name -> GPU (HIP-kernel) tensor-level function, populated by ops.hip
GPU_IMPL = {}


def _unwrap(x):
    from ..ndarray.ndarray import NDArray
    if isinstance(x, NDArray):
        return x._t
    if isinstance(x, (list, tuple)):
        return type(x)(_unwrap(v) for v in x)
    return x


def _wrap_out(out):
    from ..ndarray.ndarray import NDArray
    if isinstance(out, torch.Tensor):
        return NDArray(out)
    if isinstance(out, (list, tuple)):
        return [_wrap_out(o) for o in out]
    return out


def register(name: str):
    """Register a tensor-level op implementation under `name`."""
    def deco(fn):
        RAW[name] = fn

        @functools.wraps(fn)
        def nd_fn(*args, **kwargs):
            targs = [_unwrap(a) for a in args]
            tkw = {k: _unwrap(v) for k, v in kwargs.items()}
            with torch.set_grad_enabled(_ag.is_recording()):
                out = fn(*targs, **tkw)
            return _wrap_out(out)

        OPS[name] = nd_fn
        return fn
    return deco


def register_gpu(name: str):
    """Register the HIP-kernel implementation for op `name` (GPU path)."""
    def deco(fn):
        GPU_IMPL[name] = fn
        return fn
    return deco


def gpu_impl(name: str):
    """Fetch the HIP implementation for a hot op; fail LOUDLY if absent.

    On a GPU box the hand-written CDNA4 kernels are the compute path; a
    silent eager fallback would invalidate benchmarks, so it is an error.
    """
    fn = GPU_IMPL.get(name)
    if fn is None:
        raise MXNetError(
            f"op '{name}' reached the GPU dispatch but no HIP kernel is "
            "registered — the native extension failed to load. Build it with "
            "`python setup.py build_ext --inplace`.")
    return fn


def has_gpu_impl(name: str) -> bool:
    return name in GPU_IMPL


def get(name: str):
    return OPS[name]
