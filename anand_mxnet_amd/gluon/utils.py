"""Gluon utils (parity with python/mxnet/gluon/utils.py)."""
from __future__ import annotations

import torch

from ..ndarray.ndarray import NDArray, array
from ..context import Context
from .. import ndarray as nd

__all__ = ["split_data", "split_and_load", "clip_global_norm"]


def split_data(data, num_slice, batch_axis=0, even_split=True):
    size = data.shape[batch_axis]
    if even_split and size % num_slice != 0:
        raise ValueError(
            f"data with shape {data.shape} cannot be evenly split into "
            f"{num_slice} slices along axis {batch_axis}.")
    step = size // num_slice
    slices = []
    for i in range(num_slice):
        begin = i * step
        end = (i + 1) * step if i < num_slice - 1 else size
        slices.append(data.slice_axis(batch_axis, begin, end))
    return slices


def split_and_load(data, ctx_list, batch_axis=0, even_split=True):
    if not isinstance(data, NDArray):
        data = array(data, ctx=ctx_list[0])
    if len(ctx_list) == 1:
        return [data.as_in_context(ctx_list[0])]
    slices = split_data(data, len(ctx_list), batch_axis, even_split)
    return [s.as_in_context(ctx) for s, ctx in zip(slices, ctx_list)]


def clip_global_norm(arrays, max_norm, check_isfinite=True):
    """Rescale arrays so that the sum of their 2-norms <= max_norm.

    GPU path: fused multi_sum_sq kernel (one launch over all tensors)."""
    assert len(arrays) > 0
    from ..ops.registry import RAW
    sums = RAW["multi_sum_sq"]([a._t for a in arrays])
    total = torch.stack([s if isinstance(s, torch.Tensor)
                         else torch.as_tensor(s) for s in sums]).sum()
    total_norm = torch.sqrt(total)
    if check_isfinite and not bool(torch.isfinite(total_norm)):
        import warnings
        warnings.warn("nan or inf is detected. Clipping results will be "
                      "undefined.", stacklevel=2)
    scale = max_norm / (float(total_norm) + 1e-8)
    if scale < 1.0:
        with torch.no_grad():
            for arr in arrays:
                arr._t.detach().mul_(scale)
    return float(total_norm)
