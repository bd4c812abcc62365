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


def fold_batchnorm(block, _count=None):
    """Inference-time conv+BN(+relu) folding.

    Walks the Block tree; wherever a ``Conv2D`` is DIRECTLY followed by a
    ``BatchNorm`` inside the same container (the composition-is-sequential
    guarantee), absorbs the BN's affine transform into the conv weights
    (w'[k] = w[k] * gamma_k / sqrt(var_k + eps), b' = beta + (b - mean) *
    scale) and, when the BN carried ``fuse_relu``, moves the relu into the
    conv kernel's epilogue.  The BN layer becomes an identity.  Junction
    BNs (called with a residual addend, e.g. ResNet's bn_join) never form
    a (Conv2D, BatchNorm) sibling pair here, so they are untouched.

    Inference-only: folded convs with fused relu run the no-autograd
    kernel path; mutates conv weights in place (re-load params to train
    again).  Reference parity: the conv+bn+relu subgraph fusion the
    reference reserves for its subgraph framework.

    Returns the number of BN layers folded.
    """
    from .nn.conv_layers import Conv2D
    from .nn.basic_layers import BatchNorm
    count = [0] if _count is None else _count
    children = list(block._children.values())
    for prev, cur in zip(children, children[1:]):
        if not (isinstance(prev, Conv2D) and isinstance(cur, BatchNorm)):
            continue
        if getattr(cur, "_folded", False):
            continue
        ax = cur._axis
        want_ax = 1 if prev._layout == "NCHW" else 3
        if ax != want_ax:
            continue
        eps = cur._kwargs["eps"]
        gamma = cur.gamma.data()._t.float()
        beta = cur.beta.data()._t.float()
        rmean = cur.running_mean.data()._t.float()
        rvar = cur.running_var.data()._t.float()
        scale = gamma / torch.sqrt(rvar + eps)
        w = prev.weight.data()._t
        wshape = [1] * w.dim()
        wshape[0] = -1
        w_f = (w.float() * scale.reshape(wshape)).to(w.dtype)
        prev.weight.set_data(NDArray(w_f))
        if prev.bias is not None:
            b0 = prev.bias.data()._t.float()
            bias_f = beta + (b0 - rmean) * scale
            prev.bias.set_data(NDArray(bias_f.to(prev.bias.data()._t.dtype)))
        else:
            prev._folded_bias = (beta - rmean * scale)
        if cur._kwargs.get("fuse_relu"):
            prev._fold_relu = True
        cur._folded = True
        count[0] += 1
    for child in children:
        fold_batchnorm(child, count)
    return count[0]
