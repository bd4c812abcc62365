"""nd.random namespace (parity with python/mxnet/ndarray/random.py)."""
from __future__ import annotations

import torch

from .ndarray import NDArray, _wrap, _ctx_dev
from ..base import dtype_from_any


def _shape(shape):
    if shape is None:
        return (1,)
    if isinstance(shape, int):
        return (shape,)
    return tuple(shape)


def uniform(low=0.0, high=1.0, shape=None, dtype="float32", ctx=None, out=None):
    t = torch.empty(_shape(shape), dtype=dtype_from_any(dtype),
                    device=_ctx_dev(ctx))
    if t.dtype in (torch.bfloat16, torch.float16):
        t32 = torch.empty_like(t, dtype=torch.float32).uniform_(low, high)
        t.copy_(t32)
    else:
        t.uniform_(low, high)
    if out is not None:
        out._t.detach().copy_(t)
        return out
    return NDArray(t)


def normal(loc=0.0, scale=1.0, shape=None, dtype="float32", ctx=None, out=None):
    t = torch.empty(_shape(shape), dtype=dtype_from_any(dtype),
                    device=_ctx_dev(ctx))
    if t.dtype in (torch.bfloat16, torch.float16):
        t32 = torch.empty_like(t, dtype=torch.float32).normal_(loc, scale)
        t.copy_(t32)
    else:
        t.normal_(loc, scale)
    if out is not None:
        out._t.detach().copy_(t)
        return out
    return NDArray(t)


def randint(low, high, shape=None, dtype="int32", ctx=None):
    return NDArray(torch.randint(low, high, _shape(shape),
                                 dtype=dtype_from_any(dtype),
                                 device=_ctx_dev(ctx)))


def randn(*shape, loc=0.0, scale=1.0, dtype="float32", ctx=None):
    return normal(loc, scale, shape=shape, dtype=dtype, ctx=ctx)


def exponential(scale=1.0, shape=None, dtype="float32", ctx=None):
    t = torch.empty(_shape(shape), dtype=dtype_from_any(dtype),
                    device=_ctx_dev(ctx)).exponential_(1.0 / scale)
    return NDArray(t)


def gamma(alpha=1.0, beta=1.0, shape=None, dtype="float32", ctx=None):
    d = torch.distributions.Gamma(float(alpha), 1.0 / float(beta))
    return NDArray(d.sample(_shape(shape)).to(dtype_from_any(dtype)))


def poisson(lam=1.0, shape=None, dtype="float32", ctx=None):
    t = torch.poisson(torch.full(_shape(shape), float(lam),
                                 device=_ctx_dev(ctx)))
    return NDArray(t.to(dtype_from_any(dtype)))


def multinomial(data, shape=1, get_prob=False, dtype="int32"):
    from ..ops.registry import OPS
    return OPS["sample_multinomial"](data, shape=shape, get_prob=get_prob,
                                     dtype=dtype)


def seed(seed_state, ctx="all"):
    torch.manual_seed(seed_state)


def shuffle(x):
    return _wrap(x._t[torch.randperm(x._t.shape[0], device=x._t.device)])
