"""Custom operators (parity with python/mxnet/operator.py — CustomOp /
CustomOpProp / register; reference src/operator/custom/custom.cc runs these
Python callbacks on a worker pool; here they run inline as
torch.autograd.Functions, stream-ordered like every other op)."""
from __future__ import annotations

import functools

import torch

from .base import MXNetError
from .ndarray.ndarray import NDArray
from . import autograd as _ag
from .ops import registry as _reg

__all__ = ["CustomOp", "CustomOpProp", "register", "get_all_registered"]

_CUSTOM_OPS = {}


class CustomOp:
    """Base class for user ops: implement forward()/backward() with
    assign()-style outputs (reference API)."""

    def forward(self, is_train, req, in_data, out_data, aux):
        raise NotImplementedError

    def backward(self, req, out_grad, in_data, out_data, in_grad, aux):
        raise NotImplementedError

    def assign(self, dst, req, src):
        if req in ("write", "inplace", None):
            dst._t.detach().copy_(src._t if isinstance(src, NDArray)
                                  else torch.as_tensor(src))
        elif req == "add":
            dst._t.detach().add_(src._t if isinstance(src, NDArray)
                                 else torch.as_tensor(src))
        elif req == "null":
            pass
        else:
            raise MXNetError(f"unknown req {req}")


class CustomOpProp:
    def __init__(self, need_top_grad=True):
        self.need_top_grad_ = need_top_grad

    def list_arguments(self):
        return ["data"]

    def list_outputs(self):
        return ["output"]

    def infer_shape(self, in_shape):
        return in_shape, [in_shape[0]], []

    def infer_type(self, in_type):
        return in_type, [in_type[0]] * len(self.list_outputs()), []

    def create_operator(self, ctx, shapes, dtypes):
        raise NotImplementedError


def register(reg_name):
    """Register a CustomOpProp; the op becomes callable as
    ``nd.Custom(*inputs, op_type=reg_name)``."""

    def deco(prop_cls):
        _CUSTOM_OPS[reg_name] = prop_cls
        return prop_cls

    return deco


def get_all_registered():
    return dict(_CUSTOM_OPS)


@_reg.register("Custom")
def custom(*tensors, op_type=None, **kwargs):
    if op_type not in _CUSTOM_OPS:
        raise MXNetError(f"custom op '{op_type}' is not registered")
    prop = _CUSTOM_OPS[op_type]()
    shapes = [list(t.shape) for t in tensors]
    dtypes = [t.dtype for t in tensors]
    in_shapes, out_shapes, _ = prop.infer_shape(shapes)
    op = prop.create_operator(None, in_shapes, dtypes)
    n_out = len(prop.list_outputs())

    class _F(torch.autograd.Function):
        @staticmethod
        def forward(ctx, *ts):
            ins = [NDArray(t) for t in ts]
            outs = [NDArray(torch.empty(tuple(s), dtype=ts[0].dtype,
                                        device=ts[0].device))
                    for s in out_shapes]
            with _ag.pause():
                op.forward(_ag.is_training(), ["write"] * n_out, ins, outs,
                           [])
            ctx.save_for_backward(*ts, *[o._t for o in outs])
            ctx.n_in = len(ts)
            return tuple(o._t for o in outs) if n_out > 1 else outs[0]._t

        @staticmethod
        def backward(ctx, *gts):
            saved = ctx.saved_tensors
            ins = [NDArray(t) for t in saved[:ctx.n_in]]
            outs = [NDArray(t) for t in saved[ctx.n_in:]]
            ograds = [NDArray(g.contiguous()) for g in gts]
            igrads = [NDArray(torch.zeros_like(t._t)) for t in ins]
            with _ag.pause():
                op.backward(["write"] * len(ins), ograds, ins, outs,
                            igrads, [])
            return tuple(g._t for g in igrads)

    with torch.set_grad_enabled(_ag.is_recording()):
        out = _F.apply(*tensors)
    return out


def _reinstall_nd():
    import sys
    nd = sys.modules.get("anand_mxnet_amd.ndarray")
    if nd is not None:
        setattr(nd, "Custom", _reg.OPS["Custom"])


_reinstall_nd()
