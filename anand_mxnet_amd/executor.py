"""Graph executor (parity role of src/executor/graph_executor.cc +
include/mxnet/executor.h: Symbol.simple_bind/bind -> Executor with
Forward/Backward over engine-scheduled ops).

MI355X-native: no ahead-of-time memory planning pass — torch-ROCm's caching
allocator + stream-ordered eager execution replace PlanMemory/bulking; the
hipGraph step capture (gluon hybridize / bench) is the bulked fast path."""
from __future__ import annotations

import torch

from .base import MXNetError
from .context import Context, cpu
from .ndarray.ndarray import NDArray, zeros
from . import autograd as ag

__all__ = ["Executor"]


class Executor:
    def __init__(self, symbol, ctx=None, grad_req="write", shapes=None,
                 args=None, args_grad=None, aux_states=None):
        self._symbol = symbol
        self._ctx = ctx if isinstance(ctx, Context) else \
            (ctx[0] if isinstance(ctx, (list, tuple)) else (ctx or cpu()))
        self._grad_req = grad_req
        arg_names = symbol.list_arguments()
        aux_names = symbol.list_auxiliary_states()

        if args is None:
            shapes = shapes or {}
            arg_shapes, out_shapes, aux_shapes = symbol.infer_shape(**shapes)
            self.arg_dict = {}
            for name, shp in zip(arg_names, arg_shapes):
                if shp is None:
                    raise MXNetError(f"cannot infer shape for argument "
                                     f"'{name}'; pass it to simple_bind")
                self.arg_dict[name] = zeros(shp, ctx=self._ctx)
            self.aux_dict = {name: zeros(shp, ctx=self._ctx)
                             for name, shp in zip(aux_names, aux_shapes)}
        else:
            if isinstance(args, (list, tuple)):
                self.arg_dict = dict(zip(arg_names, args))
            else:
                self.arg_dict = dict(args)
            if aux_states is None:
                self.aux_dict = {}
            elif isinstance(aux_states, (list, tuple)):
                self.aux_dict = dict(zip(aux_names, aux_states))
            else:
                self.aux_dict = dict(aux_states)

        self._input_names = set(shapes.keys()) if shapes else set()
        if args_grad is not None:
            self._grad_dict = dict(zip(arg_names, args_grad)) \
                if isinstance(args_grad, (list, tuple)) else dict(args_grad)
        else:
            self._grad_dict = None  # lazily exposes the attached .grad
        if grad_req != "null":
            for name, arr in self.arg_dict.items():
                arr.attach_grad(grad_req if isinstance(grad_req, str)
                                else grad_req.get(name, "write"))
        self.outputs = []

    @property
    def arg_arrays(self):
        return [self.arg_dict[n] for n in self._symbol.list_arguments()]

    @property
    def grad_dict(self):
        """name -> gradient array (reference executor.py grad_dict).
        When no explicit args_grad was bound, the autograd-attached
        .grad arrays are exposed (populated after backward())."""
        if self._grad_dict is not None:
            return self._grad_dict
        if self._grad_req == "null":
            return {}
        return {n: self.arg_dict[n].grad
                for n in self._symbol.list_arguments()
                if getattr(self.arg_dict[n], "grad", None) is not None}

    @property
    def grad_arrays(self):
        return [self.arg_dict[n].grad
                for n in self._symbol.list_arguments()]

    @property
    def aux_arrays(self):
        return [self.aux_dict[n]
                for n in self._symbol.list_auxiliary_states()]

    def forward(self, is_train=False, **kwargs):
        with torch.no_grad():
            for name, val in kwargs.items():
                if name in self.arg_dict:
                    self.arg_dict[name]._t.detach().copy_(
                        val._t if isinstance(val, NDArray)
                        else torch.as_tensor(val))
        inputs = dict(self.arg_dict)
        inputs.update(self.aux_dict)
        with ag._Scope(recording=is_train and self._grad_req != "null",
                       training=is_train):
            out = self._symbol.eval(inputs, {})
        self.outputs = out if isinstance(out, list) else [out]
        return self.outputs

    def backward(self, out_grads=None):
        heads = self.outputs
        if out_grads is None:
            ag.backward(heads)
        else:
            if isinstance(out_grads, NDArray):
                out_grads = [out_grads]
            ag.backward(heads, out_grads)

    def copy_params_from(self, arg_params, aux_params=None):
        with torch.no_grad():
            for k, v in arg_params.items():
                if k in self.arg_dict:
                    self.arg_dict[k]._t.detach().copy_(
                        v._t.to(self.arg_dict[k]._t.device))
            for k, v in (aux_params or {}).items():
                if k in self.aux_dict:
                    self.aux_dict[k]._t.detach().copy_(
                        v._t.to(self.aux_dict[k]._t.device))
