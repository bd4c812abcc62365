"""Gluon Parameter / ParameterDict (parity with python/mxnet/gluon/parameter.py:
deferred init, grad_req, per-ctx data/grad, initialize, _reduce, cast)."""
from __future__ import annotations

from collections import OrderedDict

import numpy as np
import torch

from ..base import MXNetError, dtype_from_any
from ..context import Context, cpu, current_context
from ..ndarray.ndarray import NDArray
from .. import initializer as init_mod
from .. import autograd as _ag

__all__ = ["Parameter", "Constant", "ParameterDict", "DeferredInitializationError"]


class DeferredInitializationError(MXNetError):
    pass


class Parameter:
    def __init__(self, name, grad_req="write", shape=None, dtype=np.float32,
                 lr_mult=1.0, wd_mult=1.0, init=None, allow_deferred_init=False,
                 differentiable=True, stype="default", grad_stype="default"):
        self.name = name
        self._grad_req = grad_req if differentiable else "null"
        if isinstance(shape, int):
            shape = (shape,)
        self._shape = tuple(shape) if shape is not None else None
        self._dtype = dtype_from_any(dtype)
        self.lr_mult = lr_mult
        self.wd_mult = wd_mult
        self.init = init
        self.allow_deferred_init = allow_deferred_init
        self._data = None        # OrderedDict ctx -> NDArray
        self._grad = None
        self._deferred_init = ()
        self._stype = stype

    # ------------- properties -------------
    @property
    def shape(self):
        return self._shape

    @shape.setter
    def shape(self, new_shape):
        if self._shape is None:
            self._shape = tuple(new_shape)
            return
        assert len(self._shape) == len(new_shape) and \
            all(j in (0, i) or i == j for i, j in zip(new_shape, self._shape)), \
            f"Expected shape {new_shape} is incompatible with given shape {self._shape}"
        self._shape = tuple(new_shape)

    @property
    def dtype(self):
        from ..base import torch_dtype_to_np
        return torch_dtype_to_np(self._dtype)

    @property
    def grad_req(self):
        return self._grad_req

    @grad_req.setter
    def grad_req(self, req):
        assert req in ("write", "add", "null")
        self._grad_req = req
        if req == "null":
            self._grad = None
            if self._data is not None:
                for arr in self._data.values():
                    arr._t = arr._t.detach()
        elif self._data is not None and self._grad is None:
            self._init_grad()

    # ------------- init -------------
    def initialize(self, init=None, ctx=None, default_init=None,
                   force_reinit=False):
        default_init = default_init or init_mod.Uniform()
        if self._data is not None and not force_reinit:
            return
        if ctx is None:
            ctx = [current_context()]
        if isinstance(ctx, Context):
            ctx = [ctx]
        if self._shape is None or np.prod(self._shape) <= 0:
            if self.allow_deferred_init:
                self._deferred_init = (init, ctx, default_init, None)
                return
            raise MXNetError(f"Cannot initialize Parameter '{self.name}' "
                             "because it has invalid shape: "
                             f"{self._shape}.")
        self._init_impl(init, ctx, default_init, None)

    def _finish_deferred_init(self):
        if not self._deferred_init:
            return
        init, ctx, default_init, data = self._deferred_init
        self._deferred_init = ()
        if self._shape is None or np.prod(self._shape) <= 0:
            raise DeferredInitializationError(
                f"Parameter '{self.name}' has not been initialized yet "
                "(unknown shape)")
        self._init_impl(init, ctx, default_init, data)

    def _init_impl(self, initializer, ctx_list, default_init, data):
        self._data = OrderedDict()
        with _ag.pause():
            for ctx in ctx_list:
                if data is None:
                    arr = NDArray(torch.empty(self._shape, dtype=self._dtype,
                                              device=ctx.torch_device()))
                    ini = initializer if initializer is not None else \
                        (self.init if self.init is not None else default_init)
                    ini_obj = init_mod.create(ini) if isinstance(ini, str) else ini
                    ini_obj.init(self.name, arr)
                else:
                    arr = data.as_in_context(ctx).copy()
                self._data[ctx] = arr
        if self._grad_req != "null":
            self._init_grad()

    def _init_grad(self):
        self._grad = OrderedDict()
        for ctx, arr in self._data.items():
            arr.attach_grad(self._grad_req)
            self._grad[ctx] = arr.grad

    def _load_init(self, data, ctx=None, cast_dtype=False, dtype_source="current"):
        if isinstance(data, NDArray):
            pass
        else:
            data = NDArray(data)
        if cast_dtype and dtype_source == "current":
            data = data.astype(self.dtype)
        elif cast_dtype and dtype_source == "saved":
            self._dtype = data._t.dtype
        if self._shape is not None and any(s != 0 for s in self._shape):
            if tuple(self._shape) != tuple(data.shape) and \
                    all(s != 0 for s in self._shape):
                raise MXNetError(
                    f"Failed loading Parameter '{self.name}' from saved "
                    f"params: shape incompatible expected {self._shape} vs "
                    f"saved {data.shape}")
        self._shape = tuple(data.shape)
        if self._data is None:
            if self._deferred_init:
                ctx = self._deferred_init[1]
            elif ctx is None:
                ctx = [cpu()]
            if isinstance(ctx, Context):
                ctx = [ctx]
            self._init_impl(None, ctx, None, data.astype(self.dtype)
                            if not cast_dtype else data)
        else:
            with torch.no_grad():
                for arr in self._data.values():
                    arr._t.detach().copy_(data._t.to(arr._t.device,
                                                     arr._t.dtype))
        self._deferred_init = ()

    # ------------- access -------------
    def _check_initialized(self, ctx=None):
        if self._data is None:
            if self._deferred_init:
                raise DeferredInitializationError(
                    f"Parameter '{self.name}' has not been initialized yet "
                    "because initialization was deferred. Actual "
                    "initialization happens during the first forward pass.")
            raise MXNetError(
                f"Parameter '{self.name}' has not been initialized. You "
                "should initialize parameters with Block.initialize().")

    def data(self, ctx=None) -> NDArray:
        self._check_initialized()
        if ctx is None:
            if len(self._data) == 1:
                return next(iter(self._data.values()))
            ctx = current_context()
        if isinstance(ctx, Context) and ctx in self._data:
            return self._data[ctx]
        raise MXNetError(f"Parameter '{self.name}' was not initialized on "
                         f"context {ctx}")

    def list_data(self):
        self._check_initialized()
        return list(self._data.values())

    def grad(self, ctx=None) -> NDArray:
        self._check_initialized()
        if self._grad is None:
            raise MXNetError(f"Cannot get gradient for Parameter "
                             f"'{self.name}' because grad_req='null'")
        if ctx is None:
            if len(self._data) == 1:
                return next(iter(self._data.values())).grad
            ctx = current_context()
        return self._data[ctx].grad

    def list_grad(self):
        self._check_initialized()
        return [a.grad for a in self._data.values()]

    def list_ctx(self):
        if self._data is None and self._deferred_init:
            return self._deferred_init[1]
        self._check_initialized()
        return list(self._data.keys())

    def zero_grad(self):
        if self._grad is None:
            return
        with torch.no_grad():
            for arr in self._data.values():
                if arr._t.grad is not None:
                    arr._t.grad.zero_()

    def reset_ctx(self, ctx):
        if isinstance(ctx, Context):
            ctx = [ctx]
        if self._data is not None:
            data = self._reduce()
            init_list = list(self._data.keys())
            self._init_impl(None, ctx, None, data)
        elif self._deferred_init:
            i, _, d, dd = self._deferred_init
            self._deferred_init = (i, ctx, d, dd)

    def _reduce(self) -> NDArray:
        """Average over contexts onto cpu (for save_parameters)."""
        self._check_initialized()
        vals = list(self._data.values())
        acc = vals[0]._t.detach().cpu().clone()
        for v in vals[1:]:
            acc += v._t.detach().cpu()
        if len(vals) > 1:
            acc /= len(vals)
        return NDArray(acc)

    def set_data(self, data):
        self.shape = data.shape
        if self._data is None:
            assert self._deferred_init, \
                f"Parameter '{self.name}' has not been initialized"
            self._deferred_init = self._deferred_init[:3] + (data,)
            return
        with torch.no_grad():
            for arr in self._data.values():
                arr._t.detach().copy_(data._t.to(arr._t.device, arr._t.dtype))

    def cast(self, dtype):
        self._dtype = dtype_from_any(dtype)
        if self._data is None:
            return
        with _ag.pause():
            new_data = OrderedDict()
            for ctx, arr in self._data.items():
                new_data[ctx] = arr.astype(self._dtype)
            self._data = new_data
            if self._grad_req != "null":
                self._init_grad()

    def var(self):
        from .. import symbol
        return symbol.var(self.name, shape=self._shape)

    def __repr__(self):
        return f"Parameter {self.name} (shape={self._shape}, dtype={self.dtype})"


class Constant(Parameter):
    def __init__(self, name, value):
        if not isinstance(value, NDArray):
            value = NDArray(torch.as_tensor(np.asarray(value, dtype=np.float32)))
        self.value = value

        class _CInit(init_mod.Initializer):
            def _init_weight(self2, _, arr):
                with torch.no_grad():
                    arr._t.copy_(value._t.to(arr._t.device, arr._t.dtype))

        super().__init__(name, grad_req="null", shape=value.shape,
                         dtype=value._t.dtype, init=_CInit())


class ParameterDict:
    def __init__(self, prefix="", shared=None):
        self._prefix = prefix
        self._params = OrderedDict()
        self._shared = shared

    @property
    def prefix(self):
        return self._prefix

    def items(self):
        return self._params.items()

    def keys(self):
        return self._params.keys()

    def values(self):
        return self._params.values()

    def __iter__(self):
        return iter(self._params)

    def __getitem__(self, key):
        return self._params[key]

    def __contains__(self, key):
        return key in self._params

    def __len__(self):
        return len(self._params)

    def get(self, name, **kwargs):
        name = self._prefix + name
        param = self._get_impl(name)
        if param is None:
            param = Parameter(name, **kwargs)
            self._params[name] = param
        else:
            for k, v in kwargs.items():
                if hasattr(param, k) and getattr(param, k) is not None:
                    existing = getattr(param, k)
                    if k == "shape" and v is not None and existing is not None:
                        param.shape = v
                    continue
                setattr(param, k, v)
        return param

    def get_constant(self, name, value=None):
        name = self._prefix + name
        param = self._get_impl(name)
        if param is None:
            if value is None:
                raise MXNetError(f"No constant named '{name}'")
            param = Constant(name, value)
            self._params[name] = param
        return param

    def _get_impl(self, name):
        if name in self._params:
            return self._params[name]
        if self._shared is not None and name in self._shared._params:
            self._params[name] = self._shared._params[name]
            return self._params[name]
        return None

    def update(self, other):
        for k, v in other.items():
            if k in self._params and self._params[k] is not v:
                raise MXNetError(f"Cannot update self with other because "
                                 f"they have different Parameters with the "
                                 f"same name '{k}'")
            self._params[k] = v

    def initialize(self, init=None, ctx=None, verbose=False,
                   force_reinit=False):
        for _, v in self.items():
            v.initialize(None, ctx, init or init_mod.Uniform(),
                         force_reinit=force_reinit)

    def zero_grad(self):
        for v in self.values():
            v.zero_grad()

    def reset_ctx(self, ctx):
        for v in self.values():
            v.reset_ctx(ctx)

    def setattr(self, name, value):
        for v in self.values():
            setattr(v, name, value)

    def save(self, filename, strip_prefix=""):
        from ..serializer import save_ndarrays
        arg_dict = {}
        for param in self.values():
            weight = param._reduce()
            if not param.name.startswith(strip_prefix):
                raise MXNetError(f"Prefix '{strip_prefix}' is to be stripped "
                                 f"but Parameter's name '{param.name}' does "
                                 "not start with it")
            arg_dict[param.name[len(strip_prefix):]] = weight
        save_ndarrays(filename, arg_dict)

    def load(self, filename, ctx=None, allow_missing=False,
             ignore_extra=False, restore_prefix="", cast_dtype=False,
             dtype_source="current"):
        from ..serializer import load_ndarrays
        loaded = load_ndarrays(filename)
        if isinstance(loaded, list):
            raise MXNetError("params file has no names")
        arg_dict = {restore_prefix + k.split(":", 1)[-1] if ":" in k
                    else restore_prefix + k: v for k, v in loaded.items()}
        if not allow_missing:
            for name in self.keys():
                if name not in arg_dict:
                    raise MXNetError(f"Parameter '{name}' is missing in file "
                                     f"'{filename}'")
        for name in arg_dict:
            if name not in self._params:
                if not ignore_extra:
                    raise MXNetError(f"Parameter '{name}' loaded from file "
                                     f"'{filename}' is not present in this "
                                     "ParameterDict")
                continue
            self[name]._load_init(arg_dict[name], ctx, cast_dtype=cast_dtype,
                                  dtype_source=dtype_source)

    def __repr__(self):
        s = "\n".join(repr(v) for v in self.values())
        return f"ParameterDict '{self._prefix}' (\n{s}\n)"
