"""Gluon Block / HybridBlock / SymbolBlock (parity with
python/mxnet/gluon/block.py).

MI355X-native CachedOp: the reference's ``hybridize`` lowers the block to a
graph executed by CachedOp (src/imperative/cached_op.cc) with static_alloc +
op bulking to cut per-op engine overhead.  Here ``hybridize`` with
``static_alloc=True`` captures the forward (and backward, via
torch.cuda.make_graphed_callables) into a **hipGraph** keyed on input shape —
one graph replay per step instead of per-op launches, the CDNA4-idiomatic
equivalent of bulked engine segments.  Without static_alloc, hybridize is a
no-op (eager, stream-ordered)."""
from __future__ import annotations

import copy
import re
import threading
from collections import OrderedDict

import numpy as np
import torch

from ..base import MXNetError, dtype_from_any
from ..context import Context, cpu, current_context
from ..ndarray.ndarray import NDArray
from .parameter import Parameter, ParameterDict, DeferredInitializationError
from .. import ndarray as nd_mod
from .. import autograd as _ag

__all__ = ["Block", "HybridBlock", "SymbolBlock"]

_naming = threading.local()


class _BlockScope:
    """Name manager for parameter prefixes."""

    _current = threading.local()

    def __init__(self, block):
        self._block = block
        self._counter = {}
        self._old_scope = None

    @staticmethod
    def create(prefix, params, hint):
        current = getattr(_BlockScope._current, "value", None)
        if current is None:
            if prefix is None:
                if not hasattr(_naming, "counter"):
                    _naming.counter = {}
                count = _naming.counter.get(hint, 0)
                _naming.counter[hint] = count + 1
                prefix = f"{hint}{count}_"
            if params is None:
                params = ParameterDict(prefix)
            else:
                params = ParameterDict(params.prefix, params)
            return prefix, params
        if prefix is None:
            count = current._counter.get(hint, 0)
            current._counter[hint] = count + 1
            prefix = f"{hint}{count}_"
        if params is None:
            parent = current._block.params
            params = ParameterDict(parent.prefix + prefix, parent._shared)
        else:
            params = ParameterDict(params.prefix, params)
        return current._block.prefix + prefix, params

    def __enter__(self):
        if self._block._empty_prefix:
            return self
        self._old_scope = getattr(_BlockScope._current, "value", None)
        _BlockScope._current.value = self
        return self

    def __exit__(self, *a):
        if self._block._empty_prefix:
            return
        _BlockScope._current.value = self._old_scope


class Block:
    def __init__(self, prefix=None, params=None):
        self._empty_prefix = prefix == ""
        self._prefix, self._params = _BlockScope.create(
            prefix, params, self._alias())
        self._name = self._prefix[:-1] if self._prefix.endswith("_") \
            else self._prefix
        self._scope = _BlockScope(self)
        self._children = OrderedDict()
        self._reg_params = {}
        self._forward_hooks = OrderedDict()
        self._forward_pre_hooks = OrderedDict()

    def _alias(self):
        return self.__class__.__name__.lower()

    @property
    def prefix(self):
        return self._prefix

    @property
    def name(self):
        return self._name

    def name_scope(self):
        return self._scope

    @property
    def params(self):
        return self._params

    def collect_params(self, select=None) -> ParameterDict:
        ret = ParameterDict(self._params.prefix)
        if not select:
            ret.update(self.params)
        else:
            pattern = re.compile(select)
            ret.update({name: value for name, value in self.params.items()
                        if pattern.match(name)})
        for child in self._children.values():
            ret.update(child.collect_params(select=select))
        return ret

    def __setattr__(self, name, value):
        if isinstance(value, Block):
            existing = self.__dict__.get("_children")
            if existing is not None:
                existing[name] = value
        elif isinstance(value, Parameter):
            if "_reg_params" in self.__dict__:
                self._reg_params[name] = value
        super().__setattr__(name, value)

    def register_child(self, block, name=None):
        if name is None:
            name = str(len(self._children))
        self._children[name] = block

    def register_forward_hook(self, hook):
        self._forward_hooks[len(self._forward_hooks)] = hook

    def register_forward_pre_hook(self, hook):
        self._forward_pre_hooks[len(self._forward_pre_hooks)] = hook

    def apply(self, fn):
        for child in self._children.values():
            child.apply(fn)
        fn(self)
        return self

    def initialize(self, init=None, ctx=None, verbose=False,
                   force_reinit=False):
        from .. import initializer as init_mod
        self.collect_params().initialize(init or init_mod.Uniform(), ctx,
                                         verbose, force_reinit)

    def hybridize(self, active=True, **kwargs):
        for child in self._children.values():
            child.hybridize(active, **kwargs)

    def cast(self, dtype):
        for child in self._children.values():
            child.cast(dtype)
        for _, param in self.params.items():
            param.cast(dtype)

    def __call__(self, *args, **kwargs):
        for hook in self._forward_pre_hooks.values():
            hook(self, args)
        out = self.forward(*args, **kwargs)
        for hook in self._forward_hooks.values():
            hook(self, args, out)
        return out

    def forward(self, *args):
        raise NotImplementedError

    def summary(self, *inputs):
        summary_strs = [f"{self.__class__.__name__}:"]
        for name, p in self.collect_params().items():
            summary_strs.append(f"  {name}: {p.shape}")
        print("\n".join(summary_strs))

    # ---------------- save / load ----------------
    def save_parameters(self, filename, deduplicate=False):
        params = self._collect_params_with_prefix()
        from ..serializer import save_ndarrays
        arg_dict = {key: val._reduce() for key, val in params.items()}
        save_ndarrays(filename, arg_dict)

    def load_parameters(self, filename, ctx=None, allow_missing=False,
                        ignore_extra=False, cast_dtype=False,
                        dtype_source="current"):
        from ..serializer import load_ndarrays
        loaded = load_ndarrays(filename)
        params = self._collect_params_with_prefix()
        if isinstance(loaded, list):
            raise MXNetError("params file has no parameter names")
        if not loaded and not params:
            return
        if any(":" in k for k in loaded):  # Module-style arg:/aux: names
            loaded = {k.split(":", 1)[1]: v for k, v in loaded.items()}
        if not allow_missing:
            for name in params.keys():
                if name not in loaded:
                    raise MXNetError(
                        f"Parameter '{name}' is missing in file '{filename}'")
        for name in loaded:
            if name not in params:
                if not ignore_extra:
                    raise MXNetError(
                        f"Parameter '{name}' loaded from file '{filename}' "
                        "is not present in this Block")
                continue
            params[name]._load_init(loaded[name], ctx, cast_dtype=cast_dtype,
                                    dtype_source=dtype_source)

    def _collect_params_with_prefix(self, prefix=""):
        if prefix:
            prefix += "."
        ret = {prefix + key: val for key, val in self._reg_params.items()}
        for name, child in self._children.items():
            ret.update(child._collect_params_with_prefix(prefix + name))
        return ret

    def __repr__(self):
        s = f"{self.__class__.__name__}("
        for key, child in self._children.items():
            s += f"\n  ({key}): {repr(child)}"
        return s + "\n)" if self._children else s + ")"


class HybridBlock(Block):
    def __init__(self, prefix=None, params=None):
        super().__init__(prefix=prefix, params=params)
        self._active = False
        self._cached_graph = None
        self._cached_graph_obj = None
        self._graph_kwargs = {}

    def _eager_forward(self, *args):
        return super(HybridBlock, type(self)).forward(self, *args) \
            if type(self).forward is HybridBlock.forward \
            else type(self).forward(self, *args)

    def __call__(self, *args, **kwargs):
        if self._active and not kwargs:
            from .cached_op import maybe_cached_forward
            out = maybe_cached_forward(self, args)
            if out is not None:
                return out
        return super().__call__(*args, **kwargs)

    def hybridize(self, active=True, static_alloc=False, static_shape=False,
                  **kwargs):
        self._active = active
        self._graph_kwargs = dict(static_alloc=static_alloc,
                                  static_shape=static_shape, **kwargs)
        self._cached_graph = None
        self._cached_graph_obj = None
        super().hybridize(active, static_alloc=static_alloc,
                          static_shape=static_shape, **kwargs)

    def cast(self, dtype):
        self._cached_graph = None
        super().cast(dtype)

    def infer_shape(self, *args):
        """Run deferred shape inference by a dry forward on meta params."""
        # deferred-init params get shapes inside hybrid_forward (layers call
        # _finish_deferred_init after setting shapes from inputs)
        pass

    def _get_graph_inputs(self, *args):
        return args

    def forward(self, x, *args):
        if isinstance(x, NDArray):
            params = {}
            try:
                for name, p in self._reg_params.items():
                    params[name] = p.data(x.context)
            except DeferredInitializationError:
                self._infer_param_shapes(x, *args)
                for name, p in self._reg_params.items():
                    params[name] = p.data(x.context)
            return self.hybrid_forward(nd_mod, x, *args, **params)
        raise ValueError("HybridBlock input must be NDArray")

    def _infer_param_shapes(self, *args):
        """Deferred init: infer parameter shapes from input shapes."""
        self.shape_infer(*args)
        for _, p in self._reg_params.items():
            p._finish_deferred_init()

    def shape_infer(self, *args):
        """Subclasses with deferred params override to set param shapes."""
        raise DeferredInitializationError(
            f"{self.__class__.__name__} has deferred-init parameters but "
            "no shape_infer")

    def hybrid_forward(self, F, x, *args, **kwargs):
        raise NotImplementedError

    def export(self, path, epoch=0):
        """Save symbol json + params (reference HybridBlock.export)."""
        from .. import symbol as sym_mod
        s = sym_mod.trace_block(self)
        s.save(f"{path}-symbol.json")
        params = self._collect_params_with_prefix()
        from ..serializer import save_ndarrays
        # key by Parameter.name — the traced symbol's variable nodes use
        # parameter names, and Module/SymbolBlock loaders bind by them
        # (the dotted child-path keys of save_parameters are a different
        # contract).  Auxiliary states (grad_req null: BN running stats)
        # get the reference's aux: prefix so Module.load_checkpoint
        # classifies them correctly.
        arg_dict = {}
        for v in params.values():
            pfx = "aux:" if v.grad_req == "null" else "arg:"
            arg_dict[pfx + v.name] = v._reduce()
        save_ndarrays(f"{path}-{epoch:04d}.params", arg_dict)
        return f"{path}-symbol.json", f"{path}-{epoch:04d}.params"


class SymbolBlock(HybridBlock):
    """Run a loaded Symbol as a Block (reference gluon.SymbolBlock)."""

    def __init__(self, outputs, inputs, params=None):
        super().__init__(prefix="", params=params)
        self._symbol = outputs
        self._input_names = [i if isinstance(i, str) else i.name
                             for i in (inputs if isinstance(inputs, list)
                                       else [inputs])]

    @staticmethod
    def imports(symbol_file, input_names, param_file=None, ctx=None):
        from .. import symbol as sym_mod
        s = sym_mod.load(symbol_file)
        blk = SymbolBlock(s, [sym_mod.var(n) for n in (
            input_names if isinstance(input_names, list) else [input_names])])
        if param_file is not None:
            from ..serializer import load_ndarrays
            loaded = load_ndarrays(param_file)
            blk._arg_params = {k.split(":", 1)[-1]: v
                               for k, v in loaded.items()}
        else:
            blk._arg_params = {}
        if ctx is not None:
            blk._ctx = ctx
        return blk

    def forward(self, *args):
        inputs = dict(zip(self._input_names, args))
        return self._symbol.eval(inputs, self._arg_params)
