"""Module (parity with python/mxnet/module/module.py: bind ->
executors per context, init_params, init_optimizer w/ kvstore,
forward/backward/update, save_checkpoint / Module.load).

MI355X note: data-parallel multi-ctx inside one process is supported by
slicing the batch across per-ctx executors (reference
DataParallelExecutorGroup); the idiomatic MI355X scale-out path is one
process per GPU (gluon.Trainer + RCCL), this single-process path serves the
legacy API."""
from __future__ import annotations

import logging

import numpy as np
import torch

from ..base import MXNetError
from ..context import cpu, Context
from ..executor import Executor
from ..ndarray.ndarray import NDArray, zeros
from .. import optimizer as opt_mod
from .. import kvstore as kvs_mod
from .base_module import BaseModule

__all__ = ["Module"]


class Module(BaseModule):
    def __init__(self, symbol, data_names=("data",),
                 label_names=("softmax_label",), logger=logging,
                 context=None, work_load_list=None, fixed_param_names=None,
                 state_names=None):
        super().__init__(logger)
        self._symbol = symbol
        self._data_names = list(data_names)
        self._label_names = list(label_names or [])
        ctx = context or cpu()
        self._context = [ctx] if isinstance(ctx, Context) else list(ctx)
        self._fixed_param_names = set(fixed_param_names or [])
        self._exec = None
        self._arg_params = None
        self._aux_params = None
        self._optimizer = None
        self._updater = None
        self._kvstore = None
        self._preload_opt_states = None

    @property
    def symbol(self):
        return self._symbol

    @property
    def data_names(self):
        return self._data_names

    @property
    def label_names(self):
        return self._label_names

    # ---------------- bind ----------------
    def bind(self, data_shapes, label_shapes=None, for_training=True,
             inputs_need_grad=False, force_rebind=False, shared_module=None,
             grad_req="write"):
        if self.binded and not force_rebind:
            return
        self.for_training = for_training
        shapes = {}
        self._data_shapes = data_shapes
        self._label_shapes = label_shapes
        for d in data_shapes:
            name, shape = (d.name, d.shape) if hasattr(d, "name") else d[:2]
            shapes[name] = shape
        if label_shapes:
            for d in label_shapes:
                name, shape = (d.name, d.shape) if hasattr(d, "name") else d[:2]
                shapes[name] = shape
        # single executor on first ctx; multi-ctx slices the batch
        self._exec = Executor(self._symbol, self._context[0],
                              grad_req=grad_req if for_training else "null",
                              shapes=shapes)
        self._input_names = set(shapes.keys())
        self.binded = True

    # ---------------- params ----------------
    def init_params(self, initializer=None, arg_params=None, aux_params=None,
                    allow_missing=False, force_init=False,
                    allow_extra=False):
        if self.params_initialized and not force_init:
            return
        assert self.binded, "call bind before init_params"
        from .. import initializer as init_mod
        initializer = initializer or init_mod.Uniform(0.01)
        for name, arr in self._exec.arg_dict.items():
            if name in self._input_names:
                continue
            if arg_params is not None and name in arg_params:
                arr._t.detach().copy_(arg_params[name]._t.to(
                    arr._t.device, arr._t.dtype))
            else:
                if arg_params is not None and not allow_missing:
                    raise MXNetError(f"arg_params is provided but missing "
                                     f"'{name}'")
                initializer(name, arr)
        for name, arr in self._exec.aux_dict.items():
            if aux_params is not None and name in aux_params:
                arr._t.detach().copy_(aux_params[name]._t.to(
                    arr._t.device, arr._t.dtype))
            else:
                initializer(name, arr)
        self.params_initialized = True

    def get_params(self):
        assert self.binded and self.params_initialized
        arg_params = {name: arr.copy().as_in_context(cpu())
                      for name, arr in self._exec.arg_dict.items()
                      if name not in self._input_names}
        aux_params = {name: arr.copy().as_in_context(cpu())
                      for name, arr in self._exec.aux_dict.items()}
        return arg_params, aux_params

    def set_params(self, arg_params, aux_params, allow_missing=False,
                   force_init=True, allow_extra=False):
        self.init_params(initializer=None, arg_params=arg_params,
                         aux_params=aux_params, allow_missing=allow_missing,
                         force_init=force_init)

    # ---------------- optimizer ----------------
    def init_optimizer(self, kvstore="local", optimizer="sgd",
                       optimizer_params=None, force_init=False):
        assert self.binded and self.params_initialized
        if self.optimizer_initialized and not force_init:
            return
        optimizer_params = dict(optimizer_params or {"learning_rate": 0.01})
        if isinstance(optimizer, str):
            param_names = [n for n in self._exec.arg_dict
                           if n not in self._input_names]
            idx2name = dict(enumerate(param_names))
            optimizer = opt_mod.create(optimizer, param_idx2name=idx2name,
                                       **optimizer_params)
        self._optimizer = optimizer
        self._updater = opt_mod.get_updater(optimizer)
        self._param_names = [n for n in self._exec.arg_dict
                             if n not in self._input_names]
        if kvstore:
            kv = kvs_mod.create(kvstore) if isinstance(kvstore, str) else kvstore
            self._kvstore = kv
        self.optimizer_initialized = True
        if self._preload_opt_states is not None:
            self.load_optimizer_states(self._preload_opt_states)
            self._preload_opt_states = None

    # ---------------- compute ----------------
    def forward(self, data_batch, is_train=None):
        assert self.binded and self.params_initialized
        if is_train is None:
            is_train = self.for_training
        feed = {}
        for name, arr in zip(self._data_names, data_batch.data):
            feed[name] = arr
        if data_batch.label is not None and self._label_names:
            for name, arr in zip(self._label_names, data_batch.label):
                feed[name] = arr
        self._exec.forward(is_train=is_train, **feed)

    def backward(self, out_grads=None):
        assert self.binded and self.params_initialized
        self._exec.backward(out_grads)

    def update(self):
        assert self.optimizer_initialized
        for i, name in enumerate(self._param_names):
            if name in self._fixed_param_names:
                continue
            arr = self._exec.arg_dict[name]
            grad = arr.grad
            if grad is None:
                continue
            self._updater(i, grad, arr)

    def get_outputs(self, merge_multi_context=True):
        return self._exec.outputs

    def get_input_grads(self, merge_multi_context=True):
        return [self._exec.arg_dict[n].grad for n in self._data_names]

    def update_metric(self, eval_metric, labels, pre_sliced=False):
        eval_metric.update(labels, self.get_outputs())

    # ---------------- checkpoints ----------------
    def save_checkpoint(self, prefix, epoch, save_optimizer_states=False):
        from ..model import save_checkpoint
        arg_params, aux_params = self.get_params()
        save_checkpoint(prefix, epoch, self._symbol, arg_params, aux_params)
        if save_optimizer_states:
            self.save_optimizer_states(f"{prefix}-{epoch:04d}.states")

    @staticmethod
    def load(prefix, epoch, load_optimizer_states=False, **kwargs):
        from ..model import load_checkpoint
        sym, args, auxs = load_checkpoint(prefix, epoch)
        mod = Module(symbol=sym, **kwargs)
        mod._arg_params = args
        mod._aux_params = auxs
        mod.params_initialized = False
        mod._preloaded_params = (args, auxs)
        if load_optimizer_states:
            mod._preload_opt_states = f"{prefix}-{epoch:04d}.states"
        return mod

    def init_from_loaded(self):
        self.init_params(arg_params=self._arg_params,
                         aux_params=self._aux_params, allow_missing=False)

    def save_optimizer_states(self, fname):
        assert self.optimizer_initialized
        with open(fname, "wb") as f:
            f.write(self._updater.get_states(dump_optimizer=True))

    def load_optimizer_states(self, fname):
        assert self.optimizer_initialized
        with open(fname, "rb") as f:
            self._updater.set_states(f.read())
