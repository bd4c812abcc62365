"""BucketingModule (parity with python/mxnet/module/bucketing_module.py):
one Module per bucket key (sequence length), shared parameters."""
from __future__ import annotations

import logging

from ..base import MXNetError
from .base_module import BaseModule
from .module import Module

__all__ = ["BucketingModule"]


class BucketingModule(BaseModule):
    def __init__(self, sym_gen, default_bucket_key=None, logger=logging,
                 context=None, **kwargs):
        super().__init__(logger)
        self._sym_gen = sym_gen
        self._default_bucket_key = default_bucket_key
        self._context = context
        self._kwargs = kwargs
        self._buckets = {}
        self._curr_module = None
        self._shared_params = None

    def _get_module(self, bucket_key, data_shapes, label_shapes,
                    for_training=True):
        if bucket_key not in self._buckets:
            symbol, data_names, label_names = self._sym_gen(bucket_key)
            mod = Module(symbol, data_names=data_names,
                         label_names=label_names, context=self._context,
                         **self._kwargs)
            mod.bind(data_shapes, label_shapes, for_training=for_training)
            if self._shared_params is not None:
                mod.init_params(arg_params=self._shared_params[0],
                                aux_params=self._shared_params[1],
                                allow_missing=False)
            self._buckets[bucket_key] = mod
        return self._buckets[bucket_key]

    def bind(self, data_shapes, label_shapes=None, for_training=True,
             **kwargs):
        if self._default_bucket_key is None:
            raise MXNetError("default_bucket_key required to bind")
        self._curr_module = self._get_module(self._default_bucket_key,
                                             data_shapes, label_shapes,
                                             for_training)
        self.binded = True
        self.for_training = for_training

    def init_params(self, **kwargs):
        self._curr_module.init_params(**kwargs)
        self.params_initialized = True
        self._shared_params = self._curr_module.get_params()

    def init_optimizer(self, **kwargs):
        self._curr_module.init_optimizer(**kwargs)
        self.optimizer_initialized = True
        self._opt_kwargs = kwargs

    def switch_bucket(self, bucket_key, data_shapes, label_shapes=None):
        mod = self._get_module(bucket_key, data_shapes, label_shapes,
                               self.for_training)
        if mod is not self._curr_module:
            # share params from the current module
            arg, aux = self._curr_module.get_params()
            mod.init_params(arg_params=arg, aux_params=aux,
                            allow_missing=False, force_init=True)
            if self.optimizer_initialized and not mod.optimizer_initialized:
                mod.init_optimizer(**self._opt_kwargs)
                # all buckets must share ONE optimizer + updater state
                # (reference: a single updater over the shared params —
                # per-bucket momentum would diverge otherwise).  Param
                # order matches across buckets (same symbol structure).
                mod._optimizer = self._curr_module._optimizer
                mod._updater = self._curr_module._updater
        self._curr_module = mod

    def forward(self, data_batch, is_train=None):
        key = getattr(data_batch, "bucket_key", self._default_bucket_key)
        self.switch_bucket(key, data_batch.provide_data,
                           data_batch.provide_label)
        self._curr_module.forward(data_batch, is_train)

    def backward(self, out_grads=None):
        self._curr_module.backward(out_grads)

    def update(self):
        self._curr_module.update()

    def get_outputs(self):
        return self._curr_module.get_outputs()

    def update_metric(self, eval_metric, labels):
        self._curr_module.update_metric(eval_metric, labels)

    def get_params(self):
        return self._curr_module.get_params()
