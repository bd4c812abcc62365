"""SequentialModule: chain of Modules executed in order.

Reference parity: python/mxnet/module/sequential_module.py — a container
module whose forward feeds each sub-module's outputs into the next, with
`take_labels` marking which sub-module consumes the data labels (losses).
Backward runs the chain in reverse, handing each module's input gradients
to its predecessor.
"""
from __future__ import annotations

import logging

from .base_module import BaseModule


class SequentialModule(BaseModule):
    META_TAKE_LABELS = "take_labels"
    META_AUTO_WIRING = "auto_wiring"

    def __init__(self, logger=logging):
        super().__init__(logger=logger)
        self._modules = []
        self._metas = []
        self._label_shapes = None
        self.binded = False
        self.params_initialized = False
        self.optimizer_initialized = False

    def add(self, module, **kwargs):
        """Append a sub-module.  kwargs: take_labels=True marks the module
        that receives the batch labels (typically the loss head)."""
        self._modules.append(module)
        self._metas.append(kwargs)
        self.binded = False
        self.params_initialized = False
        self.optimizer_initialized = False
        return self

    # ------------------------------------------------------------------
    @property
    def data_names(self):
        if self._modules:
            return self._modules[0].data_names
        return ()

    @property
    def output_names(self):
        if self._modules:
            return self._modules[-1].output_names \
                if hasattr(self._modules[-1], "output_names") else ()
        return ()

    @property
    def label_shapes(self):
        return self._label_shapes

    def bind(self, data_shapes, label_shapes=None, for_training=True,
             inputs_need_grad=False, force_rebind=False, shared_module=None,
             grad_req="write"):
        if self.binded and not force_rebind:
            return
        assert shared_module is None, \
            "shared_module not supported for SequentialModule"
        self._label_shapes = label_shapes
        cur_shapes = data_shapes
        n = len(self._modules)
        for i, (mod, meta) in enumerate(zip(self._modules, self._metas)):
            take_labels = meta.get(self.META_TAKE_LABELS, False)
            mod.bind(cur_shapes,
                     label_shapes if take_labels else None,
                     for_training=for_training,
                     inputs_need_grad=(inputs_need_grad or i > 0),
                     grad_req=grad_req)
            if i + 1 < n:
                # wire this module's inferred output shapes to the next
                # module's data names
                shapes = {}
                for d in cur_shapes:
                    name, shape = (d.name, d.shape) if hasattr(d, "name") \
                        else d[:2]
                    shapes[name] = shape
                _, out_shapes, _ = mod.symbol.infer_shape(**shapes)
                nxt_names = self._modules[i + 1].data_names
                cur_shapes = [(nxt_names[j] if j < len(nxt_names)
                               else f"out{j}", shape)
                              for j, shape in enumerate(out_shapes)]
        self.binded = True

    def init_params(self, initializer=None, arg_params=None, aux_params=None,
                    allow_missing=False, force_init=False,
                    allow_extra=False):
        assert self.binded, "call bind before init_params"
        for mod in self._modules:
            mod.init_params(initializer=initializer, arg_params=arg_params,
                            aux_params=aux_params,
                            allow_missing=True,
                            force_init=force_init, allow_extra=True)
        self.params_initialized = True

    def get_params(self):
        args, auxs = {}, {}
        for mod in self._modules:
            a, x = mod.get_params()
            args.update(a)
            auxs.update(x)
        return args, auxs

    def set_params(self, arg_params, aux_params, allow_missing=False,
                   allow_extra=False):
        for mod in self._modules:
            mod.set_params(arg_params, aux_params, allow_missing=True,
                           allow_extra=True)

    def init_optimizer(self, kvstore="local", optimizer="sgd",
                       optimizer_params=(("learning_rate", 0.01),),
                       force_init=False):
        for mod in self._modules:
            mod.init_optimizer(kvstore=kvstore, optimizer=optimizer,
                               optimizer_params=optimizer_params,
                               force_init=force_init)
        self.optimizer_initialized = True

    # ------------------------------------------------------------------
    def forward(self, data_batch, is_train=None):
        from ..io import DataBatch
        batch = data_batch
        n = len(self._modules)
        for i, mod in enumerate(self._modules):
            mod.forward(batch, is_train=is_train)
            if i + 1 == n:
                break
            # next module consumes this module's outputs; labels flow
            # along so the take_labels module (loss head) can see them
            batch = DataBatch(data=mod.get_outputs(),
                              label=data_batch.label)

    def backward(self, out_grads=None):
        grads = out_grads
        for i in range(len(self._modules) - 1, -1, -1):
            self._modules[i].backward(out_grads=grads)
            if i > 0:
                grads = self._modules[i].get_input_grads()

    def update(self):
        for mod in self._modules:
            mod.update()

    def get_outputs(self, merge_multi_context=True):
        return self._modules[-1].get_outputs()

    def get_input_grads(self, merge_multi_context=True):
        return self._modules[0].get_input_grads()

    def update_metric(self, eval_metric, labels, pre_sliced=False):
        for mod, meta in zip(self._modules, self._metas):
            if meta.get(self.META_TAKE_LABELS, False):
                mod.update_metric(eval_metric, labels)
                return
        self._modules[-1].update_metric(eval_metric, labels)

    def install_monitor(self, mon):
        for mod in self._modules:
            if hasattr(mod, "install_monitor"):
                mod.install_monitor(mon)
