"""SVRGModule (reference: contrib/svrg_optimization/svrg_module.py).

Module wrapper implementing Stochastic Variance-Reduced Gradient: every
`update_freq` epochs a full-pass gradient snapshot is taken at the
current weights; per-batch updates use
  g_corrected = g(w) - g_snapshot_batch(w_snap) + g_full(w_snap).
"""
from __future__ import annotations

from ...module.module import Module


class SVRGModule(Module):
    def __init__(self, symbol, data_names=("data",),
                 label_names=("softmax_label",), context=None,
                 update_freq=2, **kwargs):
        super().__init__(symbol, data_names=data_names,
                         label_names=label_names, context=context, **kwargs)
        self.update_freq = update_freq
        self._mod_aux = None       # snapshot-weight module
        self._full_grads = None    # name -> mean full-batch grad at w_snap

    def bind(self, *args, **kwargs):
        super().bind(*args, **kwargs)
        self._mod_aux = Module(self._symbol, self._data_names,
                               self._label_names, context=self._context)
        self._mod_aux.bind(*args, **kwargs)

    def init_params(self, *args, **kwargs):
        super().init_params(*args, **kwargs)
        arg, aux = self.get_params()
        self._mod_aux.set_params(arg, aux)

    def update_full_grads(self, train_data):
        """Full pass at the snapshot weights: mean gradient per param."""
        arg, aux = self.get_params()
        self._mod_aux.set_params(arg, aux)
        sums, nbatch = {}, 0
        for batch in train_data:
            self._mod_aux.forward(batch, is_train=True)
            self._mod_aux.backward()
            for name in self._param_names_of(self._mod_aux):
                g = self._mod_aux._exec.arg_dict[name].grad
                if g is None:
                    continue
                sums[name] = g.copy() if name not in sums else sums[name] + g
            nbatch += 1
        train_data.reset()
        self._full_grads = {n: v / nbatch for n, v in sums.items()}

    @staticmethod
    def _param_names_of(mod):
        ex = mod._exec
        return [n for n in ex.arg_dict if n not in ex._input_names]

    def forward_backward(self, data_batch):
        super().forward_backward(data_batch)
        if self._full_grads is None:
            return
        # gradient at the snapshot weights on the same batch, then the
        # variance-reduction correction applied in place
        self._mod_aux.forward(data_batch, is_train=True)
        self._mod_aux.backward()
        for name in self._param_names_of(self):
            gcur = self._exec.arg_dict[name].grad
            if gcur is None or name not in self._full_grads:
                continue
            gsnap = self._mod_aux._exec.arg_dict[name].grad
            gcur[:] = gcur - gsnap + self._full_grads[name]

    def fit(self, train_data, **kwargs):
        """Reference loop shape: refresh the snapshot every update_freq
        epochs, then run a normal epoch."""
        num_epoch = kwargs.pop("num_epoch", 1)
        for epoch in range(num_epoch):
            if self.binded and self.params_initialized and \
                    epoch % self.update_freq == 0:
                self.update_full_grads(train_data)
            super().fit(train_data, num_epoch=1, **kwargs)
            # base fit binds/inits on the first epoch; snapshot next loop
            kwargs.pop("begin_epoch", None)
            train_data.reset()
