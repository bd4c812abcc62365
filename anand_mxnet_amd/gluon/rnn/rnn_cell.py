"""Gluon RNN cells (parity with python/mxnet/gluon/rnn/rnn_cell.py —
unfused RNN/LSTM/GRU cells; the reference's fused GPU path is cuDNN-bound
(rnn.cu) and is out of the hot path here, see docs/PARITY.md)."""
from __future__ import annotations

from ..block import HybridBlock
from ... import ndarray as nd

__all__ = ["RNNCell", "LSTMCell", "GRUCell", "SequentialRNNCell"]


class _BaseRNNCell(HybridBlock):
    def __init__(self, hidden_size, input_size=0, prefix=None, params=None):
        super().__init__(prefix=prefix, params=params)
        self._hidden_size = hidden_size
        self._input_size = input_size
        self._modified = False

    def state_info(self, batch_size=0):
        raise NotImplementedError

    def begin_state(self, batch_size=0, func=None, ctx=None, dtype="float32"):
        func = func or nd.zeros
        states = []
        for info in self.state_info(batch_size):
            states.append(func(shape=info["shape"], ctx=ctx, dtype=dtype)
                          if "ctx" in func.__code__.co_varnames
                          else func(info["shape"]))
        return states

    def unroll(self, length, inputs, begin_state=None, layout="NTC",
               merge_outputs=None):
        batch_axis = 0 if layout[0] == "N" else 1
        time_axis = 1 if layout[0] == "N" else 0
        batch = inputs.shape[batch_axis]
        if begin_state is None:
            begin_state = self.begin_state(batch, ctx=inputs.context)
        states = begin_state
        outputs = []
        for t in range(length):
            x_t = inputs.slice_axis(time_axis, t, t + 1).squeeze(time_axis)
            out, states = self(x_t, states)
            outputs.append(out)
        if merge_outputs is None or merge_outputs:
            outputs = nd.stack(*outputs, axis=time_axis)
        return outputs, states


class RNNCell(_BaseRNNCell):
    def __init__(self, hidden_size, activation="tanh", input_size=0,
                 i2h_weight_initializer=None, h2h_weight_initializer=None,
                 i2h_bias_initializer="zeros", h2h_bias_initializer="zeros",
                 **kwargs):
        super().__init__(hidden_size, input_size, **kwargs)
        self._activation = activation
        with self.name_scope():
            self.i2h_weight = self.params.get(
                "i2h_weight", shape=(hidden_size, input_size),
                init=i2h_weight_initializer, allow_deferred_init=True)
            self.h2h_weight = self.params.get(
                "h2h_weight", shape=(hidden_size, hidden_size),
                init=h2h_weight_initializer, allow_deferred_init=True)
            self.i2h_bias = self.params.get(
                "i2h_bias", shape=(hidden_size,),
                init=i2h_bias_initializer, allow_deferred_init=True)
            self.h2h_bias = self.params.get(
                "h2h_bias", shape=(hidden_size,),
                init=h2h_bias_initializer, allow_deferred_init=True)

    def state_info(self, batch_size=0):
        return [{"shape": (batch_size, self._hidden_size)}]

    def shape_infer(self, x, *args):
        self.i2h_weight.shape = (self._hidden_size, x.shape[-1])
        self.h2h_weight.shape = (self._hidden_size, self._hidden_size)
        self.i2h_bias.shape = (self._hidden_size,)
        self.h2h_bias.shape = (self._hidden_size,)

    def forward(self, x, states):
        from ..parameter import DeferredInitializationError
        try:
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        except DeferredInitializationError:
            self._infer_param_shapes(x)
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        i2h = nd.FullyConnected(x, p["i2h_weight"], p["i2h_bias"],
                                num_hidden=self._hidden_size)
        h2h = nd.FullyConnected(states[0], p["h2h_weight"], p["h2h_bias"],
                                num_hidden=self._hidden_size)
        out = nd.Activation(i2h + h2h, act_type=self._activation)
        return out, [out]


class LSTMCell(_BaseRNNCell):
    def __init__(self, hidden_size, input_size=0,
                 i2h_weight_initializer=None, h2h_weight_initializer=None,
                 i2h_bias_initializer="zeros", h2h_bias_initializer="zeros",
                 **kwargs):
        super().__init__(hidden_size, input_size, **kwargs)
        with self.name_scope():
            self.i2h_weight = self.params.get(
                "i2h_weight", shape=(4 * hidden_size, input_size),
                init=i2h_weight_initializer, allow_deferred_init=True)
            self.h2h_weight = self.params.get(
                "h2h_weight", shape=(4 * hidden_size, hidden_size),
                init=h2h_weight_initializer, allow_deferred_init=True)
            self.i2h_bias = self.params.get(
                "i2h_bias", shape=(4 * hidden_size,),
                init=i2h_bias_initializer, allow_deferred_init=True)
            self.h2h_bias = self.params.get(
                "h2h_bias", shape=(4 * hidden_size,),
                init=h2h_bias_initializer, allow_deferred_init=True)

    def state_info(self, batch_size=0):
        return [{"shape": (batch_size, self._hidden_size)},
                {"shape": (batch_size, self._hidden_size)}]

    def shape_infer(self, x, *args):
        self.i2h_weight.shape = (4 * self._hidden_size, x.shape[-1])
        self.h2h_weight.shape = (4 * self._hidden_size, self._hidden_size)
        self.i2h_bias.shape = (4 * self._hidden_size,)
        self.h2h_bias.shape = (4 * self._hidden_size,)

    def forward(self, x, states):
        from ..parameter import DeferredInitializationError
        try:
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        except DeferredInitializationError:
            self._infer_param_shapes(x)
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        nh = self._hidden_size
        gates = nd.FullyConnected(x, p["i2h_weight"], p["i2h_bias"],
                                  num_hidden=4 * nh) + \
            nd.FullyConnected(states[0], p["h2h_weight"], p["h2h_bias"],
                              num_hidden=4 * nh)
        slices = nd.split(gates, num_outputs=4, axis=1)
        i = nd.sigmoid(slices[0])
        f = nd.sigmoid(slices[1])
        g = nd.tanh(slices[2])
        o = nd.sigmoid(slices[3])
        c = f * states[1] + i * g
        h = o * nd.tanh(c)
        return h, [h, c]


class GRUCell(_BaseRNNCell):
    def __init__(self, hidden_size, input_size=0,
                 i2h_weight_initializer=None, h2h_weight_initializer=None,
                 i2h_bias_initializer="zeros", h2h_bias_initializer="zeros",
                 **kwargs):
        super().__init__(hidden_size, input_size, **kwargs)
        with self.name_scope():
            self.i2h_weight = self.params.get(
                "i2h_weight", shape=(3 * hidden_size, input_size),
                init=i2h_weight_initializer, allow_deferred_init=True)
            self.h2h_weight = self.params.get(
                "h2h_weight", shape=(3 * hidden_size, hidden_size),
                init=h2h_weight_initializer, allow_deferred_init=True)
            self.i2h_bias = self.params.get(
                "i2h_bias", shape=(3 * hidden_size,),
                init=i2h_bias_initializer, allow_deferred_init=True)
            self.h2h_bias = self.params.get(
                "h2h_bias", shape=(3 * hidden_size,),
                init=h2h_bias_initializer, allow_deferred_init=True)

    def state_info(self, batch_size=0):
        return [{"shape": (batch_size, self._hidden_size)}]

    def shape_infer(self, x, *args):
        self.i2h_weight.shape = (3 * self._hidden_size, x.shape[-1])
        self.h2h_weight.shape = (3 * self._hidden_size, self._hidden_size)
        self.i2h_bias.shape = (3 * self._hidden_size,)
        self.h2h_bias.shape = (3 * self._hidden_size,)

    def forward(self, x, states):
        from ..parameter import DeferredInitializationError
        try:
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        except DeferredInitializationError:
            self._infer_param_shapes(x)
            p = {n: v.data(x.context) for n, v in self._reg_params.items()}
        nh = self._hidden_size
        i2h = nd.FullyConnected(x, p["i2h_weight"], p["i2h_bias"],
                                num_hidden=3 * nh)
        h2h = nd.FullyConnected(states[0], p["h2h_weight"], p["h2h_bias"],
                                num_hidden=3 * nh)
        i2h_r, i2h_z, i2h_n = nd.split(i2h, num_outputs=3, axis=1)
        h2h_r, h2h_z, h2h_n = nd.split(h2h, num_outputs=3, axis=1)
        r = nd.sigmoid(i2h_r + h2h_r)
        z = nd.sigmoid(i2h_z + h2h_z)
        n = nd.tanh(i2h_n + r * h2h_n)
        h = (1 - z) * n + z * states[0]
        return h, [h]


class SequentialRNNCell(_BaseRNNCell):
    def __init__(self, prefix=None, params=None):
        super().__init__(0, prefix=prefix, params=params)

    def add(self, cell):
        self.register_child(cell)

    def state_info(self, batch_size=0):
        infos = []
        for cell in self._children.values():
            infos.extend(cell.state_info(batch_size))
        return infos

    def begin_state(self, batch_size=0, **kwargs):
        states = []
        for cell in self._children.values():
            states.extend(cell.begin_state(batch_size, **kwargs))
        return states

    def forward(self, x, states):
        next_states = []
        pos = 0
        for cell in self._children.values():
            n = len(cell.state_info())
            x, new_s = cell(x, states[pos:pos + n])
            pos += n
            next_states.extend(new_s)
        return x, next_states
