"""Gluon fused-style RNN layers (LSTM/GRU/RNN over whole sequences; parity
with python/mxnet/gluon/rnn/rnn_layer.py, implemented by unrolling cells —
the reference's GPU path is cuDNN (rnn.cu), out of scope per PARITY.md)."""
from __future__ import annotations

from ..block import Block
from .rnn_cell import LSTMCell, GRUCell, RNNCell

__all__ = ["RNN", "LSTM", "GRU"]


class _RNNLayer(Block):
    _cell_cls = None

    def __init__(self, hidden_size, num_layers=1, layout="TNC",
                 dropout=0.0, bidirectional=False, input_size=0,
                 prefix=None, params=None, **cell_args):
        super().__init__(prefix=prefix, params=params)
        assert not bidirectional, \
            "bidirectional RNN layers not implemented (docs/PARITY.md)"
        assert layout in ("TNC", "NTC")
        self._layout = layout
        self._hidden = hidden_size
        self._num_layers = num_layers
        with self.name_scope():
            self._cells = []
            for i in range(num_layers):
                cell = self._cell_cls(hidden_size,
                                      input_size if i == 0 else hidden_size,
                                      prefix=f"l{i}_", **cell_args)
                self.register_child(cell, f"l{i}")
                self._cells.append(cell)

    def begin_state(self, batch_size=0, **kwargs):
        return [c.begin_state(batch_size, **kwargs) for c in self._cells]

    def forward(self, inputs, states=None):
        time_axis = 0 if self._layout == "TNC" else 1
        length = inputs.shape[time_axis]
        batch = inputs.shape[1 - time_axis]
        return_states = states is not None
        if states is None:
            states = self.begin_state(batch, ctx=inputs.context)
        x = inputs
        out_states = []
        for cell, st in zip(self._cells, states):
            outs, new_st = cell.unroll(
                length, x, begin_state=st,
                layout="NTC" if self._layout == "NTC" else "TNC")
            x = outs
            out_states.append(new_st)
        if return_states:
            return x, out_states
        return x


class LSTM(_RNNLayer):
    _cell_cls = LSTMCell


class GRU(_RNNLayer):
    _cell_cls = GRUCell


class RNN(_RNNLayer):
    _cell_cls = RNNCell
