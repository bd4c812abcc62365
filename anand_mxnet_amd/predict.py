"""C predict API backend (reference parity: src/c_api/c_predict_api.cc —
MXPredCreate / MXPredSetInput / MXPredForward / MXPredGetOutput).

The C-linkage entry points live in the native extension
(src/bindings.hip, `MXAPred*` symbols exported from _C's .so); they drive
this class through the embedded interpreter, exactly as libmxnet's C
predict API drove its GraphExecutor.  A C program links the extension
.so (plus libpython) and runs inference on exported
`-symbol.json` + `.params` checkpoints; tests/test_predict_api.py calls
the same symbols through ctypes to prove the ABI.
"""
from __future__ import annotations

import numpy as np

from . import symbol as _sym
from .context import cpu, gpu
from .io import DataBatch
from .module.module import Module
from .ndarray import ndarray as _nd
from .serializer import load_ndarrays


class CPredictor:
    def __init__(self, symbol_json: str, params_file: str, dev_type: int,
                 dev_id: int, input_names=("data",)):
        if symbol_json.lstrip().startswith("{"):
            sym = _sym.loads(symbol_json)
        else:
            sym = _sym.load(symbol_json)
        loaded = load_ndarrays(params_file)
        arg, aux = {}, {}
        for k, v in loaded.items():
            if k.startswith("arg:"):
                arg[k[4:]] = v
            elif k.startswith("aux:"):
                aux[k[4:]] = v
            else:
                arg[k] = v
        self._sym = sym
        self._arg, self._aux = arg, aux
        self._ctx = cpu() if dev_type == 1 else gpu(dev_id)
        self._inputs = {}
        self._input_names = list(input_names)
        self._mod = None
        self._outputs = []

    def set_input(self, name: str, data: np.ndarray):
        self._inputs[name] = np.ascontiguousarray(data, dtype=np.float32)

    def forward(self):
        shapes = [(n, self._inputs[n].shape) for n in self._input_names
                  if n in self._inputs]
        if not shapes:
            raise ValueError("no inputs set before forward")
        if self._mod is None or self._shapes != shapes:
            mod = Module(self._sym, data_names=[n for n, _ in shapes],
                         label_names=None, context=self._ctx)
            mod.bind(shapes, for_training=False)
            mod.set_params(self._arg, self._aux, allow_missing=False)
            self._mod = mod
            self._shapes = shapes
        batch = DataBatch(data=[_nd.array(self._inputs[n], ctx=self._ctx)
                                for n, _ in shapes])
        self._mod.forward(batch, is_train=False)
        self._outputs = [o.asnumpy() for o in self._mod.get_outputs()]

    def num_outputs(self) -> int:
        return len(self._outputs)

    def output_shape(self, idx: int):
        return list(self._outputs[idx].shape)

    def output_data(self, idx: int) -> np.ndarray:
        return np.ascontiguousarray(self._outputs[idx], dtype=np.float32)
