"""Quantization (parity role of src/operator/quantization/* +
python/mxnet/contrib/quantization.py).

MI355X-native: the reference's int8/DP4A path is superseded by fp8
(OCP e4m3) on the CDNA4 fp8 MFMA units (BASELINE config #5).  This module
provides model conversion and amax calibration for the fp8 path; the
kernels live in src/kernels/gemm_fp8.hip."""
from __future__ import annotations

from ..base import MXNetError

__all__ = ["convert_to_fp8", "CalibrationCollector", "quantize_model"]

E4M3_MAX = 448.0


def convert_to_fp8(net, min_units=128):
    """Replace eligible Dense layers of a built Gluon network with the fp8
    forward path (in place).  A layer is eligible when its input dim tiles
    the fp8 kernel (in_units % 128 == 0)."""
    from ..gluon import nn

    def visit(block):
        for child in block._children.values():
            if type(child) is nn.Dense:
                in_units = child.weight.shape[1] if child.weight.shape else 0
                if in_units and in_units % 128 == 0 and \
                        child._units >= min_units:
                    child.__class__ = nn.Fp8Dense
            visit(child)
    visit(net)
    return net


class CalibrationCollector:
    """Per-tensor amax observer (reference calibrate.cc role): run
    representative batches, then read per-layer scales."""

    def __init__(self, net):
        self.net = net
        self.amax = {}
        self._installed = False

    def install(self):
        def hook(blk, inputs, output):
            import numpy as np
            from ..ndarray.ndarray import NDArray
            for i, x in enumerate(inputs):
                if isinstance(x, NDArray):
                    key = f"{blk.name}_in{i}"
                    v = float(x._t.detach().abs().amax())
                    self.amax[key] = max(self.amax.get(key, 0.0), v)
        self.net.apply(lambda b: b.register_forward_pre_hook(
            lambda blk, ins: hook(blk, ins, None)))
        self._installed = True

    def scales(self):
        return {k: (E4M3_MAX / v if v > 0 else 1.0)
                for k, v in self.amax.items()}


def quantize_model(sym=None, arg_params=None, aux_params=None, **kwargs):
    raise MXNetError(
        "int8 symbolic quantize_model is superseded by the fp8 path "
        "(convert_to_fp8); see docs/PARITY.md")
