"""AMP — automatic mixed precision (parity with
python/mxnet/contrib/amp/amp.py: init patches op dispatch, init_trainer
wraps the Trainer with dynamic loss scaling, convert_model casts a model).

MI355X-native: the target dtype is bf16 (CDNA4 MFMA fast path); fp16 is
accepted for parity.  ``init()`` patches the registered ops so LP16_FUNCS
inputs are cast down and FP32_FUNCS inputs are cast up, reproducing the
reference's list-driven graph pass at dispatch time (there is no separate
graph compiler to pass over — dispatch IS the graph)."""
from __future__ import annotations

import functools

import torch

from ...base import dtype_from_any
from ... import ndarray as nd_mod
from ...ops import registry as _reg
from . import lists
from .loss_scaler import LossScaler

_initialized = False
_target_dtype = torch.bfloat16


def _cast_args(args, kwargs, dtype):
    def c(x):
        if isinstance(x, torch.Tensor) and x.is_floating_point():
            return x.to(dtype)
        return x
    return [c(a) for a in args], {k: c(v) for k, v in kwargs.items()}


def init(target_dtype="bfloat16"):
    """Patch op dispatch with the AMP cast lists."""
    global _initialized, _target_dtype
    if _initialized:
        return
    _target_dtype = dtype_from_any(target_dtype)

    for name in lists.LP16_FUNCS:
        if name not in _reg.RAW:
            continue
        raw = _reg.RAW[name]

        @functools.wraps(raw)
        def lp(*args, __raw=raw, **kwargs):
            a, k = _cast_args(args, kwargs, _target_dtype)
            return __raw(*a, **k)

        _reg.RAW[name] = lp
        _rewrap(name)

    for name in lists.FP32_FUNCS:
        if name not in _reg.RAW:
            continue
        raw = _reg.RAW[name]

        @functools.wraps(raw)
        def fp(*args, __raw=raw, **kwargs):
            out_dtype = None
            for x in args:
                if isinstance(x, torch.Tensor) and x.is_floating_point():
                    out_dtype = x.dtype
                    break
            a, k = _cast_args(args, kwargs, torch.float32)
            out = __raw(*a, **k)
            if out_dtype is not None and isinstance(out, torch.Tensor):
                out = out.to(out_dtype)
            return out

        _reg.RAW[name] = fp
        _rewrap(name)
    _initialized = True


def _rewrap(name):
    """Re-generate the nd-level wrapper for a patched raw op."""
    fn = _reg.RAW[name]
    _reg.register(name)(fn)
    import sys
    nd = sys.modules["anand_mxnet_amd.ndarray"]
    setattr(nd, name, _reg.OPS[name])


def convert_model(net, target_dtype="bfloat16"):
    """Cast a Gluon model for mixed precision (BN/LN params stay fp32 via
    their cast() overrides)."""
    net.cast(target_dtype)
    return net


_scalers = {}


def init_trainer(trainer):
    """Attach a dynamic LossScaler to a Trainer (reference amp.init_trainer)."""
    scaler = LossScaler()
    _scalers[id(trainer)] = scaler
    trainer._amp_loss_scaler = scaler
    orig_step = trainer.step

    def step(batch_size, ignore_stale_grad=False):
        overflow = scaler.has_overflow(trainer._params)
        scaler.update_scale(overflow)
        if overflow:  # skip update, just clear grads
            if trainer._reducer is not None:
                trainer._reducer.zero_grads()
            return
        trainer._optimizer.rescale_grad = \
            trainer._scale / batch_size / scaler.loss_scale
        orig_step(batch_size * 1, ignore_stale_grad)

    trainer.step = step
    return trainer


class scale_loss:
    """Context manager: ``with amp.scale_loss(loss, trainer) as l: l.backward()``"""

    def __init__(self, loss, trainer):
        self._trainer = trainer
        scaler = getattr(trainer, "_amp_loss_scaler", None)
        self._scale = scaler.loss_scale if scaler else 1.0
        self._loss = loss

    def __enter__(self):
        if isinstance(self._loss, (list, tuple)):
            return [l * self._scale for l in self._loss]
        return self._loss * self._scale

    def __exit__(self, *a):
        pass


def unscale(trainer):
    scaler = getattr(trainer, "_amp_loss_scaler", None)
    if scaler is None:
        return
    inv = 1.0 / scaler.loss_scale
    for p in trainer._params:
        if p.grad_req == "null":
            continue
        for g in p.list_grad():
            g._t.detach().mul_(inv)
