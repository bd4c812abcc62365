"""Tape-based autograd (parity with python/mxnet/autograd.py: record/pause/
backward/mark_variables/Function).

MI355X-native design: the tape is torch-autograd's graph — every op in
``anand_mxnet_amd.ops`` runs under ``torch.set_grad_enabled(is_recording())``,
so outside a ``record()`` scope nothing builds a graph (MXNet semantics, where
recording is opt-in, unlike torch's default).  ``grad_req='write'`` semantics
(grad overwritten per backward, not accumulated) are reproduced by clearing
write-req leaves at each ``backward`` entry.
"""
from __future__ import annotations

import threading
import weakref

import torch

__all__ = ["record", "pause", "train_mode", "predict_mode", "is_recording",
           "is_training", "backward", "mark_variables", "set_recording",
           "set_training", "Function"]

_state = threading.local()

# registry of arrays with attached grads, for write-req clearing
_attached = weakref.WeakSet()


def _st():
    if not hasattr(_state, "recording"):
        _state.recording = False
        _state.training = False
    return _state


def is_recording() -> bool:
    return _st().recording


def is_training() -> bool:
    return _st().training


def set_recording(is_record: bool) -> bool:
    st = _st()
    prev, st.recording = st.recording, bool(is_record)
    return prev


def set_training(train: bool) -> bool:
    st = _st()
    prev, st.training = st.training, bool(train)
    return prev


class _Scope:
    def __init__(self, recording=None, training=None):
        self._rec, self._train = recording, training

    def __enter__(self):
        st = _st()
        self._prev = (st.recording, st.training)
        if self._rec is not None:
            st.recording = self._rec
        if self._train is not None:
            st.training = self._train
        return self

    def __exit__(self, *a):
        st = _st()
        st.recording, st.training = self._prev


def record(train_mode: bool = True) -> _Scope:
    return _Scope(recording=True, training=train_mode)


def pause(train_mode: bool = False) -> _Scope:
    return _Scope(recording=False, training=train_mode)


def train_mode() -> _Scope:
    return _Scope(training=True)


def predict_mode() -> _Scope:
    return _Scope(training=False)


def _register_attached(arr):
    _attached.add(arr)


def backward(heads, head_grads=None, retain_graph: bool = False,
             train_mode: bool = True):
    """Compute gradients of heads w.r.t. all attached variables."""
    from .ndarray.ndarray import NDArray

    if isinstance(heads, NDArray):
        heads = [heads]
    if head_grads is None:
        head_grads = [None] * len(heads)
    elif isinstance(head_grads, NDArray):
        head_grads = [head_grads]

    # MXNet 'write' grad_req: each backward overwrites.  Arrays whose grad is
    # a persistent bucket view (parallel.bucketed) are zeroed by the Trainer
    # instead of detached (_zero_in_place).
    for arr in list(_attached):
        if getattr(arr, "_grad_req", "write") == "write" \
                and arr._t.grad is not None \
                and not getattr(arr, "_zero_in_place", False):
            arr._t.grad = None

    ts, gs = [], []
    for h, g in zip(heads, head_grads):
        if not h._t.requires_grad:
            raise RuntimeError("backward() on an array that was not computed "
                               "inside autograd.record() (no graph attached)")
        ts.append(h._t)
        if g is None:
            gs.append(torch.ones_like(h._t))
        else:
            gs.append(g._t)
    torch.autograd.backward(ts, gs, retain_graph=retain_graph)


def mark_variables(variables, gradients, grad_reqs="write"):
    from .ndarray.ndarray import NDArray
    if isinstance(variables, NDArray):
        variables = [variables]
        gradients = [gradients]
    if isinstance(grad_reqs, str):
        grad_reqs = [grad_reqs] * len(variables)
    for v, g, req in zip(variables, gradients, grad_reqs):
        v.attach_grad(grad_req=req)
        if g is not None:
            v._t.grad = g._t


class Function:
    """Custom differentiable function (parity with mxnet.autograd.Function).

    Subclass and implement ``forward(self, *inputs)`` and
    ``backward(self, *output_grads)`` operating on NDArrays; call the instance.
    """

    def __init__(self):
        self._saved = None

    def save_for_backward(self, *arrays):
        self._saved = arrays

    @property
    def saved_tensors(self):
        return self._saved

    def forward(self, *inputs):
        raise NotImplementedError

    def backward(self, *output_grads):
        raise NotImplementedError

    def __call__(self, *inputs):
        from .ndarray.ndarray import NDArray, _wrap

        outer = self

        class _F(torch.autograd.Function):
            @staticmethod
            def forward(ctx, *ts):
                nds = [_wrap(t) for t in ts]
                with pause():
                    outs = outer.forward(*nds)
                single = not isinstance(outs, (list, tuple))
                ctx._single = single
                if single:
                    outs = (outs,)
                return tuple(o._t for o in outs) if len(outs) > 1 else outs[0]._t

            @staticmethod
            def backward(ctx, *gts):
                gnds = [_wrap(g.contiguous()) if g is not None else None for g in gts]
                with pause():
                    gins = outer.backward(*gnds)
                if not isinstance(gins, (list, tuple)):
                    gins = (gins,)
                return tuple(g._t if g is not None else None for g in gins)

        ts = [x._t for x in inputs]
        with torch.set_grad_enabled(is_recording()):
            out = _F.apply(*ts)
        if isinstance(out, tuple):
            return [_wrap(t) for t in out]
        return _wrap(out)
