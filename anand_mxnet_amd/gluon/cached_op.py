"""CachedOp — hipGraph-backed cached execution of a hybridized block
(parity role of src/imperative/cached_op.cc: static_alloc + op bulking).

MI355X-native: instead of the reference's pre-planned engine segments, the
block's forward is captured once per input signature into a hipGraph and
replayed (one launch per call).  Used for inference dispatch
(``net.hybridize(static_alloc=True)``); training-step capture is exposed in
bench.py (whole fwd+bwd+update graphs)."""
from __future__ import annotations

import torch

from ..ndarray.ndarray import NDArray
from .. import autograd as _ag


class CachedGraph:
    def __init__(self, fn):
        self._fn = fn          # callable taking/returning torch tensors
        self._graphs = {}      # signature -> (graph, static_ins, static_outs)

    @staticmethod
    def _sig(tensors):
        return tuple((tuple(t.shape), t.dtype) for t in tensors)

    def __call__(self, *tensors):
        sig = self._sig(tensors)
        entry = self._graphs.get(sig)
        if entry is None:
            # warm up twice (allocator steady-state), then capture
            for _ in range(2):
                outs = self._fn(*tensors)
            static_ins = [t.clone() for t in tensors]
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                outs = self._fn(*static_ins)
            single = not isinstance(outs, (list, tuple))
            outs_list = [outs] if single else list(outs)
            entry = (g, static_ins, outs_list, single)
            self._graphs[sig] = entry
        g, static_ins, outs_list, single = entry
        for dst, src in zip(static_ins, tensors):
            if dst.data_ptr() != src.data_ptr():
                dst.copy_(src, non_blocking=True)
        g.replay()
        return outs_list[0] if single else outs_list


def maybe_cached_forward(block, args):
    """Dispatch a hybridized block's forward through its hipGraph when
    eligible (GPU tensors, inference mode, static_alloc requested)."""
    if not getattr(block, "_active", False):
        return None
    if not block._graph_kwargs.get("static_alloc", False):
        return None
    if _ag.is_recording():
        return None  # training path stays eager (bench captures whole steps)
    tensors = [a._t for a in args if isinstance(a, NDArray)]
    if not tensors or not all(t.is_cuda for t in tensors):
        return None
    cg = getattr(block, "_cached_graph_obj", None)
    if cg is None:
        def fn(*ts):
            nds = [NDArray(t) for t in ts]
            out = block._eager_forward(*nds)
            if isinstance(out, (list, tuple)):
                return [o._t for o in out]
            return out._t
        cg = CachedGraph(fn)
        block._cached_graph_obj = cg
    out = cg(*tensors)
    if isinstance(out, list):
        return [NDArray(t) for t in out]
    return NDArray(out)
