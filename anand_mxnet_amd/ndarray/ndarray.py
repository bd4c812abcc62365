"""NDArray — the framework's value type (parity with include/mxnet/ndarray.h
``NDArray`` + python/mxnet/ndarray/ndarray.py).

MI355X-native design: an NDArray owns a torch.Tensor resident in HIP memory
(torch-ROCm's caching allocator plays the role of the reference's
GPUPooledStorageManager over the 288 GB HBM3E pool).  Asynchronous execution
comes from HIP stream ordering — every op is stream-ordered on the array's
device, and Python only blocks at sync points (``asnumpy``, ``wait_to_read``)
— the same contract the reference implements with its threaded dependency
engine (src/engine/threaded_engine.cc), realized here directly on the HIP
runtime's stream semantics.
"""
from __future__ import annotations

import numpy as np
import torch

from ..base import dtype_from_any, torch_dtype_to_np, MXNetError
from ..context import Context, current_context
from .. import autograd as _ag

__all__ = ["NDArray", "array", "zeros", "ones", "empty", "full", "arange",
           "from_torch", "waitall", "concat_nd"]


def _wrap(t: torch.Tensor) -> "NDArray":
    return NDArray(t)


def _as_t(x, ref: torch.Tensor = None):
    """Coerce x (NDArray | scalar | numpy | tensor) to a torch value for ops."""
    if isinstance(x, NDArray):
        return x._t
    if isinstance(x, torch.Tensor):
        return x
    if isinstance(x, (int, float, bool)):
        return x
    if isinstance(x, np.ndarray):
        t = torch.from_numpy(np.ascontiguousarray(x))
        if ref is not None:
            t = t.to(device=ref.device)
        return t
    raise TypeError(f"cannot coerce {type(x)} to tensor")


class NDArray:
    __slots__ = ("_t", "_grad_req", "_zero_in_place", "__weakref__")

    def __init__(self, t: torch.Tensor):
        assert isinstance(t, torch.Tensor)
        self._t = t
        self._grad_req = None
        self._zero_in_place = False

    # ---------------- properties ----------------
    @property
    def shape(self):
        return tuple(self._t.shape)

    @property
    def ndim(self):
        return self._t.dim()

    @property
    def size(self):
        return self._t.numel()

    @property
    def dtype(self):
        return torch_dtype_to_np(self._t.dtype)

    @property
    def context(self) -> Context:
        return Context.from_torch(self._t.device)

    ctx = context

    @property
    def stype(self):
        return "default"

    # ---------------- sync / conversion ----------------
    def asnumpy(self) -> np.ndarray:
        t = self._t.detach()
        if t.is_cuda:
            torch.cuda.synchronize(t.device)
        if t.dtype == torch.bfloat16:
            t = t.float()
        return t.cpu().numpy()

    def asscalar(self):
        return self.asnumpy().item()

    def item(self):
        return self.asscalar()

    def wait_to_read(self):
        if self._t.is_cuda:
            torch.cuda.synchronize(self._t.device)

    def astype(self, dtype, copy=True) -> "NDArray":
        td = dtype_from_any(dtype)
        if not copy and self._t.dtype == td:
            return self
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.to(td))

    def copyto(self, other):
        if isinstance(other, Context):
            t = self._t.detach().to(device=other.torch_device(), non_blocking=True)
            return _wrap(t)
        if isinstance(other, NDArray):
            other._t.detach().copy_(self._t.detach(), non_blocking=True)
            return other
        raise TypeError(f"copyto: bad target {type(other)}")

    def copy(self) -> "NDArray":
        return _wrap(self._t.detach().clone())

    def as_in_context(self, ctx: Context) -> "NDArray":
        if self.context == ctx:
            return self
        return self.copyto(ctx)

    as_in_ctx = as_in_context

    def to_device(self, ctx):
        return self.as_in_context(ctx)

    # ---------------- autograd ----------------
    def attach_grad(self, grad_req: str = "write", stype=None):
        self._t = self._t.detach().requires_grad_(True)
        self._grad_req = grad_req
        _ag._register_attached(self)

    @property
    def grad(self):
        g = self._t.grad
        if g is None:
            if self._grad_req is not None:
                z = torch.zeros_like(self._t.detach())
                self._t.grad = z
                return _wrap(z)
            return None
        return _wrap(g)

    def detach(self) -> "NDArray":
        return _wrap(self._t.detach())

    def backward(self, out_grad=None, retain_graph=False, train_mode=True):
        _ag.backward([self], [out_grad] if out_grad is not None else None,
                     retain_graph=retain_graph, train_mode=train_mode)

    # ---------------- shape ops (views) ----------------
    def reshape(self, *shape):
        if len(shape) == 1 and isinstance(shape[0], (list, tuple)):
            shape = tuple(shape[0])
        # MXNet magic values: 0 copy dim, -1 infer
        out = []
        for i, s in enumerate(shape):
            if s == 0:
                out.append(self._t.shape[i])
            else:
                out.append(int(s))
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.reshape(out))

    def reshape_like(self, other):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.reshape(other.shape))

    @property
    def T(self):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.t())

    def transpose(self, axes=None):
        with torch.set_grad_enabled(_ag.is_recording()):
            if axes is None:
                axes = tuple(reversed(range(self.ndim)))
            return _wrap(self._t.permute(tuple(axes)))

    def expand_dims(self, axis):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.unsqueeze(axis))

    def squeeze(self, axis=None):
        with torch.set_grad_enabled(_ag.is_recording()):
            if axis is None:
                return _wrap(self._t.squeeze())
            return _wrap(self._t.squeeze(axis))

    def flatten(self):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.reshape(self._t.shape[0], -1))

    def broadcast_to(self, shape):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t.expand(tuple(shape)))

    def broadcast_like(self, other):
        return self.broadcast_to(other.shape)

    def slice_axis(self, axis, begin, end):
        from .. import nd
        return nd.slice_axis(self, axis=axis, begin=begin, end=end)

    def split(self, num_outputs, axis=1, squeeze_axis=False):
        from .. import nd
        return nd.split(self, num_outputs=num_outputs, axis=axis,
                        squeeze_axis=squeeze_axis)

    def tostype(self, stype):
        if stype == "default":
            return self
        from .sparse import _dense_tostype
        return _dense_tostype(self, stype)

    # ---------------- indexing ----------------
    def _idx(self, key):
        if isinstance(key, NDArray):
            return key._t
        if isinstance(key, tuple):
            return tuple(k._t if isinstance(k, NDArray) else k for k in key)
        return key

    def __getitem__(self, key):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(self._t[self._idx(key)])

    def __setitem__(self, key, value):
        v = _as_t(value, self._t)
        with torch.no_grad():
            self._t[self._idx(key)] = v

    def __len__(self):
        return self._t.shape[0]

    def __iter__(self):
        for i in range(len(self)):
            yield self[i]

    # ---------------- arithmetic ----------------
    def _binop(self, other, fn):
        o = _as_t(other, self._t)
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(fn(self._t, o))

    def __add__(self, o): return self._binop(o, torch.add)
    def __radd__(self, o): return self._binop(o, torch.add)
    def __sub__(self, o): return self._binop(o, torch.sub)
    def __rsub__(self, o): return self._binop(o, lambda a, b: torch.sub(b, a) if isinstance(b, torch.Tensor) else b - a)
    def __mul__(self, o): return self._binop(o, torch.mul)
    def __rmul__(self, o): return self._binop(o, torch.mul)
    def __truediv__(self, o): return self._binop(o, torch.div)
    def __rtruediv__(self, o): return self._binop(o, lambda a, b: torch.div(torch.as_tensor(b, device=a.device, dtype=a.dtype), a) if not isinstance(b, torch.Tensor) else torch.div(b, a))
    def __pow__(self, o): return self._binop(o, torch.pow)
    def __neg__(self):
        with torch.set_grad_enabled(_ag.is_recording()):
            return _wrap(-self._t)
    def __mod__(self, o): return self._binop(o, torch.remainder)

    def __iadd__(self, o):
        with torch.no_grad():
            self._t.detach().add_(_as_t(o, self._t))
        return self

    def __isub__(self, o):
        with torch.no_grad():
            self._t.detach().sub_(_as_t(o, self._t))
        return self

    def __imul__(self, o):
        with torch.no_grad():
            self._t.detach().mul_(_as_t(o, self._t))
        return self

    def __itruediv__(self, o):
        with torch.no_grad():
            self._t.detach().div_(_as_t(o, self._t))
        return self

    def __eq__(self, o): return self._binop(o, lambda a, b: (a == b).to(a.dtype))
    def __ne__(self, o): return self._binop(o, lambda a, b: (a != b).to(a.dtype))
    def __gt__(self, o): return self._binop(o, lambda a, b: (a > b).to(a.dtype))
    def __ge__(self, o): return self._binop(o, lambda a, b: (a >= b).to(a.dtype))
    def __lt__(self, o): return self._binop(o, lambda a, b: (a < b).to(a.dtype))
    def __le__(self, o): return self._binop(o, lambda a, b: (a <= b).to(a.dtype))

    def __hash__(self):
        return id(self)

    def __bool__(self):
        if self.size != 1:
            raise ValueError("ambiguous truth value of multi-element NDArray")
        return bool(self.asscalar())

    # ---------------- reductions as methods ----------------
    def sum(self, axis=None, keepdims=False):
        from .. import nd
        return nd.sum(self, axis=axis, keepdims=keepdims)

    def mean(self, axis=None, keepdims=False):
        from .. import nd
        return nd.mean(self, axis=axis, keepdims=keepdims)

    def max(self, axis=None, keepdims=False):
        from .. import nd
        return nd.max(self, axis=axis, keepdims=keepdims)

    def min(self, axis=None, keepdims=False):
        from .. import nd
        return nd.min(self, axis=axis, keepdims=keepdims)

    def norm(self):
        from .. import nd
        return nd.norm(self)

    def argmax(self, axis=None):
        from .. import nd
        return nd.argmax(self, axis=axis)

    def abs(self):
        from .. import nd
        return nd.abs(self)

    def clip(self, a_min, a_max):
        from .. import nd
        return nd.clip(self, a_min, a_max)

    def zeros_like(self):
        return _wrap(torch.zeros_like(self._t.detach()))

    def ones_like(self):
        return _wrap(torch.ones_like(self._t.detach()))

    def __repr__(self):
        return f"\n{self.asnumpy()}\n<NDArray {self.shape} @{self.context}>"


# ---------------- creation ----------------

def _ctx_dev(ctx):
    ctx = ctx if ctx is not None else current_context()
    return ctx.torch_device()


def array(source, ctx: Context = None, dtype=None) -> NDArray:
    if isinstance(source, NDArray):
        t = source._t.detach().clone()
    elif isinstance(source, torch.Tensor):
        t = source.detach().clone()
    else:
        a = np.asarray(source)
        if dtype is None and a.dtype != np.float32:
            a = a.astype(np.float32)  # MXNet default dtype: float32
        t = torch.from_numpy(np.ascontiguousarray(a))
    if dtype is not None:
        t = t.to(dtype_from_any(dtype))
    t = t.to(device=_ctx_dev(ctx))
    return NDArray(t)


def from_torch(t: torch.Tensor) -> NDArray:
    return NDArray(t)


def empty(shape, ctx=None, dtype=None) -> NDArray:
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.empty(tuple(shape), dtype=dtype_from_any(dtype),
                               device=_ctx_dev(ctx)))


def zeros(shape, ctx=None, dtype=None) -> NDArray:
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.zeros(tuple(shape), dtype=dtype_from_any(dtype),
                               device=_ctx_dev(ctx)))


def ones(shape, ctx=None, dtype=None) -> NDArray:
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.ones(tuple(shape), dtype=dtype_from_any(dtype),
                              device=_ctx_dev(ctx)))


def full(shape, val, ctx=None, dtype=None) -> NDArray:
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.full(tuple(shape), val, dtype=dtype_from_any(dtype),
                              device=_ctx_dev(ctx)))


def arange(start, stop=None, step=1.0, ctx=None, dtype=None) -> NDArray:
    if stop is None:
        start, stop = 0, start
    td = dtype_from_any(dtype)
    return NDArray(torch.arange(start, stop, step, dtype=td,
                                device=_ctx_dev(ctx)))


def waitall():
    if torch.cuda.is_available():
        for i in range(torch.cuda.device_count()):
            torch.cuda.synchronize(i)


def concat_nd(arrays, axis=0):
    with torch.set_grad_enabled(_ag.is_recording()):
        return _wrap(torch.cat([a._t for a in arrays], dim=axis))
