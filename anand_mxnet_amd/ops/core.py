"""Core tensor ops: elementwise, broadcast, reductions, matrix manipulation,
indexing, random sampling.

Parity targets (reference layout): src/operator/tensor/elemwise_*_op*,
broadcast_reduce_op*, matrix_op*, indexing_op.*, random/sample_op.* and the
mshadow_op.h functor zoo.  On GPU the elementwise family executes through
torch-ROCm's HIP kernels (the sanctioned substrate); the *hot* fused ops live
in ops/nn.py + ops/hip with hand-written CDNA4 kernels.
"""
from __future__ import annotations

import math
import torch

from .registry import register
from ..base import dtype_from_any

# ---------------- unary elementwise ----------------
_UNARY = {
    "relu": lambda x: torch.relu(x),
    "sigmoid": torch.sigmoid,
    "tanh": torch.tanh,
    "exp": torch.exp,
    "log": torch.log,
    "log2": torch.log2,
    "log10": torch.log10,
    "sqrt": torch.sqrt,
    "rsqrt": torch.rsqrt,
    "square": torch.square,
    "abs": torch.abs,
    "sign": torch.sign,
    "floor": torch.floor,
    "ceil": torch.ceil,
    "round": torch.round,
    "trunc": torch.trunc,
    "negative": torch.neg,
    "reciprocal": torch.reciprocal,
    "erf": torch.erf,
    "erfinv": torch.erfinv,
    "sin": torch.sin,
    "cos": torch.cos,
    "tan": torch.tan,
    "arcsin": torch.asin,
    "arccos": torch.acos,
    "arctan": torch.atan,
    "sinh": torch.sinh,
    "cosh": torch.cosh,
    "arcsinh": torch.asinh,
    "arccosh": torch.acosh,
    "arctanh": torch.atanh,
    "expm1": torch.expm1,
    "log1p": torch.log1p,
    "gamma": lambda x: torch.exp(torch.lgamma(x)),
    "gammaln": torch.lgamma,
    "logical_not": lambda x: (x == 0).to(x.dtype),
}
def _ew(name, *ts):
    """Generic elementwise engine dispatch (ops/hip ew.hip) for GPU bf16
    tensors — the framework-owned long-tail surface; returns None when the
    native path does not apply (CPU oracle / substrate runs instead)."""
    t0 = ts[0]
    if not (isinstance(t0, torch.Tensor) and t0.is_cuda):
        return None
    try:
        from . import hip as _h
    except Exception:
        return None
    if len(ts) == 1:
        return _h.ew_unary_dispatch(name, t0)
    return _h.ew_binary_dispatch(name, ts[0], ts[1])


def _make_unary_op(name, fn):
    def op(x):
        r = _ew(name, x)
        return r if r is not None else fn(x)
    op.__name__ = name
    return op


for _name, _fn in _UNARY.items():
    register(_name)(_make_unary_op(_name, _fn))


@register("softsign")
def softsign(x):
    return x / (1 + torch.abs(x))


@register("softrelu")
def softrelu(x):
    return torch.nn.functional.softplus(x)


def _act_or_torch(x, name, fallback):
    # GPU: the fused CDNA4 activation kernels (misc.hip) — nd.gelu in the
    # BERT MLM head was silently running torch's gelu (substrate audit)
    if x.is_cuda and x.dtype == torch.bfloat16 and x.numel() % 8 == 0:
        from .registry import GPU_IMPL
        impl = GPU_IMPL.get("act_" + name)
        if impl is not None:
            return impl(x)
    return fallback(x)


@register("gelu")
def gelu(x):
    return _act_or_torch(x, "gelu", torch.nn.functional.gelu)


@register("gelu_tanh")
def gelu_tanh(x):
    return _act_or_torch(
        x, "gelu_tanh",
        lambda t: torch.nn.functional.gelu(t, approximate="tanh"))


@register("silu")
def silu(x):
    return _act_or_torch(x, "silu", torch.nn.functional.silu)


@register("LeakyReLU")
def leaky_relu(x, gamma=None, act_type="leaky", slope=0.25, lower_bound=None,
               upper_bound=None):
    if act_type == "leaky":
        return torch.nn.functional.leaky_relu(x, slope)
    if act_type == "prelu":
        return torch.nn.functional.prelu(x, gamma)
    if act_type == "elu":
        return torch.nn.functional.elu(x, slope)
    if act_type == "selu":
        return torch.nn.functional.selu(x)
    if act_type == "gelu":
        return torch.nn.functional.gelu(x)
    raise ValueError(f"unknown act_type {act_type}")


# ---------------- binary / broadcast ----------------
_BINARY = {
    "elemwise_add": torch.add, "broadcast_add": torch.add,
    "elemwise_sub": torch.sub, "broadcast_sub": torch.sub,
    "elemwise_mul": torch.mul, "broadcast_mul": torch.mul,
    "elemwise_div": torch.div, "broadcast_div": torch.div,
    "broadcast_power": torch.pow, "power": torch.pow,
    "broadcast_maximum": torch.maximum, "maximum": torch.maximum,
    "broadcast_minimum": torch.minimum, "minimum": torch.minimum,
    "broadcast_mod": torch.remainder, "mod": torch.remainder,
    "broadcast_hypot": torch.hypot,
    "arctan2": torch.atan2,
}
def _make_binary_op(name, fn):
    def op(a, b):
        if isinstance(a, torch.Tensor) and isinstance(b, torch.Tensor):
            r = _ew(name, a, b)
            if r is not None:
                return r
        return fn(a, b)
    op.__name__ = name
    return op


for _name, _fn in _BINARY.items():
    register(_name)(_make_binary_op(_name, _fn))

for _name, _cmp in [("broadcast_equal", torch.eq), ("broadcast_not_equal", torch.ne),
                    ("broadcast_greater", torch.gt), ("broadcast_greater_equal", torch.ge),
                    ("broadcast_lesser", torch.lt), ("broadcast_lesser_equal", torch.le),
                    ("broadcast_logical_and", torch.logical_and),
                    ("broadcast_logical_or", torch.logical_or),
                    ("broadcast_logical_xor", torch.logical_xor)]:
    register(_name)(lambda a, b, _c=_cmp: _c(a, b).to(
        a.dtype if isinstance(a, torch.Tensor) else b.dtype))


@register("add_n")
def add_n(*args):
    """Multi-input sum (reference: ElementwiseSum, ndarray_function.cu)."""
    if len(args) == 1 and isinstance(args[0], (list, tuple)):
        args = args[0]
    out = args[0].clone() if len(args) > 1 else args[0]
    for a in args[1:]:
        out = out + a
    return out


@register("clip")
def clip(x, a_min=None, a_max=None):
    return torch.clamp(x, a_min, a_max)


@register("where")
def where(cond, a, b):
    return torch.where(cond.bool(), a, b)


# ---------------- reductions ----------------
def _axes(axis):
    if axis is None:
        return None
    if isinstance(axis, int):
        return (axis,)
    return tuple(axis)


@register("sum")
def sum_(x, axis=None, keepdims=False, exclude=False):
    ax = _axes(axis)
    if exclude and ax is not None:
        ax = tuple(i for i in range(x.dim()) if i not in
                   tuple(a % x.dim() for a in ax))
    if (ax is not None and len(ax) == 1 and isinstance(x, torch.Tensor)
            and x.dim() >= 1 and ax[0] % x.dim() == x.dim() - 1
            and not keepdims and x.is_cuda
            and x.dtype == torch.bfloat16):
        try:
            from . import hip as _h
            r = _h.ew_sum_last(x.contiguous())
            if r is not None:
                return r
        except Exception:
            pass
    if ax is None:
        out = x.sum()
        return out.reshape((1,) * x.dim()) if keepdims else out
    return x.sum(dim=ax, keepdim=keepdims)


@register("mean")
def mean(x, axis=None, keepdims=False, exclude=False):
    ax = _axes(axis)
    if exclude and ax is not None:
        ax = tuple(i for i in range(x.dim()) if i not in
                   tuple(a % x.dim() for a in ax))
    if ax is None:
        out = x.float().mean().to(x.dtype)
        return out.reshape((1,) * x.dim()) if keepdims else out
    return x.float().mean(dim=ax, keepdim=keepdims).to(x.dtype)


@register("max")
def max_(x, axis=None, keepdims=False):
    ax = _axes(axis)
    if ax is None:
        out = x.max()
        return out.reshape((1,) * x.dim()) if keepdims else out
    out = x
    for a in sorted((a % x.dim() for a in ax), reverse=True):
        out = out.max(dim=a, keepdim=keepdims).values
    return out


@register("min")
def min_(x, axis=None, keepdims=False):
    ax = _axes(axis)
    if ax is None:
        out = x.min()
        return out.reshape((1,) * x.dim()) if keepdims else out
    out = x
    for a in sorted((a % x.dim() for a in ax), reverse=True):
        out = out.min(dim=a, keepdim=keepdims).values
    return out


@register("prod")
def prod(x, axis=None, keepdims=False):
    ax = _axes(axis)
    if ax is None:
        out = x.prod()
        return out.reshape((1,) * x.dim()) if keepdims else out
    out = x
    for a in sorted((a % x.dim() for a in ax), reverse=True):
        out = out.prod(dim=a, keepdim=keepdims)
    return out


@register("norm")
def norm(x, ord=2, axis=None, keepdims=False):
    """fp32-accumulated norm (reference MXNET_SAFE_ACCUMULATION semantics)."""
    ax = _axes(axis)
    xf = x.float()
    if ax is None:
        out = torch.linalg.vector_norm(xf, ord=ord)
        out = out.reshape((1,)) if not keepdims else out.reshape((1,) * x.dim())
        return out.to(x.dtype)
    return torch.linalg.vector_norm(xf, ord=ord, dim=ax, keepdim=keepdims).to(x.dtype)


@register("argmax")
def argmax(x, axis=None, keepdims=False):
    if axis is None:
        return x.flatten().argmax().to(torch.float32)
    return x.argmax(dim=axis, keepdim=keepdims).to(torch.float32)


@register("argmin")
def argmin(x, axis=None, keepdims=False):
    if axis is None:
        return x.flatten().argmin().to(torch.float32)
    return x.argmin(dim=axis, keepdim=keepdims).to(torch.float32)


@register("topk")
def topk(x, axis=-1, k=1, ret_typ="indices", is_ascend=False):
    vals, idx = torch.topk(x, k, dim=axis, largest=not is_ascend)
    if ret_typ == "indices":
        return idx.to(torch.float32)
    if ret_typ == "value":
        return vals
    return [vals, idx.to(torch.float32)]


@register("sort")
def sort(x, axis=-1, is_ascend=True):
    return torch.sort(x, dim=axis, descending=not is_ascend).values


@register("argsort")
def argsort(x, axis=-1, is_ascend=True):
    return torch.argsort(x, dim=axis, descending=not is_ascend).to(torch.float32)


# ---------------- matrix / shape manipulation ----------------
@register("dot")
def dot(a, b, transpose_a=False, transpose_b=False):
    """GEMM. GPU path: hand-written MFMA kernel via ops.hip (see nn.py dot)."""
    from . import nn as _nn
    return _nn.dot_impl(a, b, transpose_a, transpose_b)


@register("batch_dot")
def batch_dot(a, b, transpose_a=False, transpose_b=False):
    from . import nn as _nn
    return _nn.batch_dot_impl(a, b, transpose_a, transpose_b)


@register("transpose")
def transpose(x, axes=None):
    if axes is None:
        axes = tuple(reversed(range(x.dim())))
    return x.permute(tuple(axes))


@register("reshape")
def reshape(x, shape):
    out = []
    for i, s in enumerate(shape):
        if s == 0:
            out.append(x.shape[i])
        else:
            out.append(int(s))
    return x.reshape(out)


@register("expand_dims")
def expand_dims(x, axis):
    return x.unsqueeze(axis)


@register("squeeze")
def squeeze(x, axis=None):
    return x.squeeze() if axis is None else x.squeeze(axis)


@register("Flatten")
def flatten(x):
    return x.reshape(x.shape[0], -1)


@register("concat")
def concat(*args, dim=1):
    if len(args) == 1 and isinstance(args[0], (list, tuple)):
        args = args[0]
    return torch.cat(list(args), dim=dim)


@register("stack")
def stack(*args, axis=0):
    if len(args) == 1 and isinstance(args[0], (list, tuple)):
        args = args[0]
    return torch.stack(list(args), dim=axis)


@register("split")
def split(x, num_outputs, axis=1, squeeze_axis=False):
    outs = torch.chunk(x, num_outputs, dim=axis)
    if squeeze_axis:
        outs = [o.squeeze(axis) for o in outs]
    return list(outs)


@register("slice")
def slice_(x, begin, end, step=None):
    idx = []
    for i in range(len(begin)):
        b = begin[i]
        e = end[i]
        s = step[i] if step is not None else None
        idx.append(slice(b, e, s))
    return x[tuple(idx)]


@register("slice_axis")
def slice_axis(x, axis, begin, end):
    if end is None:
        end = x.shape[axis]
    return x.narrow(axis, begin, end - begin)


@register("slice_like")
def slice_like(x, like, axes=None):
    if axes is None:
        axes = range(min(x.dim(), like.dim()))
    out = x
    for a in axes:
        out = out.narrow(a, 0, like.shape[a])
    return out


@register("tile")
def tile(x, reps):
    return x.repeat(tuple(reps))


@register("repeat")
def repeat(x, repeats, axis=None):
    if axis is None:
        return x.flatten().repeat_interleave(repeats)
    return x.repeat_interleave(repeats, dim=axis)


@register("flip")
def flip(x, axis):
    if isinstance(axis, int):
        axis = (axis,)
    return torch.flip(x, tuple(axis))


@register("pad")
def pad(x, mode="constant", pad_width=None, constant_value=0):
    # MXNet pad_width: (before, after) per axis, first dims first
    pw = []
    for i in range(len(pad_width) // 2 - 1, -1, -1):
        pw += [pad_width[2 * i], pad_width[2 * i + 1]]
    tmode = mode if mode != "edge" else "replicate"
    if tmode == "constant":
        return torch.nn.functional.pad(x, pw, mode=tmode,
                                       value=constant_value)
    # non-constant torch pad only accepts trailing-dim pads: strip
    # leading zero pairs (MXNet requires zeros on N/C anyway)
    while len(pw) > 2 and pw[-1] == 0 and pw[-2] == 0:
        pw = pw[:-2]
    return torch.nn.functional.pad(x, pw, mode=tmode)


@register("Cast")
def cast(x, dtype):
    return x.to(dtype_from_any(dtype))


cast_op = cast
register("cast")(cast)


@register("amp_cast")
def amp_cast(x, dtype):
    return x.to(dtype_from_any(dtype))


@register("amp_multicast")
def amp_multicast(*args, num_outputs=None):
    xs = list(args)
    widest = xs[0].dtype
    for x in xs[1:]:
        widest = torch.promote_types(widest, x.dtype)
    return [x.to(widest) for x in xs]


@register("broadcast_to")
def broadcast_to(x, shape):
    shp = [x.shape[i] if s == 0 else s for i, s in enumerate(shape)]
    return x.expand(shp)


@register("broadcast_axis")
def broadcast_axis(x, axis, size):
    if isinstance(axis, int):
        axis, size = (axis,), (size,)
    shape = list(x.shape)
    for a, s in zip(axis, size):
        shape[a] = s
    return x.expand(shape)


@register("zeros_like")
def zeros_like(x):
    return torch.zeros_like(x)


@register("ones_like")
def ones_like(x):
    return torch.ones_like(x)


@register("full_like")
def full_like(x, fill_value):
    return torch.full_like(x, fill_value)


@register("identity")
def identity(x):
    return x.clone()


@register("BlockGrad")
def block_grad(x):
    return x.detach()


@register("stop_gradient")
def stop_gradient(x):
    return x.detach()


@register("make_loss")
def make_loss(x):
    return x


@register("L2Normalization")
def l2_normalization(x, eps=1e-10, mode="instance"):
    if mode == "instance":
        n = torch.sqrt((x.float() ** 2).sum(dim=tuple(range(1, x.dim())),
                                            keepdim=True) + eps)
        return (x.float() / n).to(x.dtype)
    if mode == "channel":
        n = torch.sqrt((x.float() ** 2).sum(dim=1, keepdim=True) + eps)
        return (x.float() / n).to(x.dtype)
    raise ValueError(mode)


@register("smooth_l1")
def smooth_l1(x, scalar=1.0):
    s2 = scalar * scalar
    ax = torch.abs(x)
    return torch.where(ax < 1.0 / s2, 0.5 * s2 * x * x, ax - 0.5 / s2)


# ---------------- indexing ----------------
@register("Embedding")
def embedding(data, weight, input_dim=None, output_dim=None, dtype=None,
              sparse_grad=False):
    from . import nn as _nn
    return _nn.embedding_impl(data, weight)


@register("take")
def take(a, indices, axis=0, mode="clip"):
    idx = indices.long()
    if mode == "clip":
        idx = idx.clamp(0, a.shape[axis] - 1)
    elif mode == "wrap":
        idx = idx % a.shape[axis]
    if (axis == 0 and a.dim() == 2 and isinstance(a, torch.Tensor)
            and a.is_cuda and a.dtype == torch.bfloat16):
        # row gather == the embedding kernel (fwd gather + scatter-add
        # grad); reference indexing_op.h Take/AddTakeGrad
        from . import nn as _nn
        out = _nn.embedding_impl(idx.reshape(-1), a)
        return out.reshape(tuple(indices.shape) + (a.shape[1],))
    return torch.index_select(a, axis, idx.flatten()).reshape(
        a.shape[:axis] + tuple(indices.shape) + a.shape[axis + 1:])


@register("batch_take")
def batch_take(a, indices):
    return a.gather(1, indices.long().unsqueeze(1)).squeeze(1)


@register("pick")
def pick(data, index, axis=-1, keepdims=False, mode="clip"):
    idx = index.long()
    if mode == "clip":
        idx = idx.clamp(0, data.shape[axis] - 1)
    out = data.gather(axis, idx.unsqueeze(axis))
    return out if keepdims else out.squeeze(axis)


@register("one_hot")
def one_hot(indices, depth, on_value=1.0, off_value=0.0, dtype="float32"):
    oh = torch.nn.functional.one_hot(indices.long(), depth).to(dtype_from_any(dtype))
    if on_value != 1.0 or off_value != 0.0:
        oh = oh * (on_value - off_value) + off_value
    return oh


@register("gather_nd")
def gather_nd(data, indices):
    idx = indices.long()
    m = idx.shape[0]
    out_shape = tuple(idx.shape[1:]) + tuple(data.shape[m:])
    flat_idx = [idx[i].flatten() for i in range(m)]
    return data[tuple(flat_idx)].reshape(out_shape)


@register("scatter_nd")
def scatter_nd(data, indices, shape):
    idx = indices.long()
    out = torch.zeros(tuple(shape), dtype=data.dtype, device=data.device)
    m = idx.shape[0]
    flat_idx = tuple(idx[i].flatten() for i in range(m))
    out.index_put_(flat_idx, data.reshape((-1,) + tuple(shape[m:])),
                   accumulate=True)
    return out


@register("SequenceMask")
def sequence_mask(data, sequence_length=None, use_sequence_length=False,
                  value=0.0, axis=0):
    if not use_sequence_length or sequence_length is None:
        return data
    maxlen = data.shape[axis]
    pos = torch.arange(maxlen, device=data.device)
    if axis == 0:
        mask = pos.unsqueeze(1) < sequence_length.unsqueeze(0)  # (T, B)
        for _ in range(data.dim() - 2):
            mask = mask.unsqueeze(-1)
    else:  # axis == 1: (B, T, ...)
        mask = pos.unsqueeze(0) < sequence_length.unsqueeze(1)
        for _ in range(data.dim() - 2):
            mask = mask.unsqueeze(-1)
    return torch.where(mask, data, torch.as_tensor(value, dtype=data.dtype,
                                                   device=data.device))


@register("SequenceLast")
def sequence_last(data, sequence_length=None, use_sequence_length=False, axis=0):
    if not use_sequence_length:
        return data.select(axis, data.shape[axis] - 1)
    idx = (sequence_length.long() - 1)
    if axis == 0:
        return data[idx, torch.arange(data.shape[1], device=data.device)]
    return data[torch.arange(data.shape[0], device=data.device), idx]


@register("SequenceReverse")
def sequence_reverse(data, sequence_length=None, use_sequence_length=False, axis=0):
    if not use_sequence_length:
        return torch.flip(data, (axis,))
    out = data.clone()
    for b in range(data.shape[1]):
        n = int(sequence_length[b])
        out[:n, b] = torch.flip(data[:n, b], (0,))
    return out


@register("boolean_mask")
def boolean_mask(data, index, axis=0):
    return data[index.bool()] if axis == 0 else data.transpose(0, axis)[index.bool()].transpose(0, axis)


# ---------------- random ----------------
_rng_salt = [7919]


def _native_sample(shape, a, b, normal, dtype, device):
    """Counter-hash sampling kernel (reference random/sample_op.* family)
    for GPU bf16 requests; None -> torch substrate."""
    if device is None or dtype_from_any(dtype) != torch.bfloat16:
        return None
    dev = torch.device(device) if not isinstance(device, torch.device)         else device
    if dev.type != "cuda":
        return None
    try:
        from ..base import lib
        C = lib()
        if C is None or not hasattr(C, "rng_sample"):
            return None
        _rng_salt[0] = (_rng_salt[0] * 2654435761 + 97) % (1 << 62)
        return C.rng_sample(list(shape), float(a), float(b),
                            _rng_salt[0], normal, dev.index or 0)
    except Exception:
        return None


@register("random_uniform")
def random_uniform(low=0.0, high=1.0, shape=(1,), dtype="float32", device=None):
    r = _native_sample(shape, low, high, False, dtype, device)
    if r is not None:
        return r
    td = dtype_from_any(dtype)
    return torch.empty(tuple(shape), dtype=td, device=device).uniform_(low, high)


@register("random_normal")
def random_normal(loc=0.0, scale=1.0, shape=(1,), dtype="float32", device=None):
    r = _native_sample(shape, loc, scale, True, dtype, device)
    if r is not None:
        return r
    td = dtype_from_any(dtype)
    return torch.empty(tuple(shape), dtype=td, device=device).normal_(loc, scale)


@register("random_randint")
def random_randint(low, high, shape=(1,), dtype="int32", device=None):
    return torch.randint(low, high, tuple(shape), dtype=dtype_from_any(dtype),
                         device=device)


@register("random_gamma")
def random_gamma(alpha=1.0, beta=1.0, shape=(1,), dtype="float32", device=None):
    d = torch.distributions.Gamma(alpha, 1.0 / beta)
    return d.sample(tuple(shape)).to(dtype_from_any(dtype))


@register("random_exponential")
def random_exponential(lam=1.0, shape=(1,), dtype="float32", device=None):
    return torch.empty(tuple(shape), dtype=dtype_from_any(dtype),
                       device=device).exponential_(lam)


@register("random_poisson")
def random_poisson(lam=1.0, shape=(1,), dtype="float32", device=None):
    return torch.poisson(torch.full(tuple(shape), float(lam),
                                    device=device)).to(dtype_from_any(dtype))


@register("sample_multinomial")
def sample_multinomial(data, shape=1, get_prob=False, dtype="int32"):
    n = shape if isinstance(shape, int) else shape[0]
    idx = torch.multinomial(data, n, replacement=True)
    if get_prob:
        lp = torch.log(data.gather(-1, idx))
        return [idx.to(dtype_from_any(dtype)), lp]
    return idx.to(dtype_from_any(dtype))


@register("shuffle")
def shuffle(x):
    return x[torch.randperm(x.shape[0], device=x.device)]


# ---------------- linalg (subset) ----------------
@register("linalg_gemm2")
def linalg_gemm2(a, b, transpose_a=False, transpose_b=False, alpha=1.0):
    from . import nn as _nn
    return alpha * _nn.batch_dot_impl(a, b, transpose_a, transpose_b) \
        if a.dim() > 2 else alpha * _nn.dot_impl(a, b, transpose_a, transpose_b)


@register("linalg_potrf")
def linalg_potrf(a):
    return torch.linalg.cholesky(a)


@register("linalg_trsm")
def linalg_trsm(a, b, transpose=False, rightside=False, lower=True, alpha=1.0):
    return alpha * torch.linalg.solve_triangular(
        a.transpose(-1, -2) if transpose else a, b,
        upper=(not lower) ^ transpose, left=not rightside)


@register("diag")
def diag(x, k=0):
    return torch.diagonal(x, offset=k, dim1=-2, dim2=-1) if x.dim() >= 2 \
        else torch.diag(x, k)


@register("index_copy")
def index_copy(old, index, new_tensor, out=None):
    """Copy rows of new_tensor into old at index (reference:
    src/operator/contrib/index_copy.cc)."""
    res = old.clone()
    res.index_copy_(0, index.long(), new_tensor.to(old.dtype))
    return res


@register("box_nms")
def box_nms(data, overlap_thresh=0.5, valid_thresh=0.0, topk=-1,
            coord_start=2, score_index=1, id_index=-1, force_suppress=False,
            in_format="corner", out_format="corner", background_id=-1):
    """Non-maximum suppression over [..., N, K] boxes (reference:
    src/operator/contrib/bounding_box.cc).  Suppressed entries get
    score -1 (reference convention); boxes kept in score order."""
    shape = data.shape
    flat = data.reshape(-1, shape[-2], shape[-1]).clone()
    for b in range(flat.shape[0]):
        page = flat[b]
        scores = page[:, score_index]
        order = torch.argsort(scores, descending=True)
        if topk > 0:
            order = order[:topk]
        keep = []
        for i in order.tolist():
            if scores[i] <= valid_thresh:
                continue
            box_i = page[i, coord_start:coord_start + 4]
            if in_format == "center":
                box_i = torch.stack([box_i[0] - box_i[2] / 2,
                                     box_i[1] - box_i[3] / 2,
                                     box_i[0] + box_i[2] / 2,
                                     box_i[1] + box_i[3] / 2])
            ok = True
            for j in keep:
                if not force_suppress and id_index >= 0 and \
                        page[i, id_index] != page[j, id_index]:
                    continue
                box_j = page[j, coord_start:coord_start + 4]
                if in_format == "center":
                    box_j = torch.stack([box_j[0] - box_j[2] / 2,
                                         box_j[1] - box_j[3] / 2,
                                         box_j[0] + box_j[2] / 2,
                                         box_j[1] + box_j[3] / 2])
                ix1 = torch.maximum(box_i[0], box_j[0])
                iy1 = torch.maximum(box_i[1], box_j[1])
                ix2 = torch.minimum(box_i[2], box_j[2])
                iy2 = torch.minimum(box_i[3], box_j[3])
                iw = (ix2 - ix1).clamp(min=0)
                ih = (iy2 - iy1).clamp(min=0)
                inter = iw * ih
                a_i = (box_i[2] - box_i[0]) * (box_i[3] - box_i[1])
                a_j = (box_j[2] - box_j[0]) * (box_j[3] - box_j[1])
                iou = inter / (a_i + a_j - inter + 1e-12)
                if float(iou) > overlap_thresh:
                    ok = False
                    break
            if ok:
                keep.append(i)
        mask = torch.ones(page.shape[0], dtype=torch.bool,
                          device=page.device)
        mask[torch.tensor(keep, dtype=torch.long)
             if keep else torch.zeros(0, dtype=torch.long)] = False
        page[mask, score_index] = -1.0
    return flat.reshape(shape)


@register("_contrib_fft")
def contrib_fft(data, compute_size=128):
    """FFT on the last axis (reference: src/operator/contrib/fft.cc —
    cuFFT).  Real input [..., d] -> interleaved re/im [..., 2d]."""
    spec = torch.fft.fft(data.float(), dim=-1)
    return torch.stack([spec.real, spec.imag], dim=-1) \
        .reshape(*data.shape[:-1], 2 * data.shape[-1]).to(data.dtype)


@register("_contrib_ifft")
def contrib_ifft(data, compute_size=128):
    """Inverse FFT (reference contrib/ifft.cc): interleaved re/im
    [..., 2d] -> real [..., d].  Reference returns the UNSCALED inverse
    (caller divides by d)."""
    d = data.shape[-1] // 2
    z = data.float().reshape(*data.shape[:-1], d, 2)
    spec = torch.complex(z[..., 0], z[..., 1])
    return (torch.fft.ifft(spec, dim=-1).real * d).to(data.dtype)


@register("_contrib_ROIAlign")
def roi_align(data, rois, pooled_size=(1, 1), spatial_scale=1.0,
              sample_ratio=-1, position_sensitive=False, aligned=False):
    """ROI align (reference: src/operator/contrib/roi_align.cu) via
    grid_sample bilinear interpolation.  rois [R, 5]."""
    ph, pw = int(pooled_size[0]), int(pooled_size[1])
    n_roi = rois.shape[0]
    C, H, W = data.shape[1], data.shape[2], data.shape[3]
    out = data.new_zeros((n_roi, C, ph, pw))
    off = 0.5 if aligned else 0.0
    r = rois.detach().float()
    for i in range(n_roi):
        b = int(r[i, 0])
        x1 = float(r[i, 1]) * spatial_scale - off
        y1 = float(r[i, 2]) * spatial_scale - off
        x2 = float(r[i, 3]) * spatial_scale - off
        y2 = float(r[i, 4]) * spatial_scale - off
        sr = sample_ratio if sample_ratio > 0 else 2
        gh, gw = ph * sr, pw * sr
        ys = torch.linspace(y1, y2, 2 * gh + 1, device=data.device)[1::2]
        xs = torch.linspace(x1, x2, 2 * gw + 1, device=data.device)[1::2]
        gy = (ys / max(H - 1, 1)) * 2 - 1
        gx = (xs / max(W - 1, 1)) * 2 - 1
        grid = torch.stack(torch.meshgrid(gy, gx, indexing="ij"),
                           dim=-1)[..., [1, 0]].unsqueeze(0)
        samp = torch.nn.functional.grid_sample(
            data[b:b + 1].float(), grid, mode="bilinear",
            padding_mode="zeros", align_corners=True)
        out[i] = torch.nn.functional.avg_pool2d(samp, sr)[0].to(data.dtype)
    return out


@register("swapaxes")
@register("SwapAxis")
def swapaxes(data, dim1=0, dim2=0):
    return data.transpose(int(dim1), int(dim2))


@register("reverse")
def reverse(data, axis=0):
    axes = [axis] if isinstance(axis, int) else list(axis)
    return torch.flip(data, axes)


@register("space_to_depth")
def space_to_depth(data, block_size):
    b = int(block_size)
    n, c, h, w = data.shape
    x = data.reshape(n, c, h // b, b, w // b, b)
    return x.permute(0, 3, 5, 1, 2, 4).reshape(n, c * b * b, h // b, w // b)


@register("depth_to_space")
def depth_to_space(data, block_size):
    b = int(block_size)
    n, c, h, w = data.shape
    x = data.reshape(n, b, b, c // (b * b), h, w)
    return x.permute(0, 3, 4, 1, 5, 2).reshape(n, c // (b * b), h * b, w * b)


@register("SliceChannel")
def slice_channel(data, num_outputs, axis=1, squeeze_axis=False):
    """Legacy split (reference: src/operator/slice_channel.cc)."""
    outs = torch.chunk(data, int(num_outputs), dim=int(axis))
    if squeeze_axis:
        outs = [o.squeeze(int(axis)) for o in outs]
    return list(outs)


@register("broadcast_like")
def broadcast_like(lhs, rhs):
    return lhs.expand(rhs.shape)


@register("shape_array")
def shape_array(data):
    return torch.tensor(list(data.shape), dtype=torch.int64)


@register("size_array")
def size_array(data):
    return torch.tensor([data.numel()], dtype=torch.int64)


# ---------------- int8 quantization ops (reference src/operator/quantization)

@register("_contrib_quantize")
@register("quantize")
def quantize(data, min_range, max_range, out_type="uint8"):
    """Affine quantization (reference: quantization/quantize.cc):
    returns (q, min, max)."""
    lo = float(min_range.reshape(-1)[0]) \
        if torch.is_tensor(min_range) else float(min_range)
    hi = float(max_range.reshape(-1)[0]) \
        if torch.is_tensor(max_range) else float(max_range)
    if out_type == "uint8":
        scale = 255.0 / max(hi - lo, 1e-12)
        q = ((data.float() - lo) * scale).round().clamp(0, 255) \
            .to(torch.uint8)
    else:  # int8: symmetric around 0 (reference quantize_v2 int8 mode)
        amax = max(abs(lo), abs(hi), 1e-12)
        scale = 127.0 / amax
        q = (data.float() * scale).round().clamp(-127, 127).to(torch.int8)
        lo, hi = -amax, amax
    return [q, torch.tensor([lo]), torch.tensor([hi])]


@register("_contrib_quantize_v2")
@register("quantize_v2")
def quantize_v2(data, out_type="int8", min_calib_range=None,
                max_calib_range=None):
    lo = min_calib_range if min_calib_range is not None \
        else float(data.min())
    hi = max_calib_range if max_calib_range is not None \
        else float(data.max())
    return quantize(data, torch.tensor([lo]), torch.tensor([hi]),
                    out_type=out_type)


@register("_contrib_dequantize")
@register("dequantize")
def dequantize(data, min_range, max_range, out_type="float32"):
    lo = float(min_range.reshape(-1)[0])
    hi = float(max_range.reshape(-1)[0])
    if data.dtype == torch.uint8:
        scale = (hi - lo) / 255.0
        return data.float() * scale + lo
    amax = max(abs(lo), abs(hi))
    return data.float() * (amax / 127.0)


def _i8_gemm(x8, w8):
    """Exact int32 x8 @ w8^T; CDNA4 i8 MFMA kernel on GPU (K % 128), torch
    integer matmul as the exact CPU oracle."""
    if x8.is_cuda and x8.shape[-1] % 128 == 0:
        from . import hip as _h  # noqa: F401 (registers/loads _C)
        from ..base import lib
        return lib().gemm_i8(x8.contiguous(), w8.contiguous())
    return torch.matmul(x8.to(torch.int32), w8.to(torch.int32).t())


@register("_contrib_quantized_fully_connected")
@register("quantized_fully_connected")
def quantized_fully_connected(data, weight, bias, min_data, max_data,
                              min_weight, max_weight, min_bias=None,
                              max_bias=None, num_hidden=None, no_bias=True,
                              flatten=True):
    """int8 x int8 -> int32 FC (reference
    quantization/quantized_fully_connected.cc).  Inputs are symmetric
    int8 with float ranges; output int32 with the combined range so a
    following requantize/dequantize recovers floats."""
    x = data.reshape(data.shape[0], -1) if flatten and data.dim() > 2 \
        else data
    out32 = _i8_gemm(x, weight)
    sd = max(abs(float(min_data.reshape(-1)[0])),
             abs(float(max_data.reshape(-1)[0]))) / 127.0
    sw = max(abs(float(min_weight.reshape(-1)[0])),
             abs(float(max_weight.reshape(-1)[0]))) / 127.0
    so = sd * sw
    if not no_bias and bias is not None:
        sb = max(abs(float(min_bias.reshape(-1)[0])),
                 abs(float(max_bias.reshape(-1)[0]))) / 127.0
        b32 = (bias.to(torch.int32).float() * (sb / so)).round() \
            .to(torch.int32)
        out32 = out32 + b32
    lim = float(2 ** 31 - 1) * so
    return [out32, torch.tensor([-lim]), torch.tensor([lim])]


@register("_contrib_quantized_conv")
@register("quantized_conv")
def quantized_conv(data, weight, bias, min_data, max_data, min_weight,
                   max_weight, kernel=None, stride=(1, 1), pad=(0, 0),
                   num_filter=None, no_bias=True, layout="NHWC"):
    """int8 convolution (reference quantized_conv.cc), NHWC.  Pointwise
    (1x1 s1) runs as the plain int8 GEMM; general windows go through an
    int32 unfold composition (exact)."""
    assert layout == "NHWC", "quantized_conv: NHWC only"
    N, H, W, C = data.shape
    K = weight.shape[0]
    R, S = (kernel if kernel else (weight.shape[1], weight.shape[2]))
    if R == 1 and S == 1 and tuple(stride) == (1, 1) and tuple(pad) == (0, 0):
        out32 = _i8_gemm(data.reshape(-1, C), weight.reshape(K, C)) \
            .reshape(N, H, W, K)
    else:
        xi = data.permute(0, 3, 1, 2).to(torch.int32).float()
        wi = weight.permute(0, 3, 1, 2).to(torch.int32).float()
        out32 = torch.nn.functional.conv2d(
            xi, wi, None, tuple(stride), tuple(pad)) \
            .round().to(torch.int32).permute(0, 2, 3, 1)
    sd = max(abs(float(min_data.reshape(-1)[0])),
             abs(float(max_data.reshape(-1)[0]))) / 127.0
    sw = max(abs(float(min_weight.reshape(-1)[0])),
             abs(float(max_weight.reshape(-1)[0]))) / 127.0
    lim = float(2 ** 31 - 1) * sd * sw
    return [out32, torch.tensor([-lim]), torch.tensor([lim])]


@register("_contrib_requantize")
@register("requantize")
def requantize(data, min_range, max_range, min_calib_range=None,
               max_calib_range=None):
    """int32 accum -> int8 with new range (reference requantize.cc)."""
    f = dequantize(data.to(torch.float32),
                   min_range, max_range) if data.dtype != torch.int32 \
        else data.float() * (max(abs(float(min_range.reshape(-1)[0])),
                                 abs(float(max_range.reshape(-1)[0])))
                             / float(2 ** 31 - 1))
    lo = min_calib_range if min_calib_range is not None else float(f.min())
    hi = max_calib_range if max_calib_range is not None else float(f.max())
    return quantize(f, torch.tensor([lo]), torch.tensor([hi]),
                    out_type="int8")
