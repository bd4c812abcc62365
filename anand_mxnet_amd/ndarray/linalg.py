"""mx.nd.linalg — linear-algebra operator namespace.

Reference surface: src/operator/tensor/la_op.{cc,cu} + linalg_impl.h
(gemm/gemm2 route through the cuBLAS wrappers there; the factorizations
are cuSOLVER).  Here the multiply family routes through the same
hand-written MFMA GEMM kernels as nd.dot (ops/hip), so the hot entry
points are native CDNA4; the factorizations (potrf/trsm/syevd/...) are
cold-path (none of the five target configs touch them) and compute via
torch.linalg in fp32, documented as such.
"""
import torch

from .ndarray import NDArray


def _t(x):
    return x._t if isinstance(x, NDArray) else x


def _wrap(t):
    return NDArray(t)


def gemm2(A, B, transpose_a=False, transpose_b=False, alpha=1.0):
    """C = alpha * op(A) @ op(B)  (la_op.cc: _linalg_gemm2).

    2-D inputs use the native TN/NN MFMA kernels via ops dot_impl;
    batched (>=3-D) inputs use the batched GEMM path.
    """
    from ..ops import nn as _nn
    a, b = _t(A), _t(B)
    if a.dim() == 2:
        out = _nn.dot_impl(a, b, transpose_a, transpose_b)
    else:
        out = _nn.batch_dot_impl(a.reshape(-1, *a.shape[-2:]),
                                 b.reshape(-1, *b.shape[-2:]),
                                 transpose_a, transpose_b)
        out = out.reshape(*a.shape[:-2], out.shape[-2], out.shape[-1])
    if alpha != 1.0:
        out = out * alpha
    return _wrap(out)


def gemm(A, B, C, transpose_a=False, transpose_b=False, alpha=1.0,
         beta=1.0):
    """C' = alpha * op(A) @ op(B) + beta * C  (la_op.cc: _linalg_gemm)."""
    prod = gemm2(A, B, transpose_a, transpose_b, alpha)
    return _wrap(prod._t + beta * _t(C))


def syrk(A, transpose=False, alpha=1.0):
    """alpha * A @ A^T (or A^T @ A) — la_op.cc: _linalg_syrk."""
    a = _t(A)
    return gemm2(A, _wrap(a), transpose_a=transpose,
                 transpose_b=not transpose, alpha=alpha)


def potrf(A):
    """Cholesky factor (lower). Cold path -> torch.linalg, fp32."""
    a = _t(A)
    return _wrap(torch.linalg.cholesky(a.float()).to(a.dtype))


def potri(A):
    """Inverse from Cholesky factor L: (L L^T)^-1."""
    a = _t(A).float()
    inv = torch.cholesky_inverse(a)
    return _wrap(inv.to(_t(A).dtype))


def trsm(A, B, transpose=False, rightside=False, lower=True, alpha=1.0):
    """Solve triangular op(A) X = alpha B (or X op(A) = alpha B)."""
    a = _t(A).float()
    b = (_t(B).float() * alpha)
    if rightside:
        x = torch.linalg.solve_triangular(
            a.transpose(-1, -2), b.transpose(-1, -2),
            upper=lower if transpose else not lower).transpose(-1, -2)
    else:
        if transpose:
            a = a.transpose(-1, -2)
            lower = not lower
        x = torch.linalg.solve_triangular(a, b, upper=not lower)
    return _wrap(x.to(_t(B).dtype))


def trmm(A, B, transpose=False, rightside=False, lower=True, alpha=1.0):
    """Triangular matrix multiply alpha op(tri(A)) @ B."""
    a = _t(A).float()
    a = torch.tril(a) if lower else torch.triu(a)
    if transpose:
        a = a.transpose(-1, -2)
    b = _t(B).float()
    out = (b @ a if rightside else a @ b) * alpha
    return _wrap(out.to(_t(B).dtype))


def sumlogdiag(A):
    """sum(log(diag(A))) per matrix — la_op.cc: _linalg_sumlogdiag."""
    a = _t(A).float()
    return _wrap(torch.log(torch.diagonal(a, dim1=-2, dim2=-1)).sum(-1)
                 .to(_t(A).dtype))


def extractdiag(A, offset=0):
    return _wrap(torch.diagonal(_t(A), offset=offset, dim1=-2, dim2=-1)
                 .contiguous())


def makediag(A, offset=0):
    return _wrap(torch.diag_embed(_t(A), offset=offset))


def det(A):
    a = _t(A).float()
    return _wrap(torch.linalg.det(a).to(_t(A).dtype))


def inverse(A):
    a = _t(A).float()
    return _wrap(torch.linalg.inv(a).to(_t(A).dtype))


def syevd(A):
    """Symmetric eigendecomposition: returns (U, L) with A = U^T L U."""
    a = _t(A).float()
    L, Q = torch.linalg.eigh(a)
    return (_wrap(Q.transpose(-1, -2).contiguous().to(_t(A).dtype)),
            _wrap(L.to(_t(A).dtype)))


def gelqf(A):
    """LQ factorization A = L Q (rows orthonormal Q)."""
    a = _t(A).float()
    Q, R = torch.linalg.qr(a.transpose(-1, -2))
    return (_wrap(R.transpose(-1, -2).contiguous().to(_t(A).dtype)),
            _wrap(Q.transpose(-1, -2).contiguous().to(_t(A).dtype)))
