"""Row-sparse NDArray (reference parity: python/mxnet/ndarray/sparse.py —
RowSparseNDArray / row_sparse_array / the `.retain` + `tostype` surface,
and src/operator/tensor/ sparse kernels for the ops used by embedding-style
workloads).

MI355X-native design: a RowSparseNDArray is (indices int64 [nnz],
values [nnz, *row_shape]) on the torch substrate.  It is a GRADIENT /
PARAMETER-SLICE format: what MXNet used it for is sparse embedding grads
(`Embedding(sparse_grad=True)`), the optimizer's lazy row update, and
kvstore row_sparse_pull — all of which operate row-wise, so no CSR math
is needed (CSRNDArray remains out of scope, documented in docs/PARITY.md).
"""
from __future__ import annotations

import warnings

import numpy as np
import torch

# torch's CSR support warns 'beta' on every construction
warnings.filterwarnings("ignore",
                        message="Sparse CSR tensor support is in beta")

from ..base import MXNetError
from .ndarray import NDArray


class RowSparseNDArray:
    """Rows `indices` carry `values`; all other rows are zero."""

    stype = "row_sparse"

    def __init__(self, indices: torch.Tensor, values: torch.Tensor,
                 shape):
        assert indices.dtype == torch.int64
        self._indices = indices
        self._values = values
        self._shape = tuple(shape)

    # ---------------- reference API surface ----------------
    @property
    def shape(self):
        return self._shape

    @property
    def dtype(self):
        from ..base import torch_dtype_to_np
        return torch_dtype_to_np(self._values.dtype)

    @property
    def context(self):
        from ..context import Context
        return Context.from_torch(self._values.device)

    @property
    def indices(self) -> NDArray:
        return NDArray(self._indices)

    @property
    def data(self) -> NDArray:
        return NDArray(self._values)

    def tostype(self, stype):
        if stype == "row_sparse":
            return self
        if stype != "default":
            raise MXNetError(f"unsupported stype {stype}")
        dense = torch.zeros(self._shape, dtype=self._values.dtype,
                            device=self._values.device)
        if self._indices.numel():
            dense.index_add_(0, self._indices, self._values)
        return NDArray(dense)

    def asnumpy(self):
        return self.tostype("default").asnumpy()

    def retain(self, rows) -> "RowSparseNDArray":
        """Keep only the given rows (reference: sparse.retain)."""
        rows_t = rows._t.long() if isinstance(rows, NDArray) else \
            torch.as_tensor(rows, dtype=torch.int64,
                            device=self._indices.device)
        keep = torch.isin(self._indices, rows_t)
        return RowSparseNDArray(self._indices[keep], self._values[keep],
                                self._shape)

    def copyto(self, other):
        if isinstance(other, NDArray):
            other._t.copy_(self.tostype("default")._t)
            return other
        raise MXNetError("copyto: unsupported target")

    def wait_to_read(self):
        if self._values.is_cuda:
            torch.cuda.synchronize(self._values.device)

    def __repr__(self):
        return (f"<RowSparseNDArray {self._shape} "
                f"nnz-rows={self._indices.numel()}>")


def row_sparse_array(arg1, shape=None, ctx=None, dtype=None):
    """Create a RowSparseNDArray from (data, indices) (reference API)."""
    from .. import nd
    if isinstance(arg1, tuple) and len(arg1) == 2:
        data, indices = arg1
        data_nd = data if isinstance(data, NDArray) else nd.array(
            data, ctx=ctx, dtype=dtype)
        idx_t = indices._t.long() if isinstance(indices, NDArray) else \
            torch.as_tensor(np.asarray(indices), dtype=torch.int64,
                            device=data_nd._t.device)
        assert shape is not None, "row_sparse_array needs an explicit shape"
        return RowSparseNDArray(idx_t, data_nd._t, shape)
    # dense input -> compress nonzero rows
    dense = arg1 if isinstance(arg1, NDArray) else nd.array(
        arg1, ctx=ctx, dtype=dtype)
    t = dense._t
    nz = (t.reshape(t.shape[0], -1).abs().sum(1) != 0).nonzero().flatten()
    return RowSparseNDArray(nz, t[nz].clone(), tuple(t.shape))


def _dense_tostype(nd_arr: NDArray, stype):
    """tostype('row_sparse'/'csr') on a dense NDArray."""
    if stype == "default":
        return nd_arr
    if stype == "row_sparse":
        return row_sparse_array(nd_arr)
    if stype == "csr":
        return csr_matrix(nd_arr)
    raise MXNetError(f"unsupported stype {stype}")


def sparse_row_update(weight: NDArray, grad: RowSparseNDArray, lr: float,
                      wd: float = 0.0):
    """Lazy SGD row update: only touched rows move (reference:
    optimizer lazy_update=True for row_sparse grads)."""
    w = weight._t
    idx, vals = grad._indices, grad._values
    if idx.numel() == 0:
        return
    rows = w[idx]
    rows.add_(vals + wd * rows, alpha=-lr)
    w[idx] = rows


class CSRNDArray:
    """Compressed sparse row matrix (reference parity:
    python/mxnet/ndarray/sparse.py — CSRNDArray; src/operator/tensor/
    cast_storage / dot CSR kernels).  Backed by torch.sparse_csr."""

    stype = "csr"

    def __init__(self, data: torch.Tensor, indices: torch.Tensor,
                 indptr: torch.Tensor, shape):
        self._shape = tuple(shape)
        self._csr = torch.sparse_csr_tensor(
            indptr.long(), indices.long(), data, size=self._shape)

    # ---------------- reference API surface ----------------
    @property
    def shape(self):
        return self._shape

    @property
    def dtype(self):
        from ..base import torch_dtype_to_np
        return torch_dtype_to_np(self._csr.values().dtype)

    @property
    def context(self):
        from ..context import Context
        return Context.from_torch(self._csr.device)

    @property
    def data(self) -> NDArray:
        return NDArray(self._csr.values())

    @property
    def indices(self) -> NDArray:
        return NDArray(self._csr.col_indices())

    @property
    def indptr(self) -> NDArray:
        return NDArray(self._csr.crow_indices())

    def tostype(self, stype):
        if stype == "csr":
            return self
        if stype == "default":
            return NDArray(self._csr.to_dense())
        if stype == "row_sparse":
            return row_sparse_array(NDArray(self._csr.to_dense()))
        raise MXNetError(f"unsupported stype {stype}")

    def asnumpy(self):
        return self._csr.to_dense().numpy()

    def dot(self, rhs) -> NDArray:
        """csr @ dense (reference: dot(csr, default) forward)."""
        r = rhs._t if isinstance(rhs, NDArray) else torch.as_tensor(rhs)
        return NDArray(self._csr.matmul(r.to(self._csr.values().dtype)))

    def __getitem__(self, key):
        if isinstance(key, slice):
            dense = self._csr.to_dense()[key]
            return csr_matrix(NDArray(dense))
        raise MXNetError("CSRNDArray: only row-slice indexing")

    def wait_to_read(self):
        pass

    def __repr__(self):
        return (f"<CSRNDArray {self._shape} "
                f"nnz={self._csr.values().numel()}>")


def csr_matrix(arg1, shape=None, ctx=None, dtype=None):
    """Create a CSRNDArray from (data, indices, indptr) or dense input
    (reference API mx.nd.sparse.csr_matrix)."""
    from .. import nd as _nd
    if isinstance(arg1, tuple) and len(arg1) == 3:
        data, indices, indptr = arg1
        to_t = lambda v, dt=None: (
            v._t if isinstance(v, NDArray)
            else torch.as_tensor(np.asarray(v), dtype=dt))
        assert shape is not None, "csr_matrix needs an explicit shape"
        return CSRNDArray(to_t(data, torch.float32),
                          to_t(indices, torch.int64),
                          to_t(indptr, torch.int64), shape)
    dense = arg1 if isinstance(arg1, NDArray) else _nd.array(
        arg1, ctx=ctx, dtype=dtype)
    sp = dense._t.to_sparse_csr()
    return CSRNDArray(sp.values(), sp.col_indices(), sp.crow_indices(),
                      dense.shape)
