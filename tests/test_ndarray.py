import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd
from anand_mxnet_amd.test_utils import assert_almost_equal


def test_creation():
    a = nd.zeros((2, 3))
    assert a.shape == (2, 3)
    assert a.dtype == np.float32
    b = nd.ones((4,), dtype="float64")
    assert b.asnumpy().sum() == 4.0
    c = nd.full((2, 2), 7)
    assert c.asnumpy().tolist() == [[7, 7], [7, 7]]
    d = nd.arange(0, 10, 2)
    assert d.asnumpy().tolist() == [0, 2, 4, 6, 8]
    e = nd.array(np.eye(3))
    assert e.dtype == np.float32


def test_arithmetic():
    a = nd.array([[1.0, 2.0], [3.0, 4.0]])
    b = nd.array([[10.0, 20.0], [30.0, 40.0]])
    assert_almost_equal((a + b).asnumpy(), [[11, 22], [33, 44]])
    assert_almost_equal((b - a).asnumpy(), [[9, 18], [27, 36]])
    assert_almost_equal((a * 2).asnumpy(), [[2, 4], [6, 8]])
    assert_almost_equal((2 * a).asnumpy(), [[2, 4], [6, 8]])
    assert_almost_equal((b / a).asnumpy(), [[10, 10], [10, 10]])
    assert_almost_equal((1.0 / a).asnumpy(), 1.0 / a.asnumpy())
    assert_almost_equal((-a).asnumpy(), -a.asnumpy())
    assert_almost_equal((a ** 2).asnumpy(), a.asnumpy() ** 2)


def test_inplace():
    a = nd.ones((2, 2))
    a += 1
    assert a.asnumpy().tolist() == [[2, 2], [2, 2]]
    a *= 3
    assert a.asnumpy().tolist() == [[6, 6], [6, 6]]


def test_indexing():
    a = nd.array(np.arange(12).reshape(3, 4))
    assert a[1].asnumpy().tolist() == [4, 5, 6, 7]
    assert a[1, 2].asscalar() == 6
    assert a[0:2, 1:3].shape == (2, 2)
    a[0] = 100.0
    assert a[0].asnumpy().tolist() == [100] * 4
    a[1, 1] = -1
    assert a[1, 1].asscalar() == -1


def test_reshape_transpose():
    a = nd.array(np.arange(6).reshape(2, 3))
    assert a.reshape(3, 2).shape == (3, 2)
    assert a.reshape((-1,)).shape == (6,)
    assert a.reshape(0, 3).shape == (2, 3)
    assert a.T.shape == (3, 2)
    assert a.transpose((1, 0)).shape == (3, 2)
    assert a.expand_dims(0).shape == (1, 2, 3)
    assert a.expand_dims(0).squeeze(0).shape == (2, 3)
    assert a.flatten().shape == (2, 3)


def test_reductions():
    a = nd.array(np.arange(6, dtype=np.float32).reshape(2, 3))
    assert a.sum().asscalar() == 15
    assert_almost_equal(a.sum(axis=0).asnumpy(), [3, 5, 7])
    assert_almost_equal(a.mean(axis=1).asnumpy(), [1, 4])
    assert a.max().asscalar() == 5
    assert a.min().asscalar() == 0
    assert nd.sum(a, axis=1, keepdims=True).shape == (2, 1)
    assert nd.sum(a, axis=0, exclude=True).shape == (2,)
    assert abs(a.norm().asscalar() - np.sqrt((np.arange(6) ** 2).sum())) < 1e-5


def test_broadcast_ops():
    a = nd.ones((2, 1, 3))
    b = nd.ones((1, 4, 3))
    assert nd.broadcast_add(a, b).shape == (2, 4, 3)
    assert nd.broadcast_maximum(a, b * 2).asnumpy().max() == 2
    c = nd.broadcast_to(nd.ones((1, 3)), (5, 3))
    assert c.shape == (5, 3)


def test_concat_split_stack():
    a = nd.ones((2, 3))
    b = nd.zeros((2, 3))
    c = nd.concat(a, b, dim=0)
    assert c.shape == (4, 3)
    parts = nd.split(c, num_outputs=2, axis=0)
    assert parts[0].shape == (2, 3)
    s = nd.stack(a, b, axis=0)
    assert s.shape == (2, 2, 3)


def test_dot():
    a = np.random.uniform(-1, 1, (4, 5)).astype(np.float32)
    b = np.random.uniform(-1, 1, (5, 6)).astype(np.float32)
    assert_almost_equal(nd.dot(nd.array(a), nd.array(b)).asnumpy(),
                        a @ b, rtol=1e-4, atol=1e-5)
    assert_almost_equal(
        nd.dot(nd.array(a.T), nd.array(b), transpose_a=True).asnumpy(),
        a @ b, rtol=1e-4, atol=1e-5)
    assert_almost_equal(
        nd.dot(nd.array(a), nd.array(b.T), transpose_b=True).asnumpy(),
        a @ b, rtol=1e-4, atol=1e-5)


def test_batch_dot():
    a = np.random.uniform(-1, 1, (3, 4, 5)).astype(np.float32)
    b = np.random.uniform(-1, 1, (3, 5, 6)).astype(np.float32)
    assert_almost_equal(nd.batch_dot(nd.array(a), nd.array(b)).asnumpy(),
                        a @ b, rtol=1e-4, atol=1e-5)


def test_indexing_ops():
    w = nd.array(np.arange(12).reshape(4, 3))
    idx = nd.array([0, 2])
    assert nd.take(w, idx).asnumpy().tolist() == [[0, 1, 2], [6, 7, 8]]
    assert nd.Embedding(idx, w).shape == (2, 3)
    data = nd.array([[0.1, 0.9], [0.8, 0.2]])
    picked = nd.pick(data, nd.array([1, 0]))
    assert_almost_equal(picked.asnumpy(), [0.9, 0.8])
    oh = nd.one_hot(nd.array([0, 2]), 3)
    assert oh.asnumpy().tolist() == [[1, 0, 0], [0, 0, 1]]


def test_astype_copy_context():
    a = nd.ones((2, 2))
    b = a.astype("float64")
    assert b.dtype == np.float64
    c = a.copy()
    c[0, 0] = 5
    assert a[0, 0].asscalar() == 1
    assert a.context == mx.cpu()
    d = a.as_in_context(mx.cpu())
    assert d is a


def test_bf16_roundtrip():
    a = nd.ones((2, 2), dtype="bfloat16")
    assert a.dtype == "bfloat16"
    assert a.asnumpy().dtype == np.float32
    b = a.astype("float32") * 3
    assert b.asnumpy().tolist() == [[3, 3], [3, 3]]


def test_clip_where():
    a = nd.array([-2.0, 0.5, 3.0])
    assert nd.clip(a, -1, 1).asnumpy().tolist() == [-1, 0.5, 1]
    w = nd.where(a > 0, nd.ones_like(a), nd.zeros_like(a))
    assert w.asnumpy().tolist() == [0, 1, 1]


def test_waitall():
    nd.waitall()
    a = nd.ones((4,))
    a.wait_to_read()


def test_row_sparse_roundtrip_and_retain():
    from anand_mxnet_amd.ndarray.sparse import (RowSparseNDArray,
                                                row_sparse_array)
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import nd
    dense = np.zeros((6, 3), np.float32)
    dense[1] = [1, 2, 3]
    dense[4] = [4, 5, 6]
    rs = nd.array(dense).tostype("row_sparse")
    assert isinstance(rs, RowSparseNDArray)
    assert rs.stype == "row_sparse"
    assert sorted(rs.indices.asnumpy().tolist()) == [1, 4]
    np.testing.assert_allclose(rs.tostype("default").asnumpy(), dense)
    kept = rs.retain([4])
    np.testing.assert_allclose(kept.asnumpy()[4], [4, 5, 6])
    assert kept.asnumpy()[1].sum() == 0
    # explicit (data, indices) constructor
    rs2 = row_sparse_array((nd.array(np.ones((2, 3), np.float32)),
                            [0, 5]), shape=(6, 3))
    assert rs2.asnumpy().sum() == 6


def test_sparse_row_update():
    from anand_mxnet_amd.ndarray.sparse import (row_sparse_array,
                                                sparse_row_update)
    from anand_mxnet_amd import nd
    w = nd.array(np.ones((5, 2), np.float32))
    g = row_sparse_array((nd.array(np.full((2, 2), 2.0, np.float32)),
                          [1, 3]), shape=(5, 2))
    sparse_row_update(w, g, lr=0.5)
    out = w.asnumpy()
    np.testing.assert_allclose(out[1], [0, 0])   # 1 - 0.5*2
    np.testing.assert_allclose(out[0], [1, 1])   # untouched


def test_csr_matrix():
    from anand_mxnet_amd.ndarray.sparse import csr_matrix
    m = csr_matrix(([1.0, 2.0, 3.0], [0, 2, 1], [0, 2, 3]), shape=(2, 3))
    np.testing.assert_allclose(m.asnumpy(), [[1, 0, 2], [0, 3, 0]])
    assert m.stype == "csr"
    d = nd.array(np.arange(12, dtype="float32").reshape(3, 4))
    out = m.dot(d)
    np.testing.assert_allclose(out.asnumpy(),
                               m.asnumpy() @ d.asnumpy())
    dense = nd.array(np.array([[0, 1.0], [2.0, 0]], dtype="float32"))
    c = dense.tostype("csr")
    np.testing.assert_allclose(c.tostype("default").asnumpy(),
                               dense.asnumpy())
    np.testing.assert_allclose(c[0:1].asnumpy(), [[0, 1.0]])
    rs = c.tostype("row_sparse")
    assert rs.stype == "row_sparse"
    np.testing.assert_allclose(rs.tostype("default").asnumpy(),
                               dense.asnumpy())
    assert list(m.indptr.asnumpy()) == [0, 2, 3]
    assert list(m.indices.asnumpy()) == [0, 2, 1]


# ---------------------------------------------------------------------------
# mx.nd.linalg surface (reference: src/operator/tensor/la_op.cc)
# ---------------------------------------------------------------------------

def test_linalg_gemm_family():
    import numpy as np
    import anand_mxnet_amd as mx
    rs = np.random.RandomState(3)
    A = mx.nd.array(rs.rand(16, 8).astype("float32"))
    B = mx.nd.array(rs.rand(8, 12).astype("float32"))
    C = mx.nd.array(rs.rand(16, 12).astype("float32"))
    out = mx.nd.linalg.gemm2(A, B, alpha=2.0)
    assert np.allclose(out.asnumpy(), 2.0 * A.asnumpy() @ B.asnumpy(),
                       atol=1e-4)
    out = mx.nd.linalg.gemm(A, B, C, alpha=0.5, beta=2.0)
    assert np.allclose(out.asnumpy(),
                       0.5 * A.asnumpy() @ B.asnumpy() + 2.0 * C.asnumpy(),
                       atol=1e-4)
    # transposes
    out = mx.nd.linalg.gemm2(A, C, transpose_a=True)
    assert np.allclose(out.asnumpy(), A.asnumpy().T @ C.asnumpy(),
                       atol=1e-4)
    # batched
    Ab = mx.nd.array(rs.rand(4, 8, 8).astype("float32"))
    Bb = mx.nd.array(rs.rand(4, 8, 8).astype("float32"))
    out = mx.nd.linalg.gemm2(Ab, Bb)
    assert np.allclose(out.asnumpy(), Ab.asnumpy() @ Bb.asnumpy(),
                       atol=1e-4)
    s = mx.nd.linalg.syrk(A)
    assert np.allclose(s.asnumpy(), A.asnumpy() @ A.asnumpy().T, atol=1e-4)


def test_linalg_factorizations():
    import numpy as np
    import anand_mxnet_amd as mx
    rs = np.random.RandomState(5)
    M = rs.rand(8, 8).astype("float32")
    P = M @ M.T + 8 * np.eye(8, dtype="float32")
    L = mx.nd.linalg.potrf(mx.nd.array(P))
    assert np.allclose(L.asnumpy() @ L.asnumpy().T, P, atol=1e-3)
    inv = mx.nd.linalg.potri(L)
    assert np.allclose(inv.asnumpy(), np.linalg.inv(P), atol=1e-3)
    X = mx.nd.linalg.trsm(L, mx.nd.array(np.eye(8, dtype="float32")))
    assert np.allclose(L.asnumpy() @ X.asnumpy(), np.eye(8), atol=1e-3)
    sld = mx.nd.linalg.sumlogdiag(mx.nd.array(P))
    assert np.allclose(sld.asnumpy(), np.log(np.diag(P)).sum(), atol=1e-3)
    d = mx.nd.linalg.det(mx.nd.array(P))
    assert np.allclose(d.asnumpy(), np.linalg.det(P), rtol=1e-3)
    U, lam = mx.nd.linalg.syevd(mx.nd.array(P))
    rec = U.asnumpy().T @ np.diag(lam.asnumpy()) @ U.asnumpy()
    assert np.allclose(rec, P, atol=1e-2)
    Lq, Q = mx.nd.linalg.gelqf(mx.nd.array(M[:4]))
    assert np.allclose(Lq.asnumpy() @ Q.asnumpy(), M[:4], atol=1e-3)
    assert np.allclose(Q.asnumpy() @ Q.asnumpy().T, np.eye(4), atol=1e-3)


def test_linalg_triangular_and_diag():
    import numpy as np
    import anand_mxnet_amd as mx
    rs = np.random.RandomState(9)
    A = rs.rand(6, 6).astype("float32")
    B = rs.rand(6, 4).astype("float32")
    # trmm: lower-triangular multiply, left and right, with transpose
    out = mx.nd.linalg.trmm(mx.nd.array(A), mx.nd.array(B), alpha=2.0)
    assert np.allclose(out.asnumpy(), 2.0 * np.tril(A) @ B, atol=1e-4)
    out = mx.nd.linalg.trmm(mx.nd.array(A), mx.nd.array(B.T),
                            rightside=True, transpose=True)
    assert np.allclose(out.asnumpy(), B.T @ np.tril(A).T, atol=1e-4)
    d = mx.nd.linalg.extractdiag(mx.nd.array(A))
    assert np.allclose(d.asnumpy(), np.diag(A))
    m = mx.nd.linalg.makediag(d)
    assert np.allclose(m.asnumpy(), np.diag(np.diag(A)))
