"""Operator edge coverage (reference parity:
tests/python/unittest/test_operator.py — the long tail beyond the
hot-path ops covered in test_ndarray/test_gluon)."""
import numpy as np

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd


def _np(x):
    return x.asnumpy()


def test_pad_tile_repeat_flip():
    a = nd.array(np.arange(12, dtype="float32").reshape(1, 1, 3, 4))
    p = nd.pad(a, mode="constant", pad_width=(0, 0, 0, 0, 1, 1, 2, 2),
               constant_value=7.0)
    assert p.shape == (1, 1, 5, 8)
    assert float(_np(p)[0, 0, 0, 0]) == 7.0
    assert float(_np(p)[0, 0, 1, 2]) == 0.0
    t = nd.tile(nd.array([[1, 2]]), reps=(2, 3))
    assert t.shape == (2, 6)
    r = nd.repeat(nd.array([1, 2, 3]), repeats=2)
    np.testing.assert_allclose(_np(r), [1, 1, 2, 2, 3, 3])
    f = nd.flip(nd.array([[1, 2], [3, 4]]), axis=1)
    np.testing.assert_allclose(_np(f), [[2, 1], [4, 3]])


def test_topk_sort_argsort():
    x = nd.array([[3.0, 1.0, 2.0], [0.0, 5.0, 4.0]])
    t = nd.topk(x, k=2, axis=1, ret_typ="value")
    np.testing.assert_allclose(_np(t), [[3, 2], [5, 4]])
    s = nd.sort(x, axis=1)
    np.testing.assert_allclose(_np(s), [[1, 2, 3], [0, 4, 5]])
    a = nd.argsort(x, axis=1)
    np.testing.assert_allclose(_np(a), [[1, 2, 0], [0, 2, 1]])


def test_one_hot_pick_diag():
    oh = nd.one_hot(nd.array([0, 2]), depth=3)
    np.testing.assert_allclose(_np(oh), [[1, 0, 0], [0, 0, 1]])
    x = nd.array([[1.0, 2.0], [3.0, 4.0]])
    p = nd.pick(x, nd.array([1, 0]), axis=1)
    np.testing.assert_allclose(_np(p), [2, 3])
    d = nd.diag(x)
    np.testing.assert_allclose(_np(d), [1, 4])


def test_sequence_ops():
    # [seq, batch, feat] layout (reference SequenceMask semantics)
    x = nd.array(np.arange(2 * 3 * 1, dtype="float32").reshape(3, 2, 1)
                 + 1.0)
    m = nd.SequenceMask(x, sequence_length=nd.array([2, 1]),
                        use_sequence_length=True, value=-1.0)
    out = _np(m)
    assert out[2, 0, 0] == -1.0 and out[1, 1, 0] == -1.0
    assert out[1, 0, 0] != -1.0
    last = nd.SequenceLast(x, sequence_length=nd.array([2, 1]),
                           use_sequence_length=True)
    np.testing.assert_allclose(_np(last).ravel(),
                               [_np(x)[1, 0, 0], _np(x)[0, 1, 0]])
    rev = nd.SequenceReverse(x, sequence_length=nd.array([2, 3]),
                             use_sequence_length=True)
    assert _np(rev)[0, 0, 0] == _np(x)[1, 0, 0]


def test_gather_scatter_boolean():
    x = nd.array(np.arange(6, dtype="float32").reshape(2, 3))
    idx = nd.array(np.array([[0, 1], [2, 0]]), dtype="int64")
    g = nd.gather_nd(x, idx)
    np.testing.assert_allclose(_np(g), [2.0, 3.0])
    s = nd.scatter_nd(nd.array([9.0, 8.0]), idx, shape=(2, 3))
    assert _np(s)[0, 2] == 9.0 and _np(s)[1, 0] == 8.0
    bm = nd.boolean_mask(x, nd.array([0, 1]))
    np.testing.assert_allclose(_np(bm), [[3, 4, 5]])


def test_norm_l2_normalize_smoothl1():
    x = nd.array([[3.0, 4.0], [6.0, 8.0]])
    n = nd.norm(x, ord=2, axis=1)
    np.testing.assert_allclose(_np(n), [5, 10])
    l2 = nd.L2Normalization(x, mode="instance")
    np.testing.assert_allclose(_np(l2)[0], [0.6, 0.8], rtol=1e-5)
    s = nd.smooth_l1(nd.array([-2.0, 0.5]), scalar=1.0)
    np.testing.assert_allclose(_np(s), [1.5, 0.125], rtol=1e-5)


def test_norm_layers():
    x = nd.array(np.random.RandomState(0)
                 .normal(size=(2, 4, 3)).astype("float32"))
    ga = nd.ones((4,))
    be = nd.zeros((4,))
    out = nd.InstanceNorm(x, ga, be, eps=1e-5)
    m = _np(out).mean(axis=2)
    np.testing.assert_allclose(m, np.zeros_like(m), atol=1e-5)
    g = nd.GroupNorm(x.reshape(2, 4, 3, 1), nd.ones((4,)), nd.zeros((4,)),
                     num_groups=2, eps=1e-5)
    assert g.shape == (2, 4, 3, 1)
    r = nd.RMSNorm(x, nd.ones((3,)), eps=1e-6)
    ref = _np(x) / np.sqrt((_np(x) ** 2).mean(-1, keepdims=True) + 1e-6)
    np.testing.assert_allclose(_np(r), ref, rtol=1e-4)


def test_leakyrelu_modes():
    x = nd.array([-1.0, 2.0])
    lk = nd.LeakyReLU(x, act_type="leaky", slope=0.1)
    np.testing.assert_allclose(_np(lk), [-0.1, 2.0], rtol=1e-6)
    el = nd.LeakyReLU(x, act_type="elu", slope=1.0)
    np.testing.assert_allclose(_np(el), [np.expm1(-1.0), 2.0], rtol=1e-5)
    pr = nd.LeakyReLU(x, nd.array([0.25]), act_type="prelu")
    np.testing.assert_allclose(_np(pr), [-0.25, 2.0], rtol=1e-6)


def test_broadcast_axis_slice_like_batch_take():
    x = nd.array(np.arange(4, dtype="float32").reshape(1, 4))
    b = nd.broadcast_axis(x, axis=0, size=3)
    assert b.shape == (3, 4)
    y = nd.zeros((2, 3))
    sl = nd.slice_like(nd.array(np.arange(12).reshape(3, 4)), y)
    assert sl.shape == (2, 3)
    bt = nd.batch_take(nd.array([[1.0, 2.0], [3.0, 4.0]]),
                       nd.array([1, 0], dtype="int64"))
    np.testing.assert_allclose(_np(bt), [2, 3])


def test_arctan2_hypot_erf():
    a = nd.array([1.0, -1.0])
    b = nd.array([1.0, 1.0])
    np.testing.assert_allclose(_np(nd.arctan2(a, b)),
                               np.arctan2([1, -1], [1, 1]), rtol=1e-6)
    np.testing.assert_allclose(_np(nd.broadcast_hypot(a, b)),
                               np.hypot([1, -1], [1, 1]), rtol=1e-6)
    np.testing.assert_allclose(_np(nd.erf(a)),
                               [0.8427008, -0.8427008], rtol=1e-5)
    np.testing.assert_allclose(_np(nd.erfinv(nd.array([0.5]))),
                               [0.4769363], rtol=1e-4)


def test_linalg_ops():
    rs = np.random.RandomState(1)
    a = rs.normal(size=(3, 3)).astype("float32")
    spd = a @ a.T + 3 * np.eye(3, dtype="float32")
    l = nd.linalg_potrf(nd.array(spd))
    np.testing.assert_allclose(_np(l) @ _np(l).T, spd, rtol=1e-4,
                               atol=1e-4)
    g = nd.linalg_gemm2(nd.array(a), nd.array(a), transpose_b=True)
    np.testing.assert_allclose(_np(g), a @ a.T, rtol=1e-5, atol=1e-5)
    # solve L x = b via trsm
    bvec = rs.normal(size=(3, 1)).astype("float32")
    x = nd.linalg_trsm(l, nd.array(bvec), rightside=False, lower=True)
    np.testing.assert_allclose(_np(l) @ _np(x), bvec, rtol=1e-4,
                               atol=1e-4)


def test_upsampling_lrn():
    x = nd.array(np.arange(16, dtype="float32").reshape(1, 1, 4, 4))
    u = nd.UpSampling(x, scale=2, sample_type="nearest")
    assert u.shape == (1, 1, 8, 8)
    assert float(_np(u)[0, 0, 0, 1]) == 0.0 and \
        float(_np(u)[0, 0, 0, 2]) == 1.0
    l = nd.LRN(x, nsize=3)
    assert l.shape == x.shape


def test_spatial_transformer_identity():
    x = nd.array(np.random.RandomState(0)
                 .normal(size=(2, 3, 5, 5)).astype("float32"))
    loc = nd.array(np.tile(np.array([1, 0, 0, 0, 1, 0], dtype="float32"),
                           (2, 1)))
    out = nd.SpatialTransformer(x, loc, target_shape=(5, 5))
    np.testing.assert_allclose(_np(out), _np(x), rtol=1e-4, atol=1e-4)
    grid = nd.GridGenerator(loc, transform_type="affine",
                            target_shape=(5, 5))
    assert grid.shape == (2, 2, 5, 5)
    samp = nd.BilinearSampler(x, grid)
    np.testing.assert_allclose(_np(samp), _np(x), rtol=1e-4, atol=1e-4)


def test_roi_pooling():
    x = nd.array(np.arange(16, dtype="float32").reshape(1, 1, 4, 4))
    rois = nd.array(np.array([[0, 0, 0, 3, 3], [0, 2, 2, 3, 3]],
                             dtype="float32"))
    out = nd.ROIPooling(x, rois, pooled_size=(2, 2), spatial_scale=1.0)
    np.testing.assert_allclose(_np(out)[0].ravel(), [5, 7, 13, 15])
    np.testing.assert_allclose(_np(out)[1].ravel(), [10, 11, 14, 15])


def test_ctc_loss():
    rs = np.random.RandomState(3)
    data = nd.array(rs.normal(size=(10, 2, 5)).astype("float32"))
    lab = nd.array(np.array([[1, 2, 0], [3, 0, 0]], dtype="float32"))
    c = nd.ctc_loss(data, lab)
    assert c.shape == (2,)
    assert np.all(_np(c) > 0)


def test_box_nms_index_copy_adamw():
    d = nd.array(np.array([[[0, 0.9, 0, 0, 10, 10],
                            [0, 0.8, 1, 1, 11, 11],
                            [0, 0.7, 50, 50, 60, 60]]], dtype="float32"))
    o = nd.box_nms(d, overlap_thresh=0.5, coord_start=2, score_index=1,
                   id_index=0)
    np.testing.assert_allclose(_np(o)[0, :, 1], [0.9, -1.0, 0.7])
    x = nd.zeros((4, 2))
    ic = nd.index_copy(x, nd.array([1, 3], dtype="int64"),
                       nd.array(np.ones((2, 2), dtype="float32")))
    np.testing.assert_allclose(_np(ic)[[1, 3]], np.ones((2, 2)))
    np.testing.assert_allclose(_np(ic)[[0, 2]], np.zeros((2, 2)))
    w = nd.array(np.ones(4, dtype="float32"))
    g = nd.array(np.full(4, 0.1, dtype="float32"))
    m, v = nd.zeros((4,)), nd.zeros((4,))
    nd.adamw_update(w, g, m, v, 1.0, lr=0.1, wd=0.01)
    assert np.all(_np(w) < 1.0)
    w32 = nd.array(np.ones(4, dtype="float32"))
    wb = w32.astype("float16")
    m2, v2 = nd.zeros((4,)), nd.zeros((4,))
    nd.mp_adamw_update(wb, g.astype("float16"), m2, v2, w32, 1.0, lr=0.1)
    assert np.all(_np(w32) < 1.0)


def test_fft_ifft_roialign():
    x = nd.array(np.array([[1.0, 2.0, 3.0, 4.0]], dtype="float32"))
    f = nd._contrib_fft(x)
    np.testing.assert_allclose(_np(f)[0],
                               [10, 0, -2, 2, -2, 0, -2, -2], atol=1e-5)
    inv = nd._contrib_ifft(f)
    np.testing.assert_allclose(_np(inv)[0] / 4, [1, 2, 3, 4], atol=1e-5)
    xx = nd.array(np.arange(16, dtype="float32").reshape(1, 1, 4, 4))
    rois = nd.array(np.array([[0, 0, 0, 3, 3]], dtype="float32"))
    ra = nd._contrib_ROIAlign(xx, rois, pooled_size=(2, 2),
                              spatial_scale=1.0)
    np.testing.assert_allclose(_np(ra).ravel(),
                               [3.75, 5.25, 9.75, 11.25], atol=1e-4)


def test_space_depth_swap_reverse():
    x = nd.array(np.random.RandomState(4)
                 .randn(2, 3, 4, 4).astype("float32"))
    s = nd.space_to_depth(x, 2)
    assert s.shape == (2, 12, 2, 2)
    d = nd.depth_to_space(s, 2)
    np.testing.assert_allclose(_np(d), _np(x))
    assert nd.swapaxes(x, 1, 3).shape == (2, 4, 4, 3)
    np.testing.assert_allclose(_np(nd.reverse(x, axis=2)),
                               _np(x)[:, :, ::-1])
    np.testing.assert_allclose(_np(nd.broadcast_like(
        nd.ones((1, 4)), nd.zeros((3, 4)))), np.ones((3, 4)))
    np.testing.assert_allclose(_np(nd.shape_array(x)), [2, 3, 4, 4])
    np.testing.assert_allclose(_np(nd.size_array(x)), [96])
    outs = nd.SliceChannel(x, 3, axis=1, squeeze_axis=True)
    assert len(outs) == 3 and outs[0].shape == (2, 4, 4)


def test_regression_outputs():
    from anand_mxnet_amd import autograd
    pred = nd.array(np.array([[1.0], [2.0]], dtype="float32"))
    pred.attach_grad()
    lab = nd.array(np.array([0.5, 0.5], dtype="float32"))
    with autograd.record():
        out = nd.LinearRegressionOutput(pred, lab)
    out.backward()
    np.testing.assert_allclose(_np(pred.grad).ravel(), [0.25, 0.75])
    pred2 = nd.array(np.array([[0.0]], dtype="float32"))
    pred2.attach_grad()
    with autograd.record():
        out = nd.LogisticRegressionOutput(pred2,
                                          nd.array(np.array([1.0],
                                                   dtype="float32")))
    np.testing.assert_allclose(_np(out), [[0.5]])
    out.backward()
    np.testing.assert_allclose(_np(pred2.grad), [[-0.5]])
    pred3 = nd.array(np.array([[2.0]], dtype="float32"))
    pred3.attach_grad()
    with autograd.record():
        out = nd.MAERegressionOutput(pred3,
                                     nd.array(np.array([0.0],
                                              dtype="float32")))
    out.backward()
    np.testing.assert_allclose(_np(pred3.grad), [[1.0]])


def test_nd_contrib_namespace():
    x = nd.array(np.array([[1.0, 2.0, 3.0, 4.0]], dtype="float32"))
    assert nd.contrib.fft(x).shape == (1, 8)
    assert abs(float(_np(nd.contrib.div_sqrt_dim(x))[0, 0]) - 0.5) < 1e-6
    import pytest as _pytest
    with _pytest.raises(AttributeError):
        nd.contrib.nonexistent_op


def test_int8_quantize_roundtrip():
    x = nd.array(np.array([0.0, 0.25, 0.5, 1.0], dtype="float32"))
    q, lo, hi = nd.quantize(x, nd.array([0.0]), nd.array([1.0]),
                            out_type="uint8")
    d = nd.dequantize(q, lo, hi)
    np.testing.assert_allclose(_np(d), _np(x), atol=1.0 / 255)
    x2 = nd.array(np.array([-2.0, 0.5, 2.0], dtype="float32"))
    q2, l2, h2 = nd.quantize_v2(x2, out_type="int8")
    d2 = nd.dequantize(q2, l2, h2)
    np.testing.assert_allclose(_np(d2), _np(x2), atol=2.0 / 127)
    assert float(_np(q2)[0]) == -127 and float(_np(q2)[2]) == 127


def test_quantized_fully_connected_pipeline():
    """quantize -> int8 FC -> dequantize recovers the float FC within
    int8 quantization error (reference quantized_fully_connected)."""
    import torch
    from anand_mxnet_amd.ops.registry import get as get_op
    rng = np.random.RandomState(0)
    x = torch.tensor(rng.uniform(-1, 1, (8, 128)).astype(np.float32))
    w = torch.tensor(rng.uniform(-1, 1, (16, 128)).astype(np.float32))
    q = get_op("quantize_v2")
    qfc = get_op("quantized_fully_connected")
    deq = get_op("dequantize")
    xq, xmin, xmax = q(x, out_type="int8")
    wq, wmin, wmax = q(w, out_type="int8")
    out32, omin, omax = qfc(xq, wq, None, xmin, xmax, wmin, wmax)
    o = out32.asnumpy()
    assert o.dtype == np.int32
    # combined-scale dequant
    sd = float(xmax.asscalar()) / 127.0
    sw = float(wmax.asscalar()) / 127.0
    f = o.astype(np.float64) * (sd * sw)
    ref = (x @ w.t()).numpy()
    err = np.abs(f - ref).max()
    assert err < 0.15, err


def test_quantized_conv_pointwise():
    import torch
    from anand_mxnet_amd.ops.registry import get as get_op
    rng = np.random.RandomState(1)
    x = torch.tensor(rng.uniform(-1, 1, (2, 4, 4, 128)).astype(np.float32))
    w = torch.tensor(rng.uniform(-1, 1, (32, 1, 1, 128)).astype(np.float32))
    q = get_op("quantize_v2")
    qc = get_op("quantized_conv")
    xq, xmin, xmax = q(x, out_type="int8")
    wq, wmin, wmax = q(w, out_type="int8")
    out32, omin, omax = qc(xq, wq, None, xmin, xmax, wmin, wmax,
                           kernel=(1, 1))
    sd = float(xmax.asscalar()) / 127.0
    sw = float(wmax.asscalar()) / 127.0
    f = out32.asnumpy().astype(np.float64) * (sd * sw)
    ref = torch.nn.functional.conv2d(
        x.permute(0, 3, 1, 2), w.permute(0, 3, 1, 2)).permute(0, 2, 3, 1)
    assert np.abs(f - ref.numpy()).max() < 0.15


def test_broadcast_ops_hypothesis_fuzz():
    """Property fuzz: nd broadcast add/mul/max against numpy across
    random broadcastable shape pairs (reference
    elemwise_binary_broadcast_op semantics)."""
    from hypothesis import given, settings, strategies as st
    import numpy as np
    import anand_mxnet_amd as mx

    @st.composite
    def bcast_pair(draw):
        ndim = draw(st.integers(1, 4))
        base = [draw(st.integers(1, 5)) for _ in range(ndim)]
        a = [d if draw(st.booleans()) else 1 for d in base]
        b = [d if (a[i] == 1 or draw(st.booleans())) else 1
             for i, d in enumerate(base)]
        return tuple(a), tuple(b)

    @settings(max_examples=30, deadline=None)
    @given(bcast_pair(), st.integers(0, 2 ** 31 - 1))
    def check(shapes, seed):
        sa, sb = shapes
        rs = np.random.RandomState(seed)
        a = (rs.rand(*sa) * 4 - 2).astype("float32")
        b = (rs.rand(*sb) * 4 - 2).astype("float32")
        na, nb = mx.nd.array(a), mx.nd.array(b)
        assert np.allclose((na + nb).asnumpy(), a + b, atol=1e-6)
        assert np.allclose((na * nb).asnumpy(), a * b, atol=1e-6)
        assert np.allclose(mx.nd.broadcast_maximum(na, nb).asnumpy(),
                           np.maximum(a, b), atol=1e-6)

    check()


def test_reduce_ops_hypothesis_fuzz():
    """Property fuzz: sum/mean/max over random axis subsets (keepdims
    both ways) vs numpy (reference broadcast_reduce_op semantics)."""
    from hypothesis import given, settings, strategies as st
    import numpy as np
    import anand_mxnet_amd as mx

    @st.composite
    def case(draw):
        ndim = draw(st.integers(1, 4))
        shape = tuple(draw(st.integers(1, 5)) for _ in range(ndim))
        axes = tuple(sorted(draw(st.sets(st.integers(0, ndim - 1),
                                         max_size=ndim))))
        keep = draw(st.booleans())
        return shape, (axes if axes else None), keep

    @settings(max_examples=30, deadline=None)
    @given(case(), st.integers(0, 2 ** 31 - 1))
    def check(c, seed):
        shape, axes, keep = c
        rs = np.random.RandomState(seed)
        a = (rs.rand(*shape) * 4 - 2).astype("float32")
        na = mx.nd.array(a)
        assert np.allclose(
            mx.nd.sum(na, axis=axes, keepdims=keep).asnumpy(),
            a.sum(axis=axes, keepdims=keep), atol=1e-5)
        assert np.allclose(
            mx.nd.mean(na, axis=axes, keepdims=keep).asnumpy(),
            a.mean(axis=axes, keepdims=keep), atol=1e-5)
        assert np.allclose(
            mx.nd.max(na, axis=axes, keepdims=keep).asnumpy(),
            a.max(axis=axes, keepdims=keep), atol=1e-6)

    check()


def test_matrix_op_hypothesis_fuzz():
    """Property fuzz: transpose/reshape/slice compositions vs numpy
    (reference matrix_op semantics — views must not alias-corrupt)."""
    from hypothesis import given, settings, strategies as st
    import numpy as np
    import anand_mxnet_amd as mx

    @settings(max_examples=25, deadline=None)
    @given(st.tuples(st.integers(1, 6), st.integers(1, 6),
                     st.integers(1, 6)),
           st.permutations([0, 1, 2]), st.integers(0, 2 ** 31 - 1))
    def check(shape, perm, seed):
        rs = np.random.RandomState(seed)
        a = (rs.rand(*shape) * 4 - 2).astype("float32")
        na = mx.nd.array(a)
        t = mx.nd.transpose(na, axes=tuple(perm))
        assert np.allclose(t.asnumpy(), a.transpose(perm))
        r = mx.nd.reshape(na, shape=(-1,))
        assert np.allclose(r.asnumpy(), a.reshape(-1))
        s0 = min(1, shape[0] - 1)
        sl = na[s0:]
        assert np.allclose(sl.asnumpy(), a[s0:])
        # slicing then mutating the slice must write through (view)
        sl[:] = 0
        assert float(na.asnumpy()[s0:].sum()) == 0.0

    check()
