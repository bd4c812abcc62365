"""Metric + loss tests (reference: test_metric.py / test_loss.py)."""
import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, metric, gluon, autograd
from anand_mxnet_amd.test_utils import assert_almost_equal


def test_accuracy():
    m = metric.Accuracy()
    pred = nd.array([[0.1, 0.9], [0.8, 0.2], [0.3, 0.7]])
    label = nd.array([1, 0, 0])
    m.update([label], [pred])
    name, acc = m.get()
    assert name == "accuracy"
    assert abs(acc - 2.0 / 3) < 1e-6


def test_topk():
    m = metric.TopKAccuracy(top_k=2)
    pred = nd.array([[0.1, 0.5, 0.4], [0.6, 0.3, 0.1]])
    label = nd.array([2, 1])
    m.update([label], [pred])
    assert abs(m.get()[1] - 1.0) < 1e-6


def test_f1_mse_rmse_mae():
    f1 = metric.F1()
    f1.update([nd.array([1, 0, 1])], [nd.array([[0.2, 0.8], [0.9, 0.1],
                                                [0.4, 0.6]])])
    assert f1.get()[1] == 1.0
    for cls, val in [(metric.MSE, 0.01), (metric.RMSE, 0.1),
                     (metric.MAE, 0.1)]:
        m = cls()
        m.update([nd.zeros((4,))], [nd.full((4,), 0.1)])
        assert abs(m.get()[1] - val) < 1e-6


def test_perplexity_ce():
    pred = nd.array([[0.5, 0.5], [0.9, 0.1]])
    label = nd.array([0, 0])
    ce = metric.CrossEntropy()
    ce.update([label], [pred])
    expected = -(np.log(0.5) + np.log(0.9)) / 2
    assert abs(ce.get()[1] - expected) < 1e-5
    p = metric.Perplexity(ignore_label=None)
    p.update([label], [pred])
    assert abs(p.get()[1] - np.exp(expected)) < 1e-4


def test_composite_and_custom():
    comp = metric.CompositeEvalMetric()
    comp.add(metric.Accuracy())
    comp.add(metric.CrossEntropy())
    pred = nd.array([[0.0, 1.0]])
    comp.update([nd.array([1])], [pred])
    names, vals = comp.get()
    assert len(names) == 2
    m = metric.np_fn(lambda l, p: float(np.sum(l == p.argmax(-1))),
                     name="mymetric")
    m.update([nd.array([1])], [pred])
    assert m.get()[1] == 1.0


def test_metric_create_by_name():
    assert isinstance(metric.create("acc"), metric.Accuracy)
    assert isinstance(metric.create(["acc", "mse"]),
                      metric.CompositeEvalMetric)


# ---------------- losses ----------------

def _check_loss_grad(loss_fn, pred_shape, label, **kw):
    pred = mx.test_utils.rand_ndarray(pred_shape)
    pred.attach_grad()
    with autograd.record():
        loss = loss_fn(pred, label)
    loss.backward()
    assert loss.shape[0] == pred_shape[0]
    assert np.all(np.isfinite(pred.grad.asnumpy()))
    return loss


def test_softmax_ce_loss():
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
    label = nd.array([1, 0, 2, 1])
    loss = _check_loss_grad(loss_fn, (4, 5), label)
    # cross-check vs numpy
    pred = nd.array(np.random.randn(4, 5).astype(np.float32))
    l = loss_fn(pred, label).asnumpy()
    p = pred.asnumpy()
    e = np.exp(p - p.max(-1, keepdims=True))
    sm = e / e.sum(-1, keepdims=True)
    ref = -np.log(sm[np.arange(4), label.asnumpy().astype(int)])
    assert_almost_equal(l, ref, rtol=1e-4, atol=1e-5)


def test_l1_l2_huber():
    label = nd.zeros((4, 3))
    for fn in [gluon.loss.L2Loss(), gluon.loss.L1Loss(),
               gluon.loss.HuberLoss()]:
        _check_loss_grad(fn, (4, 3), label)


def test_sigmoid_bce():
    label = nd.array(np.random.randint(0, 2, (4, 3)).astype(np.float32))
    loss_fn = gluon.loss.SigmoidBCELoss()
    l = _check_loss_grad(loss_fn, (4, 3), label)
    pred = nd.zeros((2, 2))
    l0 = loss_fn(pred, nd.zeros((2, 2))).asnumpy()
    assert_almost_equal(l0, np.full((2,), np.log(2)), rtol=1e-5, atol=1e-6)


def test_kl_hinge_logistic_triplet():
    _check_loss_grad(gluon.loss.KLDivLoss(),
                     (4, 5), nd.softmax(mx.test_utils.rand_ndarray((4, 5))))
    label = nd.array(np.random.choice([-1, 1], (4, 3)).astype(np.float32))
    _check_loss_grad(gluon.loss.HingeLoss(), (4, 3), label)
    _check_loss_grad(gluon.loss.SquaredHingeLoss(), (4, 3), label)
    _check_loss_grad(gluon.loss.LogisticLoss(), (4, 3), label)
    t = gluon.loss.TripletLoss()
    p = mx.test_utils.rand_ndarray((4, 8))
    pos = mx.test_utils.rand_ndarray((4, 8))
    neg = mx.test_utils.rand_ndarray((4, 8))
    p.attach_grad()
    with autograd.record():
        loss = t(p, pos, neg)
    loss.backward()
    assert np.all(np.isfinite(p.grad.asnumpy()))


def test_softmax_output_op():
    """Legacy fused softmax+CE gradient op (Module-era classifier)."""
    data = mx.test_utils.rand_ndarray((4, 10))
    label = nd.array([0, 3, 2, 9])
    data.attach_grad()
    with autograd.record():
        out = nd.SoftmaxOutput(data, label, normalization="batch")
        s = out.sum()
    s.backward()
    prob = out.asnumpy()
    oh = np.eye(10)[label.asnumpy().astype(int)]
    assert_almost_equal(data.grad.asnumpy(), (prob - oh) / 4,
                        rtol=1e-4, atol=1e-5)


def test_ctc_cosine_poisson_losses():
    from anand_mxnet_amd import gluon
    rs = np.random.RandomState(7)
    ce = gluon.loss.CosineEmbeddingLoss(margin=0.5)
    a = nd.array(np.array([[1.0, 0.0], [0.0, 1.0]], dtype="float32"))
    b = nd.array(np.array([[1.0, 0.0], [1.0, 0.0]], dtype="float32"))
    lab = nd.array(np.array([1.0, -1.0], dtype="float32"))
    out = ce(a, b, lab).asnumpy()
    np.testing.assert_allclose(out, [0.0, 0.0], atol=1e-5)  # cos=1 / cos=0
    pl = gluon.loss.PoissonNLLLoss(from_logits=True)
    v = float(pl(nd.array([[0.0]]), nd.array([[2.0]])).asnumpy())
    np.testing.assert_allclose(v, 1.0, rtol=1e-5)  # e^0 - 2*0
    ctc = gluon.loss.CTCLoss()
    pred = nd.array(rs.randn(2, 10, 5).astype("float32"))
    labels = nd.array(np.array([[1, 2, 0], [3, 0, 0]], dtype="float32"))
    out = ctc(pred, labels).asnumpy()
    assert out.shape == (2,) and np.all(out > 0)


def test_mcc_metric():
    import anand_mxnet_amd as mx
    m = mx.metric.MCC()
    # perfect binary predictions
    lab = nd.array(np.array([1, 0, 1, 0], dtype="float32"))
    pred = nd.array(np.array([[0.1, 0.9], [0.8, 0.2], [0.2, 0.8],
                              [0.7, 0.3]], dtype="float32"))
    m.update(lab, pred)
    assert abs(m.get()[1] - 1.0) < 1e-6
    m.reset()
    # anti-correlated -> -1
    m.update(nd.array(np.array([0, 1], dtype="float32")),
             nd.array(np.array([[0.1, 0.9], [0.9, 0.1]],
                               dtype="float32")))
    assert abs(m.get()[1] + 1.0) < 1e-6
    assert mx.metric.create("mcc").name == "mcc"


def test_f1_mcc_vs_sklearn():
    from sklearn.metrics import f1_score, matthews_corrcoef
    import anand_mxnet_amd as mx
    rs = np.random.RandomState(11)
    lab = rs.randint(0, 2, 64).astype("float32")
    prob = rs.rand(64, 2).astype("float32")
    pred_cls = prob.argmax(axis=1)
    f1 = mx.metric.F1()
    f1.update(nd.array(lab), nd.array(prob))
    np.testing.assert_allclose(f1.get()[1], f1_score(lab, pred_cls),
                               rtol=1e-6)
    mcc = mx.metric.MCC()
    mcc.update(nd.array(lab), nd.array(prob))
    np.testing.assert_allclose(mcc.get()[1],
                               matthews_corrcoef(lab, pred_cls),
                               rtol=1e-6, atol=1e-12)


def test_topk_perplexity_pearson():
    import anand_mxnet_amd as mx
    m = mx.metric.TopKAccuracy(top_k=2)
    lab = nd.array(np.array([0, 1, 2], dtype="float32"))
    prob = nd.array(np.array([[0.5, 0.3, 0.2],    # top2 {0,1} hit
                              [0.1, 0.2, 0.7],    # top2 {2,1} hit
                              [0.6, 0.3, 0.1]],   # top2 {0,1} miss
                             dtype="float32"))
    m.update(lab, prob)
    np.testing.assert_allclose(m.get()[1], 2.0 / 3, rtol=1e-6)
    p = mx.metric.Perplexity(ignore_label=None)
    probs = nd.array(np.array([[0.5, 0.5], [0.25, 0.75]],
                              dtype="float32"))
    p.update(nd.array(np.array([0, 1], dtype="float32")), probs)
    expect = np.exp(-(np.log(0.5) + np.log(0.75)) / 2)
    np.testing.assert_allclose(p.get()[1], expect, rtol=1e-5)
    pc = mx.metric.PearsonCorrelation()
    a = np.array([1.0, 2, 3, 4], dtype="float32")
    b = np.array([2.0, 4, 6.2, 7.9], dtype="float32")
    pc.update(nd.array(a), nd.array(b))
    np.testing.assert_allclose(pc.get()[1], np.corrcoef(a, b)[0, 1],
                               rtol=1e-4)
