"""Module API + Symbol tests (reference: test_module.py, symbol json
round-trips)."""
import logging
import os

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, sym
from anand_mxnet_amd.io import NDArrayIter
from anand_mxnet_amd.module import Module


def _mlp_symbol():
    data = sym.var("data")
    label = sym.var("softmax_label")
    h = sym.FullyConnected(data=data, weight=sym.var("fc1_weight"),
                           bias=sym.var("fc1_bias"), num_hidden=32,
                           name="fc1")
    h = sym.Activation(h, act_type="relu")
    h = sym.FullyConnected(data=h, weight=sym.var("fc2_weight"),
                           bias=sym.var("fc2_bias"), num_hidden=10,
                           name="fc2")
    return sym.SoftmaxOutput(h, label, name="softmax", normalization="batch")


def test_symbol_infer_shape():
    s = _mlp_symbol()
    args = s.list_arguments()
    assert "data" in args and "fc1_weight" in args
    arg_shapes, out_shapes, _ = s.infer_shape(data=(8, 20),
                                              softmax_label=(8,))
    shapes = dict(zip(args, arg_shapes))
    assert shapes["fc1_weight"] == (32, 20)
    assert shapes["fc2_weight"] == (10, 32)
    assert out_shapes[0] == (8, 10)


def test_symbol_json_roundtrip(tmp_path):
    s = _mlp_symbol()
    f = str(tmp_path / "m-symbol.json")
    s.save(f)
    s2 = sym.load(f)
    assert s2.list_arguments() == s.list_arguments()
    import json
    j = json.loads(open(f).read())
    assert "nodes" in j and "arg_nodes" in j and "heads" in j


def test_symbol_eval():
    a = sym.var("a")
    b = sym.var("b")
    c = (a + b)
    out = c.eval({"a": nd.ones((2, 2)), "b": nd.ones((2, 2)) * 2})
    assert out.asnumpy().tolist() == [[3, 3], [3, 3]]


def test_simple_bind_forward_backward():
    s = _mlp_symbol()
    ex = s.simple_bind(ctx=mx.cpu(), data=(4, 20), softmax_label=(4,))
    ex.arg_dict["fc1_weight"][...] = \
        np.random.uniform(-0.1, 0.1, (32, 20)).astype(np.float32)
    ex.arg_dict["fc2_weight"][...] = \
        np.random.uniform(-0.1, 0.1, (10, 32)).astype(np.float32)
    outs = ex.forward(is_train=True, data=nd.ones((4, 20)),
                      softmax_label=nd.array([0, 1, 2, 3]))
    assert outs[0].shape == (4, 10)
    ex.backward()
    g = ex.arg_dict["fc1_weight"].grad
    assert g is not None and np.isfinite(g.asnumpy()).all()


def test_module_fit_mlp():
    np.random.seed(0)
    # learnable toy problem: 2 gaussians
    n = 200
    x = np.random.randn(n, 20).astype(np.float32)
    yl = (x[:, 0] > 0).astype(np.float32)
    x[:, 1] += yl * 2
    train = NDArrayIter(x, yl, batch_size=20, shuffle=True)
    mod = Module(_mlp_symbol(), context=mx.cpu())
    mod.fit(train, num_epoch=4, optimizer="sgd",
            optimizer_params={"learning_rate": 0.1},
            eval_metric="acc")
    score = mod.score(train, "acc")
    assert score[0][1] > 0.8, score


def test_module_save_load_checkpoint(tmp_path):
    x = np.random.randn(32, 20).astype(np.float32)
    y = np.random.randint(0, 10, (32,)).astype(np.float32)
    train = NDArrayIter(x, y, batch_size=8)
    mod = Module(_mlp_symbol(), context=mx.cpu())
    mod.bind(train.provide_data, train.provide_label)
    mod.init_params()
    prefix = str(tmp_path / "mlp")
    mod.save_checkpoint(prefix, 1)
    assert os.path.exists(prefix + "-symbol.json")
    assert os.path.exists(prefix + "-0001.params")
    mod2 = Module.load(prefix, 1)
    mod2.bind(train.provide_data, train.provide_label)
    mod2.init_from_loaded()
    a1, _ = mod.get_params()
    a2, _ = mod2.get_params()
    for k in a1:
        np.testing.assert_allclose(a1[k].asnumpy(), a2[k].asnumpy())


def test_module_predict():
    x = np.random.randn(16, 20).astype(np.float32)
    y = np.zeros((16,), np.float32)
    it = NDArrayIter(x, y, batch_size=8)
    mod = Module(_mlp_symbol(), context=mx.cpu())
    mod.bind(it.provide_data, it.provide_label, for_training=False)
    mod.init_params()
    out = mod.predict(it)
    assert out.shape == (16, 10)


def test_conv_symbol_module():
    data = sym.var("data")
    label = sym.var("softmax_label")
    h = sym.Convolution(data=data, weight=sym.var("c1_w"),
                        bias=sym.var("c1_b"), kernel=(3, 3), num_filter=8,
                        pad=(1, 1), name="c1")
    h = sym.Activation(h, act_type="relu")
    h = sym.Pooling(h, kernel=(2, 2), stride=(2, 2), pool_type="max")
    h = sym.Flatten(h)
    h = sym.FullyConnected(data=h, weight=sym.var("fc_w"),
                           bias=sym.var("fc_b"), num_hidden=10)
    s = sym.SoftmaxOutput(h, label, normalization="batch")
    ex = s.simple_bind(ctx=mx.cpu(), data=(2, 3, 8, 8), softmax_label=(2,))
    outs = ex.forward(is_train=False, data=nd.ones((2, 3, 8, 8)),
                      softmax_label=nd.zeros((2,)))
    assert outs[0].shape == (2, 10)


def test_sequential_module_train():
    from anand_mxnet_amd.module import SequentialModule
    np.random.seed(3)
    n = 400
    x = np.random.randn(n, 20).astype(np.float32)
    yl = (x[:, 0] > 0).astype(np.float32)
    x[:, 1] += yl * 2
    train = NDArrayIter(x, yl, batch_size=20, shuffle=True)

    # body module (features) -> head module (classifier + loss)
    data = sym.var("data")
    h = sym.FullyConnected(data=data, weight=sym.var("fc1_weight"),
                           bias=sym.var("fc1_bias"), num_hidden=32,
                           name="fc1")
    body_sym = sym.Activation(h, act_type="relu")

    data2 = sym.var("data")
    label = sym.var("softmax_label")
    h2 = sym.FullyConnected(data=data2, weight=sym.var("fc2_weight"),
                            bias=sym.var("fc2_bias"), num_hidden=2,
                            name="fc2")
    head_sym = sym.SoftmaxOutput(h2, label, name="softmax",
                                 normalization="batch")

    seq = SequentialModule()
    seq.add(Module(body_sym, label_names=()))
    seq.add(Module(head_sym), take_labels=True)
    seq.bind(train.provide_data, train.provide_label)
    seq.init_params()
    seq.init_optimizer(optimizer="sgd",
                       optimizer_params={"learning_rate": 0.1})

    import anand_mxnet_amd as mx
    metric = mx.metric.Accuracy()
    for _ in range(4):
        train.reset()
        metric.reset()
        for batch in train:
            seq.forward(batch, is_train=True)
            seq.backward()
            seq.update()
            seq.update_metric(metric, batch.label)
    assert metric.get()[1] > 0.8, metric.get()


def test_bucketing_module():
    """BucketingModule shares params across bucket lengths (reference:
    module/bucketing_module.py; RNN bucketing example pattern)."""
    import numpy as np
    from anand_mxnet_amd.module.bucketing_module import BucketingModule
    from anand_mxnet_amd.io import DataBatch

    def sym_gen(seq_len):
        data = sym.var("data")
        w = sym.var("fc_weight")
        b = sym.var("fc_bias")
        h = sym.FullyConnected(data=data, weight=w, bias=b, num_hidden=4,
                               name="fc")
        label = sym.var("softmax_label")
        out = sym.SoftmaxOutput(h, label, name="softmax")
        return out, ("data",), ("softmax_label",)

    bm = BucketingModule(sym_gen, default_bucket_key=8, context=mx.cpu())
    bm.bind(data_shapes=[("data", (2, 8))],
            label_shapes=[("softmax_label", (2,))])
    bm.init_params(initializer=mx.init.Xavier())
    bm.init_optimizer(optimizer="sgd",
                      optimizer_params={"learning_rate": 0.1})

    rs = np.random.RandomState(0)

    def batch(n_feat):
        b = DataBatch(data=[mx.nd.array(rs.normal(size=(2, n_feat)))],
                      label=[mx.nd.array([0.0, 1.0])],
                      provide_data=[("data", (2, n_feat))],
                      provide_label=[("softmax_label", (2,))])
        b.bucket_key = n_feat
        return b

    # default bucket
    bm.forward(batch(8), is_train=True)
    bm.backward()
    bm.update()
    out8 = bm.get_outputs()[0].asnumpy()
    assert out8.shape == (2, 4)
    # same-width second bucket shares (and continues training) the params
    bm.forward(batch(8), is_train=True)
    bm.backward()
    bm.update()
    arg, _ = bm.get_params()
    assert "fc_weight" in arg
    assert len(bm._buckets) == 1


def test_bucketing_module_switch_sync():
    """Params flow across bucket switches and the optimizer state is
    SHARED (reference: one updater over shared params)."""
    import numpy as np
    from anand_mxnet_amd.module.bucketing_module import BucketingModule
    from anand_mxnet_amd.io import DataBatch

    def sym_gen(key):
        data = sym.var("data")
        h = sym.FullyConnected(data=data, weight=sym.var("fc_weight"),
                               bias=sym.var("fc_bias"), num_hidden=4,
                               name="fc")
        return (sym.SoftmaxOutput(h, sym.var("softmax_label"),
                                  name="softmax"),
                ("data",), ("softmax_label",))

    bm = BucketingModule(sym_gen, default_bucket_key=8, context=mx.cpu())
    bm.bind(data_shapes=[("data", (2, 8))],
            label_shapes=[("softmax_label", (2,))])
    bm.init_params(initializer=mx.init.Xavier())
    bm.init_optimizer(optimizer="sgd",
                      optimizer_params={"learning_rate": 0.1,
                                        "momentum": 0.9})

    def batch(key):
        b = DataBatch(data=[mx.nd.array(np.ones((2, 8), dtype="float32"))],
                      label=[mx.nd.array([0.0, 1.0])],
                      provide_data=[("data", (2, 8))],
                      provide_label=[("softmax_label", (2,))])
        b.bucket_key = key
        return b

    bm.forward(batch(8), is_train=True)
    bm.backward()
    bm.update()
    bm.forward(batch(99), is_train=True)
    bm.backward()
    bm.update()
    assert sorted(bm._buckets) == [8, 99]
    assert bm._buckets[8]._updater is bm._buckets[99]._updater
    # switching back syncs bucket 99's training into bucket 8
    bm.forward(batch(8), is_train=True)
    a8, _ = bm._buckets[8].get_params()
    a99, _ = bm._buckets[99].get_params()
    np.testing.assert_allclose(a8["fc_weight"].asnumpy(),
                               a99["fc_weight"].asnumpy())
