import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, gluon, autograd
from anand_mxnet_amd.gluon import nn
from anand_mxnet_amd.test_utils import assert_almost_equal


def test_dense_shapes():
    net = nn.Dense(10, in_units=4)
    net.initialize()
    out = net(nd.ones((2, 4)))
    assert out.shape == (2, 10)
    assert net.weight.shape == (10, 4)
    assert net.bias.shape == (10,)


def test_deferred_init():
    net = nn.Dense(7)
    net.initialize()
    with pytest.raises(gluon.DeferredInitializationError):
        net.weight.data()
    out = net(nd.ones((3, 5)))
    assert out.shape == (3, 7)
    assert net.weight.shape == (7, 5)


def test_sequential():
    net = nn.HybridSequential()
    net.add(nn.Dense(16, activation="relu"),
            nn.Dropout(0.5),
            nn.Dense(4))
    net.initialize()
    out = net(nd.ones((2, 8)))
    assert out.shape == (2, 4)
    assert len(net) == 3
    assert isinstance(net[0], nn.Dense)


def test_conv_block_nchw():
    net = nn.Conv2D(8, kernel_size=3, padding=1, in_channels=3)
    net.initialize()
    out = net(nd.ones((2, 3, 8, 8)))
    assert out.shape == (2, 8, 8, 8)
    assert net.weight.shape == (8, 3, 3, 3)


def test_conv_block_nhwc():
    net = nn.Conv2D(8, kernel_size=3, padding=1, strides=2, layout="NHWC")
    net.initialize()
    out = net(nd.ones((2, 9, 9, 4)))
    assert out.shape == (2, 5, 5, 8)
    assert net.weight.shape == (8, 3, 3, 4)


def test_conv_nhwc_matches_nchw():
    np.random.seed(0)
    x = np.random.uniform(-1, 1, (2, 5, 7, 3)).astype(np.float32)  # NHWC
    w = np.random.uniform(-1, 1, (6, 3, 3, 3)).astype(np.float32)  # KCRS
    out_nchw = nd.Convolution(nd.array(x.transpose(0, 3, 1, 2)),
                              nd.array(w), None, kernel=(3, 3), pad=(1, 1),
                              num_filter=6, no_bias=True, layout="NCHW")
    out_nhwc = nd.Convolution(nd.array(x),
                              nd.array(w.transpose(0, 2, 3, 1)), None,
                              kernel=(3, 3), pad=(1, 1), num_filter=6,
                              no_bias=True, layout="NHWC")
    assert_almost_equal(out_nhwc.asnumpy(),
                        out_nchw.asnumpy().transpose(0, 2, 3, 1),
                        rtol=1e-4, atol=1e-5)


def test_batchnorm_train_vs_eval():
    bn = nn.BatchNorm(in_channels=4)
    bn.initialize()
    x = nd.array(np.random.uniform(-1, 1, (8, 4, 3, 3)).astype(np.float32))
    with autograd.record():
        y_train = bn(x)
    m = bn.running_mean.data().asnumpy()
    assert not np.allclose(m, 0)  # running stats updated
    y_eval = bn(x)
    assert y_train.shape == y_eval.shape


def test_pooling_blocks():
    x = nd.ones((2, 3, 8, 8))
    assert nn.MaxPool2D(2, 2)(x).shape == (2, 3, 4, 4)
    assert nn.AvgPool2D(2, 2)(x).shape == (2, 3, 4, 4)
    assert nn.GlobalAvgPool2D()(x).shape == (2, 3, 1, 1)


def test_embedding_block():
    emb = nn.Embedding(20, 8)
    emb.initialize()
    out = emb(nd.array([[1, 2], [3, 4]]))
    assert out.shape == (2, 2, 8)


def test_layernorm_block():
    ln = nn.LayerNorm(in_channels=6)
    ln.initialize()
    x = nd.array(np.random.uniform(-1, 1, (4, 6)).astype(np.float32))
    y = ln(x).asnumpy()
    assert np.allclose(y.mean(axis=-1), 0, atol=1e-5)
    assert np.allclose(y.std(axis=-1), 1, atol=1e-2)


def test_collect_params_select():
    net = nn.HybridSequential(prefix="net_")
    with net.name_scope():
        net.add(nn.Dense(4, prefix="fc1_"), nn.Dense(2, prefix="fc2_"))
    params = net.collect_params()
    assert len(params) == 4
    weights = net.collect_params(".*weight")
    assert len(weights) == 2


def test_trainer_updates_params():
    net = nn.Dense(1, in_units=1, use_bias=False)
    net.initialize(mx.init.Constant(2.0))
    trainer = gluon.Trainer(net.collect_params(), "sgd",
                            {"learning_rate": 0.1})
    x = nd.array([[1.0]])
    with autograd.record():
        loss = (net(x) ** 2).sum()
    loss.backward()
    trainer.step(1)
    w = net.weight.data().asscalar()
    # dL/dw = 2*w*x^2 = 4; w_new = 2 - 0.1*4 = 1.6
    assert abs(w - 1.6) < 1e-5


def test_cast_bf16():
    net = nn.Dense(4, in_units=3)
    net.initialize()
    net.cast("bfloat16")
    out = net(nd.ones((2, 3), dtype="bfloat16"))
    assert out.dtype == "bfloat16"


def test_save_load_parameters(tmp_path):
    net = nn.HybridSequential()
    net.add(nn.Dense(5, activation="relu"), nn.Dense(3))
    net.initialize()
    x = nd.random.uniform(shape=(2, 4))
    y1 = net(x)
    f = str(tmp_path / "net.params")
    net.save_parameters(f)
    net2 = nn.HybridSequential()
    net2.add(nn.Dense(5, activation="relu"), nn.Dense(3))
    net2.load_parameters(f)
    assert_almost_equal(net2(x).asnumpy(), y1.asnumpy())


def test_constant_param():
    class Net(gluon.HybridBlock):
        def __init__(self):
            super().__init__()
            self.c = self.params.get_constant("c", nd.array([1.0, 2.0]))

        def hybrid_forward(self, F, x, c):
            return x * c

    net = Net()
    net.initialize()
    out = net(nd.ones((2,)))
    assert_almost_equal(out.asnumpy(), [1, 2])


def test_grad_req_null():
    net = nn.Dense(2, in_units=2)
    net.initialize()
    net.weight.grad_req = "null"
    x = nd.ones((1, 2))
    with autograd.record():
        y = net(x).sum()
    y.backward()
    assert net.bias.grad(mx.cpu()) is not None


def test_lstm_cell():
    from anand_mxnet_amd.gluon import rnn
    cell = rnn.LSTMCell(10)
    cell.initialize()
    x = nd.random.uniform(shape=(4, 6))
    states = cell.begin_state(4)
    out, new_states = cell(x, states)
    assert out.shape == (4, 10)
    assert len(new_states) == 2


def test_rnn_layer_unroll():
    from anand_mxnet_amd.gluon import rnn
    lstm = rnn.LSTM(8, num_layers=2, layout="NTC")
    lstm.initialize()
    x = nd.random.uniform(shape=(3, 5, 4))  # N,T,C
    out = lstm(x)
    assert out.shape == (3, 5, 8)


def test_split_and_load():
    from anand_mxnet_amd.gluon.utils import split_and_load
    data = nd.ones((8, 2))
    parts = split_and_load(data, [mx.cpu(), mx.cpu()])
    assert len(parts) == 2 and parts[0].shape == (4, 2)


def test_clip_global_norm():
    from anand_mxnet_amd.gluon.utils import clip_global_norm
    arrays = [nd.full((2,), 3.0), nd.full((2,), 4.0)]
    norm = clip_global_norm(arrays, 1.0)
    assert abs(norm - np.sqrt(9 * 2 + 16 * 2)) < 1e-4
    total = sum((a.asnumpy() ** 2).sum() for a in arrays)
    assert abs(np.sqrt(total) - 1.0) < 1e-3


def test_conv_pool_3d_1d_family():
    from anand_mxnet_amd.gluon import nn as gnn
    rs = np.random.RandomState(2)
    x3 = nd.array(rs.randn(2, 3, 4, 5, 6).astype("float32"))
    c3 = gnn.Conv3D(4, 3, padding=1)
    c3.initialize()
    assert c3(x3).shape == (2, 4, 4, 5, 6)
    assert gnn.MaxPool3D()(x3).shape == (2, 3, 2, 2, 3)
    assert gnn.AvgPool3D()(x3).shape == (2, 3, 2, 2, 3)
    assert gnn.GlobalAvgPool3D()(x3).shape == (2, 3, 1, 1, 1)
    assert gnn.GlobalMaxPool3D()(x3).shape == (2, 3, 1, 1, 1)
    x1 = nd.array(rs.randn(2, 3, 10).astype("float32"))
    ct = gnn.Conv1DTranspose(5, 3, strides=2)
    ct.initialize()
    assert ct(x1).shape == (2, 5, 21)
    assert gnn.GlobalMaxPool1D()(x1).shape == (2, 3, 1)
    assert gnn.GlobalAvgPool1D()(x1).shape == (2, 3, 1)
    x2 = nd.array(np.arange(16, dtype="float32").reshape(1, 1, 4, 4))
    out = gnn.ReflectionPad2D(1)(x2)
    assert out.shape == (1, 1, 6, 6)
    assert float(out.asnumpy()[0, 0, 0, 0]) == 5.0  # reflected corner
    # 3D autograd flows
    from anand_mxnet_amd import autograd
    with autograd.record():
        loss = (c3(x3) ** 2).sum()
    loss.backward()
    assert c3.weight.grad() is not None


def test_new_layers_save_load_roundtrip(tmp_path):
    from anand_mxnet_amd.gluon import nn as gnn
    net = gnn.HybridSequential()
    net.add(gnn.Conv3D(4, 3, padding=1))
    net.add(gnn.Conv1DTranspose(2, 3, in_channels=0))
    rs = np.random.RandomState(1)
    x = nd.array(rs.randn(1, 2, 4, 4, 4).astype("float32"))
    net.initialize()
    # build conv3d only (shapes differ per child); save/load params
    y = net._children["0"](x)
    f = str(tmp_path / "c3.params")
    net._children["0"].save_parameters(f)
    net2 = gnn.Conv3D(4, 3, padding=1)
    net2.load_parameters(f)
    y2 = net2(x)
    np.testing.assert_allclose(y.asnumpy(), y2.asnumpy(), rtol=1e-6)


def test_model_zoo_nhwc_layouts():
    """Every CNN zoo family except inception builds and runs in NHWC (the
    MI355X performance layout) as well as NCHW."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import nd
    from anand_mxnet_amd.gluon.model_zoo import vision as zoo
    factories = [
        ("resnet18_v1", 64), ("vgg11", 64), ("alexnet", 224),
        ("mobilenet0_25", 64), ("squeezenet1_1", 64),
    ]
    for name, size in factories:
        for layout in ("NCHW", "NHWC"):
            net = getattr(zoo, name)(layout=layout)
            net.initialize(mx.init.Xavier())
            shape = (1, 3, size, size) if layout == "NCHW" \
                else (1, size, size, 3)
            y = net(nd.array(np.random.rand(*shape).astype("float32")))
            assert y.shape == (1, 1000), (name, layout)


def test_densenet_nhwc():
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import nd
    from anand_mxnet_amd.gluon.model_zoo.vision.densenet import DenseNet
    net = DenseNet(16, 8, [2, 2], classes=10, layout="NHWC")
    net.initialize(mx.init.Xavier())
    y = net(nd.array(np.random.rand(1, 56, 56, 3).astype("float32")))
    assert y.shape == (1, 10)


def test_fold_batchnorm_resnet_inference():
    """utils.fold_batchnorm: conv+BN(+relu) fold matches the unfolded
    net in inference mode (reference parity: conv+bn+relu subgraph
    fusion, SURVEY 3.2 subgraph framework)."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.gluon.model_zoo import vision as zoo
    from anand_mxnet_amd.gluon.utils import fold_batchnorm

    net = zoo.resnet18_v1(layout="NHWC")
    net.initialize(mx.init.Xavier())
    x = mx.nd.array(np.random.RandomState(2)
                    .rand(2, 32, 32, 3).astype("float32"))
    net(x)  # materialize deferred shapes
    # non-trivial running stats
    for _, p in net.collect_params().items():
        if p.name.endswith("running_mean"):
            p.set_data(mx.nd.array(
                np.random.RandomState(0).rand(p.shape[0]).astype("float32")
                - 0.5))
        if p.name.endswith("running_var"):
            p.set_data(mx.nd.array(
                0.5 + np.random.RandomState(1).rand(p.shape[0])
                .astype("float32")))
    ref = net(x).asnumpy()
    n = fold_batchnorm(net)
    # resnet18: stem BN + 2 body BNs per basic block fold; junction BNs
    # (bn_join, addend form) must remain
    assert n >= 9, n
    out = net(x).asnumpy()
    assert np.allclose(out, ref, rtol=1e-3, atol=1e-3), \
        np.abs(out - ref).max()


def test_fold_batchnorm_edge_cases():
    """fold with a biased conv folds into the existing bias; a BN whose
    axis does not match the conv layout's channel axis is skipped."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.gluon import nn
    from anand_mxnet_amd.gluon.utils import fold_batchnorm

    # biased conv: fold lands in conv.bias, outputs must match
    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Conv2D(8, 3, padding=1, layout="NHWC", in_channels=4,
                          use_bias=True))
        net.add(nn.BatchNorm(axis=3, in_channels=8))
    net.initialize(mx.init.Xavier())
    x = mx.nd.array(np.random.RandomState(1)
                    .rand(2, 6, 6, 4).astype("float32"))
    net(x)
    bn = list(net._children.values())[1]
    bn.running_mean.set_data(mx.nd.array(
        np.random.RandomState(2).rand(8).astype("float32") - 0.5))
    bn.running_var.set_data(mx.nd.array(
        0.5 + np.random.RandomState(3).rand(8).astype("float32")))
    ref = net(x).asnumpy()
    assert fold_batchnorm(net) == 1
    conv = list(net._children.values())[0]
    assert conv.bias is not None  # folded INTO the parameter
    assert getattr(conv, "_folded_bias", None) is None
    out = net(x).asnumpy()
    assert np.allclose(out, ref, atol=2e-3), np.abs(out - ref).max()

    # axis mismatch: nothing folds
    net2 = nn.HybridSequential()
    with net2.name_scope():
        net2.add(nn.Conv2D(8, 1, layout="NHWC", in_channels=4))
        net2.add(nn.BatchNorm(axis=1, in_channels=6))
    net2.initialize(mx.init.Xavier())
    assert fold_batchnorm(net2) == 0


def test_hybridize_cached_execution():
    """net.hybridize(): outputs identical before/after, cache reused
    across calls, re-specializes on a new input shape, and backward
    still flows (reference HybridBlock._build_cache -> CachedOp)."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import autograd as ag
    from anand_mxnet_amd.gluon import nn

    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Dense(8, in_units=6, activation="relu"))
        net.add(nn.Dense(3, in_units=8))
    net.initialize(mx.init.Xavier())
    x = mx.nd.array(np.random.RandomState(0).rand(4, 6).astype("float32"))
    ref = net(x).asnumpy()
    net.hybridize(static_alloc=True)
    out1 = net(x).asnumpy()
    out2 = net(x).asnumpy()  # cached call
    assert np.allclose(out1, ref, atol=1e-5)
    assert np.allclose(out2, ref, atol=1e-5)
    # new batch size: re-specialization
    x2 = mx.nd.array(np.random.RandomState(1).rand(7, 6).astype("float32"))
    assert net(x2).shape == (7, 3)
    # backward through the hybridized net
    with ag.record():
        loss = (net(x) ** 2).mean()
    loss.backward()
    g = net[0].weight.data().grad
    assert g is not None


def test_export_and_symbolblock_imports(tmp_path):
    """HybridBlock.export() -> -symbol.json + -0000.params ->
    SymbolBlock.imports reproduces the outputs (reference block.py
    export / SymbolBlock.imports deployment path)."""
    import os
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.gluon import nn, SymbolBlock

    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Dense(8, in_units=5, activation="relu"))
        net.add(nn.Dense(2, in_units=8))
    net.initialize(mx.init.Xavier())
    x = mx.nd.array(np.random.RandomState(3).rand(3, 5).astype("float32"))
    net.hybridize()
    ref = net(x).asnumpy()
    prefix = str(tmp_path / "m")
    net.export(prefix)
    assert os.path.exists(prefix + "-symbol.json")
    assert os.path.exists(prefix + "-0000.params")
    sb = SymbolBlock.imports(prefix + "-symbol.json", ["data"],
                             prefix + "-0000.params")
    out = sb(x).asnumpy()
    assert np.allclose(out, ref, atol=1e-5), np.abs(out - ref).max()


def test_export_imports_conv_bn_net(tmp_path):
    """export/imports roundtrip for a conv+BN+pool net (the trace_block
    layer set beyond Dense)."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.gluon import nn, SymbolBlock

    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Conv2D(8, 3, padding=1, in_channels=3))
        net.add(nn.BatchNorm(axis=1, in_channels=8))
        net.add(nn.Activation("relu"))
        net.add(nn.MaxPool2D(pool_size=2, strides=2))
        net.add(nn.Flatten())
        net.add(nn.Dense(4, in_units=8 * 8 * 8))
    net.initialize(mx.init.Xavier())
    x = mx.nd.array(np.random.RandomState(5)
                    .rand(2, 3, 16, 16).astype("float32"))
    net(x)  # materialize + set BN running stats shape
    ref = net(x).asnumpy()
    prefix = str(tmp_path / "cnn")
    net.export(prefix)
    sb = SymbolBlock.imports(prefix + "-symbol.json", ["data"],
                             prefix + "-0000.params")
    out = sb(x).asnumpy()
    assert np.allclose(out, ref, rtol=1e-4, atol=1e-4), \
        np.abs(out - ref).max()


def test_load_parameters_flags(tmp_path):
    """allow_missing / ignore_extra / cast_dtype contracts of
    load_parameters (reference block.py)."""
    import numpy as np
    import pytest
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.base import MXNetError
    from anand_mxnet_amd.gluon import nn

    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Dense(4, in_units=3))
        net.add(nn.Dense(2, in_units=4))
    net.initialize(mx.init.Constant(0.5))
    f = str(tmp_path / "p.params")
    net.save_parameters(f)

    # ignore_extra: a smaller net loads from the full file
    small = nn.HybridSequential()
    with small.name_scope():
        small.add(nn.Dense(4, in_units=3))
    small.initialize(mx.init.Zero())
    with pytest.raises(MXNetError):
        small.load_parameters(f)  # extra keys present
    small.load_parameters(f, ignore_extra=True)
    assert float(small[0].weight.data().asnumpy().mean()) == 0.5

    # allow_missing: a bigger net loads what exists
    big = nn.HybridSequential()
    with big.name_scope():
        big.add(nn.Dense(4, in_units=3))
        big.add(nn.Dense(2, in_units=4))
        big.add(nn.Dense(1, in_units=2))
    big.initialize(mx.init.Zero())
    with pytest.raises(MXNetError):
        big.load_parameters(f, ignore_extra=True)  # third layer missing
    big.load_parameters(f, ignore_extra=True, allow_missing=True)
    assert float(big[0].weight.data().asnumpy().mean()) == 0.5
    assert float(big[2].weight.data().asnumpy().mean()) == 0.0
