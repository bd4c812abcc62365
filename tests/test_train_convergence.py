"""Small end-to-end convergence tests (reference: tests/python/train/
test_mlp.py / test_conv.py — a few epochs must reach an accuracy bar).

Config #1 (BASELINE.json): 'MNIST LeNet via Gluon on ctx=cpu,
KVStore=local'.  The environment is offline (no MNIST download), so the
runs use a synthetic separable image problem of MNIST shape."""
import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import gluon, nd, autograd


def _synthetic_mnist(n=512, classes=5, seed=0):
    """Images whose class is encoded as a bright patch location."""
    rng = np.random.RandomState(seed)
    x = rng.uniform(0, 0.3, (n, 1, 28, 28)).astype(np.float32)
    y = rng.randint(0, classes, n)
    for i in range(n):
        r = 4 + (y[i] // 2) * 9
        c = 4 + (y[i] % 2) * 9 + (y[i] // 4) * 5
        x[i, 0, r:r + 6, c:c + 6] += 1.0
    return x, y.astype(np.float32)


def _lenet(classes):
    net = gluon.nn.HybridSequential()
    net.add(gluon.nn.Conv2D(8, 5, activation="relu"),
            gluon.nn.MaxPool2D(2, 2),
            gluon.nn.Conv2D(16, 5, activation="relu"),
            gluon.nn.MaxPool2D(2, 2),
            gluon.nn.Flatten(),
            gluon.nn.Dense(64, activation="relu"),
            gluon.nn.Dense(classes))
    return net


def test_mnist_lenet_gluon_cpu():
    """The config-#1 plumbing smoke: LeNet + Trainer(kvstore='local')."""
    import torch
    np.random.seed(0)
    torch.manual_seed(0)  # pin init (conftest randomizes the seed)
    x, y = _synthetic_mnist()
    net = _lenet(5)
    net.initialize(mx.init.Xavier())
    trainer = gluon.Trainer(net.collect_params(), "sgd",
                            {"learning_rate": 0.05, "momentum": 0.9},
                            kvstore="local")
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
    metric = mx.metric.Accuracy()
    bs = 64
    for epoch in range(3):
        metric.reset()
        for i in range(0, len(x), bs):
            data = nd.array(x[i:i + bs])
            label = nd.array(y[i:i + bs])
            with autograd.record():
                out = net(data)
                loss = loss_fn(out, label)
            loss.backward()
            trainer.step(data.shape[0])
            metric.update([label], [out])
    assert metric.get()[1] > 0.9, metric.get()


def test_mlp_module_convergence():
    """Module.fit on a separable problem (reference test_mlp pattern)."""
    from anand_mxnet_amd import sym
    from anand_mxnet_amd.io import NDArrayIter
    from anand_mxnet_amd.module import Module
    np.random.seed(1)
    n = 400
    x = np.random.randn(n, 10).astype(np.float32)
    y = (x[:, :3].sum(1) > 0).astype(np.float32)
    data = sym.var("data")
    label = sym.var("softmax_label")
    h = sym.FullyConnected(data=data, weight=sym.var("w1"),
                           bias=sym.var("b1"), num_hidden=16)
    h = sym.Activation(h, act_type="relu")
    h = sym.FullyConnected(data=h, weight=sym.var("w2"),
                           bias=sym.var("b2"), num_hidden=2)
    s = sym.SoftmaxOutput(h, label, normalization="batch")
    it = NDArrayIter(x, y, batch_size=40, shuffle=True)
    mod = Module(s)
    mod.fit(it, num_epoch=10, optimizer="sgd",
            optimizer_params={"learning_rate": 0.5})
    acc = mod.score(it, "acc")[0][1]
    assert acc > 0.9, acc
