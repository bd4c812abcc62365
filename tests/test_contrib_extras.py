"""gluon.contrib estimator / SyncBatchNorm / contrib.text tests
(reference parity: tests/python/unittest/test_gluon_estimator.py,
test_gluon_contrib.py, test_contrib_text.py)."""
import os

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import gluon, nd
from anand_mxnet_amd.gluon import nn


def test_estimator_fit_and_handlers(tmp_path):
    from anand_mxnet_amd.gluon.contrib.estimator import (
        Estimator, CheckpointHandler)
    np.random.seed(0)
    x = np.random.randn(200, 16).astype(np.float32)
    y = (x[:, 0] > 0).astype(np.float32)
    x[:, 1] += y * 3
    from anand_mxnet_amd.io import NDArrayIter
    it = NDArrayIter(x, y, batch_size=20, shuffle=True)

    net = nn.HybridSequential()
    net.add(nn.Dense(32, activation="relu"))
    net.add(nn.Dense(2))
    net.initialize(mx.init.Xavier())
    trainer = gluon.Trainer(net.collect_params(), "sgd",
                            {"learning_rate": 0.1})
    acc = mx.metric.Accuracy()
    est = Estimator(net, gluon.loss.SoftmaxCrossEntropyLoss(),
                    train_metrics=acc, trainer=trainer)
    ckpt = CheckpointHandler(str(tmp_path), "m")
    est.fit(it, epochs=4, event_handlers=[ckpt])
    assert acc.get()[1] > 0.8, acc.get()
    assert os.path.exists(str(tmp_path / "m-epoch0.params"))


def test_estimator_early_stopping():
    from anand_mxnet_amd.gluon.contrib.estimator import (
        Estimator, EarlyStoppingHandler)
    x = np.random.randn(40, 8).astype(np.float32)
    y = np.zeros((40,), np.float32)
    from anand_mxnet_amd.io import NDArrayIter
    it = NDArrayIter(x, y, batch_size=10)
    net = nn.Dense(2)
    net.initialize()
    est = Estimator(net, gluon.loss.SoftmaxCrossEntropyLoss())
    stop = EarlyStoppingHandler(monitor=est.train_loss_metric, patience=0,
                                mode="max")
    est.fit(it, epochs=50, event_handlers=[stop])
    # loss does not increase -> early stop long before 50 epochs
    assert stop.stop_training


def test_sync_batchnorm_single_process_matches_bn():
    from anand_mxnet_amd.gluon.contrib.nn import SyncBatchNorm
    np.random.seed(1)
    x = nd.array(np.random.randn(4, 6, 5, 5).astype(np.float32))
    sbn = SyncBatchNorm(in_channels=6)
    sbn.initialize()
    bn = nn.BatchNorm(axis=1, in_channels=6)
    bn.initialize()
    from anand_mxnet_amd import autograd
    with autograd.record():
        y1 = sbn(x)
    with autograd.record():
        y2 = bn(x)
    np.testing.assert_allclose(y1.asnumpy(), y2.asnumpy(), rtol=1e-4,
                               atol=1e-4)


def test_contrib_text_vocab_embedding(tmp_path):
    from anand_mxnet_amd.contrib.text import (Vocabulary, CustomEmbedding,
                                              utils)
    counter = utils.count_tokens_from_str("hello world hello amd mi355x "
                                          "amd amd")
    v = Vocabulary(counter, min_freq=1, reserved_tokens=["<pad>"])
    assert v.to_tokens(v.to_indices(["amd", "hello"])) == ["amd", "hello"]
    assert v.to_indices("unseen-token") == 0
    p = tmp_path / "emb.txt"
    p.write_text("amd 1.0 2.0 3.0\nhello 0.5 0.5 0.5\n")
    emb = CustomEmbedding(str(p))
    assert emb.vec_len == 3
    vec = emb.get_vecs_by_tokens("amd")
    np.testing.assert_allclose(vec.asnumpy(), [1.0, 2.0, 3.0])
    emb.update_token_vectors("amd", nd.array(np.array([[9., 9., 9.]],
                                                      dtype=np.float32)))
    np.testing.assert_allclose(emb.get_vecs_by_tokens("amd").asnumpy(),
                               [9., 9., 9.])


def _syncbn_worker(rank, world, port):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import numpy as np2
        import anand_mxnet_amd as mx2
        from anand_mxnet_amd import nd as nd2, autograd
        from anand_mxnet_amd.gluon.contrib.nn import SyncBatchNorm
        np2.random.seed(7)  # same on both ranks
        full = np2.random.randn(8, 4, 3, 3).astype(np.float32)
        part = full[rank * 4:(rank + 1) * 4]
        sbn = SyncBatchNorm(in_channels=4)
        sbn.initialize()
        with autograd.record():
            y = sbn(nd2.array(part))
        # reference: plain BN over the FULL batch
        import torch
        t = torch.from_numpy(full)
        rm = torch.zeros(4)
        rv = torch.ones(4)
        ref = torch.nn.functional.batch_norm(
            t, rm, rv, torch.ones(4), torch.zeros(4), True, 0.1, 1e-5)
        np2.testing.assert_allclose(
            y.asnumpy(), ref[rank * 4:(rank + 1) * 4].numpy(),
            rtol=1e-3, atol=1e-3)
        # running stats match the full-batch BN's
        np2.testing.assert_allclose(
            sbn.running_mean.data().asnumpy(), rm.numpy(), rtol=1e-3,
            atol=1e-3)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_sync_batchnorm_two_procs():
    import torch.multiprocessing as tmp
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = tmp.get_context("spawn")
    procs = [ctx.Process(target=_syncbn_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(110)
    for p in procs:
        assert p.exitcode == 0, f"syncbn worker failed: {p.exitcode}"


def test_svrg_module():
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.contrib.svrg_optimization import SVRGModule

    np.random.seed(0)
    data = mx.sym.var("data")
    out = mx.sym.FullyConnected(
        data=data, weight=mx.sym.var("fc_weight"),
        bias=mx.sym.var("fc_bias"), num_hidden=4, name="fc")
    out = mx.sym.SoftmaxOutput(out, mx.sym.var("softmax_label"),
                               name="softmax")
    mod = SVRGModule(out, data_names=("data",),
                     label_names=("softmax_label",), context=mx.cpu(),
                     update_freq=1)
    x = np.random.normal(size=(32, 8)).astype("float32")
    y = np.random.randint(0, 4, size=(32,)).astype("float32")
    it = mx.io.NDArrayIter(x, y, batch_size=8, label_name="softmax_label")
    mod.fit(it, num_epoch=2, optimizer="sgd",
            optimizer_params={"learning_rate": 0.1})
    assert mod._full_grads is not None and len(mod._full_grads) > 0


def test_runtime_and_tensorboard(tmp_path):
    from anand_mxnet_amd import runtime
    feats = runtime.Features()
    assert feats.is_enabled("BF16")
    from anand_mxnet_amd.contrib.tensorboard import LogMetricsCallback
    import anand_mxnet_amd as mx

    class P:
        eval_metric = mx.metric.Accuracy()
    cb = LogMetricsCallback(str(tmp_path))
    import numpy as np
    P.eval_metric.update(mx.nd.array([0, 1]), mx.nd.array(
        np.array([[0.9, 0.1], [0.2, 0.8]], dtype="float32")))
    cb(P)
