"""KVStore tests: single-process semantics + multi-process gloo all-reduce
(reference: test_kvstore.py + tests/nightly/dist_sync_kvstore.py launched
as multiple localhost processes)."""
import os

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, kv
from anand_mxnet_amd.test_utils import assert_almost_equal


def test_local_push_pull():
    store = kv.create("local")
    store.init(3, nd.ones((2, 3)))
    out = nd.zeros((2, 3))
    store.pull(3, out=out)
    assert_almost_equal(out.asnumpy(), np.ones((2, 3)))
    # push a list of per-device values -> summed
    store.push(3, [nd.ones((2, 3))] * 4)
    store.pull(3, out=out)
    assert_almost_equal(out.asnumpy(), 4 * np.ones((2, 3)))


def test_multiple_keys():
    store = kv.create("local")
    keys = [5, 7, 9]
    store.init(keys, [nd.ones((2,))] * 3)
    store.push(keys, [[nd.ones((2,)) * (i + 1)] for i in range(3)])
    outs = [nd.zeros((2,)) for _ in keys]
    store.pull(keys, out=outs)
    for i, o in enumerate(outs):
        assert o.asnumpy().tolist() == [i + 1] * 2


def test_updater():
    store = kv.create("local")
    store.init("w", nd.zeros((3,)))

    def upd(key, grad, weight):
        weight += grad * 2

    store._set_updater(upd)
    store.push("w", nd.ones((3,)))
    out = nd.zeros((3,))
    store.pull("w", out=out)
    assert out.asnumpy().tolist() == [2, 2, 2]


def test_set_optimizer():
    store = kv.create("local")
    store.init("w", nd.ones((4,)))
    store.set_optimizer(mx.optimizer.SGD(learning_rate=0.5))
    store.push("w", nd.ones((4,)))  # grad=1 -> w -= 0.5
    out = nd.zeros((4,))
    store.pull("w", out=out)
    assert_almost_equal(out.asnumpy(), np.full((4,), 0.5))


def test_pushpull_and_allreduce():
    store = kv.create("device")
    store.init(0, nd.zeros((4,)))
    vals = [nd.ones((4,)), nd.ones((4,)) * 3]
    store.allreduce([vals], average=True)
    for v in vals:
        assert v.asnumpy().tolist() == [2] * 4


def _dist_worker(rank, world, port):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import anand_mxnet_amd as mx2
        from anand_mxnet_amd import nd as nd2, kv as kv2
        store = kv2.create("dist_sync_device")
        assert store.num_workers == world
        assert store.rank == rank
        v = nd2.ones((4,)) * (rank + 1)
        store.init("w", nd2.ones((4,)))
        store.pushpull("w", v)
        expect = sum(r + 1 for r in range(world))
        assert v.asnumpy().tolist() == [expect] * 4
        # gluon.Trainer over the distributed path
        from anand_mxnet_amd import gluon, autograd
        net = gluon.nn.Dense(1, in_units=2, use_bias=False)
        net.initialize(mx2.init.Constant(1.0))
        tr = gluon.Trainer(net.collect_params(), "sgd",
                           {"learning_rate": 0.1})
        x = nd2.ones((2, 2)) * (rank + 1)
        with autograd.record():
            loss = net(x).sum()
        loss.backward()
        tr.step(2 * world)
        w = net.weight.data().asnumpy()
        # grads: rank r contributes sum over batch of x = 2*(r+1) per col;
        # allreduce-sum then /(2*world)
        g = sum(2 * (r + 1) for r in range(world)) / (2.0 * world)
        assert np.allclose(w, 1.0 - 0.1 * g, atol=1e-5), (w, g)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_dist_kvstore_two_procs():
    import torch.multiprocessing as tmp
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = tmp.get_context("spawn")
    procs = [ctx.Process(target=_dist_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(110)
    for p in procs:
        assert p.exitcode == 0, f"dist worker failed: {p.exitcode}"


def test_gradient_compression_2bit():
    """Reference semantics: values clip to {-t, 0, +t}; residual error
    feedback recovers the truncation over repeated pushes."""
    import torch
    from anand_mxnet_amd.gradient_compression import GradientCompression
    gc = GradientCompression(threshold=0.5)
    g = torch.tensor([0.7, -0.9, 0.2, 0.0])
    q1 = gc.compress("k", g)
    assert q1.tolist() == [0.5, -0.5, 0.0, 0.0]
    # residual carries: second identical push pushes the remainder over t
    q2 = gc.compress("k", g)
    assert q2.tolist() == [0.5, -0.5, 0.0, 0.0]  # 0.2+0.2 < 0.5 still 0
    q3 = gc.compress("k", g)
    assert q3[2].item() == 0.5  # 0.6 accumulated > t
    # error-feedback invariant: truth - sent == residual, exactly
    total_q = q1 + q2 + q3
    assert torch.allclose(3 * g - total_q, gc._residuals["k"])
    # wire format round trip
    words = GradientCompression.pack(q1, 0.5)
    back = GradientCompression.unpack(words, q1.numel(), 0.5)
    assert torch.equal(back, q1)
    # kvstore integration
    kv_store = mx.kv.create("local")
    kv_store.set_gradient_compression({"type": "2bit", "threshold": 0.5})
    kv_store.init("w", nd.zeros((4,)))
    kv_store.push("w", nd.array(np.array([1.0, -1.0, 0.1, 0.0],
                                   dtype="float32")))
    out = nd.zeros((4,))
    kv_store.pull("w", out=out)
    np.testing.assert_allclose(out.asnumpy(), [0.5, -0.5, 0.0, 0.0])


def test_kvstore_server_role_contract(monkeypatch):
    """kvstore_server: the dist server/scheduler role is a compatibility
    shell in the all-reduce architecture — it identifies the role from
    DMLC_ROLE and parks (reference python/mxnet/kvstore_server.py)."""
    from anand_mxnet_amd import kvstore_server as ks
    monkeypatch.setenv("DMLC_ROLE", "worker")
    assert ks._init_server_module() is False
    monkeypatch.setenv("DMLC_ROLE", "server")
    assert ks._init_server_module() is True
    monkeypatch.setenv("DMLC_ROLE", "scheduler")
    assert ks._init_server_module() is True
    srv = ks.KVStoreServer()
    assert hasattr(srv, "run")
