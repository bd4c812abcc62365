"""Multi-rank GPU validation of the RCCL bucketed-overlap path.

Runs 2 ranks on the single leased MI355X (both folded onto cuda:0) so the
hook-launched bucket collectives, grad-steal <-> bucket-view interplay and
fixed launch order execute with real GPU tensors, and asserts the 2-rank
data-parallel step equals the 1-rank oracle on the concatenated global
batch (reference parity: tests/python/gpu/test_nccl.py + tests/nightly
dist kvstore — SURVEY.md §5).

RCCL (like NCCL) may refuse two ranks on one device; the workers then fall
back to gloo over CUDA tensors, which still executes the identical
hook/bucket/trainer code (the backend only changes who moves the bytes).
The chosen backend is written to backend.txt for the log."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _spawn(target, world=2, timeout=220, extra=()):
    import torch.multiprocessing as tmp
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = tmp.get_context("spawn")
    procs = [ctx.Process(target=target, args=(r, world, port) + tuple(extra))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout)
    ok = all(p.exitcode == 0 for p in procs)
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert ok, f"dist GPU worker failed: {[p.exitcode for p in procs]}"


def _init_dist(rank, world, port):
    """Init nccl (RCCL); fall back to gloo if RCCL refuses shared devices."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    backend = os.environ.get("MXA_DIST_BACKEND", "nccl")
    try:
        dist.init_process_group(backend, rank=rank, world_size=world)
        # probe the communicator: duplicate-GPU refusal surfaces here
        t = torch.ones(8, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
    except Exception:
        if dist.is_initialized():
            dist.destroy_process_group()
        os.environ["MASTER_PORT"] = str(port + 1)
        dist.init_process_group("gloo", rank=rank, world_size=world)
    return dist


def _make_net(mx, gluon):
    net = gluon.nn.HybridSequential()
    net.add(gluon.nn.Dense(64, activation="relu", in_units=32),
            gluon.nn.Dense(10, in_units=64))
    net.initialize(mx.init.Xavier(), ctx=mx.gpu(0))
    net.cast("bfloat16")
    return net


def _shard(rank):
    rng = np.random.RandomState(100 + rank)
    x = rng.uniform(-1, 1, (8, 32)).astype(np.float32)
    y = rng.randint(0, 10, (8,)).astype(np.int32)
    return x, y


def _save_init_worker(rank, world, port, outdir):
    """Record the common initial weights (what rank 0 would broadcast)."""
    torch.cuda.set_device(0)
    dist = _init_dist(rank, world, port)
    try:
        import anand_mxnet_amd as mx
        from anand_mxnet_amd import gluon
        net = _make_net(mx, gluon)
        tr = gluon.Trainer(net.collect_params(), "sgd",
                           {"learning_rate": 0.1})
        tr._init_kvstore()  # broadcasts rank-0 weights -> all ranks agree
        torch.cuda.synchronize()
        if rank == 0:
            net.save_parameters(os.path.join(outdir, "init.params"))
    finally:
        dist.destroy_process_group()


def _train_worker(rank, world, port, outdir):
    torch.cuda.set_device(0)
    dist = _init_dist(rank, world, port)
    try:
        import anand_mxnet_amd as mx
        from anand_mxnet_amd import gluon, nd, autograd
        net = _make_net(mx, gluon)
        net.load_parameters(os.path.join(outdir, "init.params"),
                            ctx=mx.gpu(0))
        tr = gluon.Trainer(net.collect_params(), "sgd",
                           {"learning_rate": 0.1, "momentum": 0.9,
                            "multi_precision": True})
        x_np, y_np = _shard(rank)  # DIFFERENT data shard per rank
        loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
        x = nd.array(x_np, ctx=mx.gpu(0), dtype="bfloat16")
        y = nd.array(y_np, ctx=mx.gpu(0))
        for _ in range(3):
            with autograd.record():
                loss = loss_fn(net(x), y)
            loss.backward()
            tr.step(8 * world)
        torch.cuda.synchronize()
        if rank == 0:
            net.save_parameters(os.path.join(outdir, "dist.params"))
            with open(os.path.join(outdir, "backend.txt"), "w") as f:
                f.write(dist.get_backend())
    finally:
        dist.destroy_process_group()


def test_two_rank_step_matches_single_rank_oracle(tmp_path):
    outdir = str(tmp_path)
    _spawn(_save_init_worker, extra=(outdir,))
    assert os.path.exists(os.path.join(outdir, "init.params"))
    _spawn(_train_worker, extra=(outdir,))
    backend = open(os.path.join(outdir, "backend.txt")).read().strip()

    # single-rank oracle: same init, concatenated global batch
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import gluon, nd, autograd
    net = _make_net(mx, gluon)
    net.load_parameters(os.path.join(outdir, "init.params"), ctx=mx.gpu(0))
    tr = gluon.Trainer(net.collect_params(), "sgd",
                       {"learning_rate": 0.1, "momentum": 0.9,
                        "multi_precision": True})
    xs, ys = zip(*[_shard(r) for r in range(2)])
    x = nd.array(np.concatenate(xs), ctx=mx.gpu(0), dtype="bfloat16")
    y = nd.array(np.concatenate(ys), ctx=mx.gpu(0))
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
    for _ in range(3):
        with autograd.record():
            loss = loss_fn(net(x), y)
        loss.backward()
        tr.step(16)
    torch.cuda.synchronize()

    net_d = _make_net(mx, gluon)
    net_d.load_parameters(os.path.join(outdir, "dist.params"),
                          ctx=mx.gpu(0))
    for (kd, pd), (ko, po) in zip(sorted(net_d.collect_params().items()),
                                  sorted(net.collect_params().items())):
        d = pd.data().asnumpy().astype(np.float32)
        o = po.data().asnumpy().astype(np.float32)
        np.testing.assert_allclose(
            d, o, rtol=2e-2, atol=2e-2,
            err_msg=f"{kd} diverged after 3 steps (backend={backend})")


def test_two_rank_allreduce_raw(tmp_path):
    """Bare BucketedAllReducer over CUDA tensors across 2 ranks."""
    _spawn(_raw_reducer_worker, extra=(str(tmp_path),))


def _raw_reducer_worker(rank, world, port, outdir):
    torch.cuda.set_device(0)
    dist = _init_dist(rank, world, port)
    try:
        from anand_mxnet_amd.parallel.bucketed import BucketedAllReducer
        torch.manual_seed(3)
        params = [torch.randn(128, 128, device="cuda:0",
                              requires_grad=True) for _ in range(5)]
        red = BucketedAllReducer(params, bucket_bytes=64 * 1024,
                                 average=False)
        assert red.overlap
        x = torch.ones(4, 128, device="cuda:0") * (rank + 1)
        sum((x @ p).sum() for p in params).backward()
        red.finish()
        torch.cuda.synchronize()
        val = sum(4.0 * (r + 1) for r in range(world))
        for p in params:
            got = p.grad.float().cpu().numpy()
            np.testing.assert_allclose(got, np.full((128, 128), val),
                                       rtol=1e-4)
    finally:
        dist.destroy_process_group()
