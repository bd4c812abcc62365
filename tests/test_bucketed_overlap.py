"""Hook-launched bucketed all-reduce machinery (parallel/bucketed.py).

The GPU path launches collectives from post-accumulate-grad hooks as
backward fills each bucket; MXA_FORCE_OVERLAP=1 runs the *same* hook code
over gloo on CPU so ordering, rearm, no_sync and overrun handling are
covered without a GPU (reference pattern: tests/nightly dist kvstore tests
run the full dist path on localhost processes — SURVEY.md §5)."""
import os

import numpy as np
import pytest


def _spawn(target, world=2, timeout=110):
    import torch.multiprocessing as tmp
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = tmp.get_context("spawn")
    procs = [ctx.Process(target=target, args=(r, world, port))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout)
    for p in procs:
        assert p.exitcode == 0, f"worker failed: {p.exitcode}"


def _init(rank, world, port):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["MXA_FORCE_OVERLAP"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    return dist


def _overlap_worker(rank, world, port):
    dist = _init(rank, world, port)
    try:
        import torch
        from anand_mxnet_amd.parallel.bucketed import BucketedAllReducer
        torch.manual_seed(7)  # same params on both ranks
        # several buckets: small bucket_bytes forces >1
        params = [torch.randn(64, 64, requires_grad=True) for _ in range(6)]
        red = BucketedAllReducer(params, bucket_bytes=16 * 1024,
                                 average=False)
        assert len(red.buckets) > 1
        assert red.overlap
        x = torch.randn(8, 64) * (rank + 1)
        loss = sum((x @ p).sum() for p in params)
        loss.backward()
        red.finish()
        # oracle: grad[k, j] = sum over ranks of x_r.sum(0)[k]
        expect = None
        for r in range(world):
            xr = x / (rank + 1) * (r + 1)
            c = xr.sum(0)
            e = c[:, None].expand(64, 64)
            expect = e if expect is None else expect + e
        for p in params:
            np.testing.assert_allclose(p.grad.numpy(), expect.numpy(),
                                       rtol=1e-4, atol=1e-4)
        # hooks actually launched before finish (overlap worked): after
        # finish every bucket is reset
        assert all(b.work is None and b.pending == 0 for b in red.buckets)
    finally:
        dist.destroy_process_group()


def _nosync_worker(rank, world, port):
    dist = _init(rank, world, port)
    try:
        import torch
        from anand_mxnet_amd.parallel.bucketed import BucketedAllReducer
        torch.manual_seed(7)
        params = [torch.randn(32, 32, requires_grad=True) for _ in range(4)]
        red = BucketedAllReducer(params, bucket_bytes=8 * 1024,
                                 average=False)
        x1 = torch.ones(4, 32) * (rank + 1)
        x2 = torch.ones(4, 32) * (rank + 2)
        with red.no_sync():
            sum((x1 @ p).sum() for p in params).backward()
        sum((x2 @ p).sum() for p in params).backward()
        red.finish()
        # expect sum over ranks of (4*(r+1) + 4*(r+2)) in every entry
        val = sum(4.0 * (r + 1) + 4.0 * (r + 2) for r in range(world))
        for p in params:
            np.testing.assert_allclose(p.grad.numpy(),
                                       np.full((32, 32), val), rtol=1e-5)
    finally:
        dist.destroy_process_group()


def _overrun_worker(rank, world, port):
    dist = _init(rank, world, port)
    try:
        import torch
        from anand_mxnet_amd.parallel.bucketed import BucketedAllReducer
        torch.manual_seed(7)
        params = [torch.randn(16, 16, requires_grad=True) for _ in range(2)]
        red = BucketedAllReducer(params, average=False)
        x = torch.ones(2, 16)
        # two UNgated backwards: the first launches the bucket collective,
        # the second accumulates after launch -> finish must raise loudly
        sum((x @ p).sum() for p in params).backward()
        sum((x @ p).sum() for p in params).backward()
        raised = False
        try:
            red.finish()
        except RuntimeError:
            raised = True
        assert raised, "finish() must reject accumulation after launch"
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_overlap_hooks_two_procs():
    _spawn(_overlap_worker)


@pytest.mark.timeout(120)
def test_no_sync_accumulation_two_procs():
    _spawn(_nosync_worker)


@pytest.mark.timeout(120)
def test_overrun_detection_two_procs():
    _spawn(_overrun_worker)
