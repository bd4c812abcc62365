"""Gluon DataLoader (parity with python/mxnet/gluon/data/dataloader.py).

MI355X-native notes: worker->main transport uses torch shared-memory
tensors (torch.multiprocessing; the reference uses a POSIX-shm
cpu_shared_storage_manager + ForkingPickler reduce_ndarray — same
mechanism, torch's allocator provides it here).  Batches can be pinned
(pin_memory=True) so the H2D copy runs async on the engine copy stream.
"""
from __future__ import annotations

import multiprocessing as mp

import numpy as np
import torch

from ...ndarray.ndarray import NDArray, array
from .sampler import SequentialSampler, RandomSampler, BatchSampler

__all__ = ["DataLoader", "default_batchify_fn"]


def default_batchify_fn(data):
    """Stack samples into a batch NDArray."""
    if isinstance(data[0], NDArray):
        return NDArray(torch.stack([d._t for d in data]))
    if isinstance(data[0], torch.Tensor):
        return NDArray(torch.stack(list(data)))
    if isinstance(data[0], tuple):
        data = zip(*data)
        return [default_batchify_fn(i) for i in data]
    if isinstance(data[0], np.ndarray):
        return NDArray(torch.from_numpy(np.stack(data)))
    arr = np.asarray(data)
    if arr.dtype == np.float64:
        arr = arr.astype(np.float32)
    return NDArray(torch.from_numpy(arr))


def _pin(batch):
    if isinstance(batch, list):
        return [_pin(b) for b in batch]
    if isinstance(batch, NDArray):
        return NDArray(batch._t.pin_memory())
    return batch


class _WorkerFn:
    def __init__(self, dataset, batchify_fn):
        self._dataset = dataset
        self._batchify_fn = batchify_fn

    def __call__(self, samples):
        batch = self._batchify_fn([self._dataset[i] for i in samples])
        # move into torch shared memory for zero-copy transport
        def _share(b):
            if isinstance(b, list):
                return [_share(x) for x in b]
            b._t.share_memory_()
            return b
        return _share(batch)


class DataLoader:
    def __init__(self, dataset, batch_size=None, shuffle=False, sampler=None,
                 last_batch=None, batch_sampler=None, batchify_fn=None,
                 num_workers=0, pin_memory=False, prefetch=None,
                 thread_pool=False, timeout=120):
        self._dataset = dataset
        self._pin_memory = pin_memory
        self._timeout = timeout
        if batch_sampler is None:
            if batch_size is None:
                raise ValueError("batch_size must be specified unless "
                                 "batch_sampler is specified")
            if sampler is None:
                sampler = RandomSampler(len(dataset)) if shuffle \
                    else SequentialSampler(len(dataset))
            elif shuffle:
                raise ValueError("shuffle must not be specified if sampler "
                                 "is specified")
            batch_sampler = BatchSampler(sampler, batch_size,
                                         last_batch or "keep")
        elif batch_size is not None or shuffle or sampler is not None or \
                last_batch is not None:
            raise ValueError("batch_size, shuffle, sampler and last_batch "
                             "must not be specified if batch_sampler is "
                             "specified.")
        self._batch_sampler = batch_sampler
        self._num_workers = max(0, num_workers)
        self._batchify_fn = batchify_fn or default_batchify_fn
        self._prefetch = max(0, prefetch or 2 * self._num_workers)
        self._pool = None
        if self._num_workers > 0:
            ctx = mp.get_context("fork")
            self._pool = ctx.Pool(self._num_workers)

    def __iter__(self):
        if self._pool is None:
            for batch_idx in self._batch_sampler:
                batch = self._batchify_fn([self._dataset[i]
                                           for i in batch_idx])
                yield _pin(batch) if self._pin_memory else batch
            return

        fn = _WorkerFn(self._dataset, self._batchify_fn)
        it = iter(self._batch_sampler)
        pending = []
        done = False
        while True:
            while not done and len(pending) < self._prefetch + 1:
                try:
                    samples = next(it)
                except StopIteration:
                    done = True
                    break
                pending.append(self._pool.apply_async(fn, (samples,)))
            if not pending:
                break
            batch = pending.pop(0).get(self._timeout)
            yield _pin(batch) if self._pin_memory else batch

    def __len__(self):
        return len(self._batch_sampler)

    def __del__(self):
        if self._pool is not None:
            self._pool.terminate()
