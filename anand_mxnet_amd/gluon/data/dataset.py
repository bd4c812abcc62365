"""Gluon datasets (parity with python/mxnet/gluon/data/dataset.py)."""
from __future__ import annotations

import numpy as np

from ...ndarray.ndarray import NDArray, array

__all__ = ["Dataset", "ArrayDataset", "SimpleDataset", "RecordFileDataset",
           "SyntheticDataset"]


class Dataset:
    def __getitem__(self, idx):
        raise NotImplementedError

    def __len__(self):
        raise NotImplementedError

    def transform(self, fn, lazy=True):
        trans = _LazyTransformDataset(self, fn)
        if lazy:
            return trans
        return SimpleDataset([trans[i] for i in range(len(trans))])

    def transform_first(self, fn, lazy=True):
        return self.transform(_TransformFirstClosure(fn), lazy)


class SimpleDataset(Dataset):
    def __init__(self, data):
        self._data = data

    def __len__(self):
        return len(self._data)

    def __getitem__(self, idx):
        return self._data[idx]


class _LazyTransformDataset(Dataset):
    def __init__(self, data, fn):
        self._data = data
        self._fn = fn

    def __len__(self):
        return len(self._data)

    def __getitem__(self, idx):
        item = self._data[idx]
        if isinstance(item, tuple):
            return self._fn(*item)
        return self._fn(item)


class _TransformFirstClosure:
    def __init__(self, fn):
        self._fn = fn

    def __call__(self, x, *args):
        if args:
            return (self._fn(x),) + args
        return self._fn(x)


class ArrayDataset(Dataset):
    def __init__(self, *args):
        assert len(args) > 0
        self._length = len(args[0])
        self._data = []
        for i, data in enumerate(args):
            assert len(data) == self._length, \
                f"All arrays must have the same length; got {len(data)} " \
                f"at position {i} vs {self._length}"
            if isinstance(data, NDArray) and data.ndim == 1:
                data = data.asnumpy()
            self._data.append(data)

    def __getitem__(self, idx):
        if len(self._data) == 1:
            return self._data[0][idx]
        return tuple(d[idx] for d in self._data)

    def __len__(self):
        return self._length


class RecordFileDataset(Dataset):
    """Dataset over a RecordIO file (reference recordio parity)."""

    def __init__(self, filename):
        from ... import recordio
        idx_file = filename[:-4] + ".idx" if filename.endswith(".rec") \
            else filename + ".idx"
        self._record = recordio.MXIndexedRecordIO(idx_file, filename, "r")

    def __getitem__(self, idx):
        return self._record.read_idx(self._record.keys[idx])

    def __len__(self):
        return len(self._record.keys)


class SyntheticDataset(Dataset):
    """Random-data dataset for benchmarking (no-network environment)."""

    def __init__(self, shapes, dtypes=None, length=1024, classes=None,
                 seed=0):
        self._shapes = shapes
        self._dtypes = dtypes or [np.float32] * len(shapes)
        self._length = length
        self._classes = classes
        rng = np.random.RandomState(seed)
        self._items = []
        for shape, dt in zip(shapes, self._dtypes):
            if np.issubdtype(np.dtype(dt), np.integer):
                self._items.append(rng.randint(0, classes or 10,
                                               size=shape).astype(dt))
            else:
                self._items.append(rng.uniform(-1, 1, size=shape).astype(dt))

    def __getitem__(self, idx):
        return tuple(self._items)

    def __len__(self):
        return self._length
