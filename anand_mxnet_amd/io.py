"""Data iterators (parity with python/mxnet/io/io.py: DataIter, DataBatch,
DataDesc, NDArrayIter, ResizeIter + src/io's PrefetcherIter role)."""
from __future__ import annotations

import threading
import queue
from collections import namedtuple

import numpy as np

from .base import MXNetError
from .ndarray.ndarray import NDArray, array

__all__ = ["DataDesc", "DataBatch", "DataIter", "NDArrayIter", "ResizeIter",
           "LibSVMIter", "ImageRecordIter",
           "PrefetchingIter", "MNISTIter", "CSVIter"]


class DataDesc(namedtuple("DataDesc", ["name", "shape", "dtype", "layout"])):
    def __new__(cls, name, shape, dtype=np.float32, layout="NCHW"):
        return super().__new__(cls, name, tuple(shape), dtype, layout)


class DataBatch:
    def __init__(self, data, label=None, pad=0, index=None,
                 provide_data=None, provide_label=None):
        self.data = data
        self.label = label
        self.pad = pad
        self.index = index
        self.provide_data = provide_data
        self.provide_label = provide_label


class DataIter:
    def __init__(self, batch_size=0):
        self.batch_size = batch_size

    def __iter__(self):
        return self

    def reset(self):
        pass

    def next(self):
        raise NotImplementedError

    def __next__(self):
        return self.next()

    def iter_next(self):
        try:
            self._next_batch = self.next()
            return True
        except StopIteration:
            self._next_batch = None
            return False


def _init_data(data, allow_empty, default_name):
    if data is None:
        assert allow_empty
        return []
    if isinstance(data, (np.ndarray, NDArray)):
        data = [data]
    if isinstance(data, list):
        data = dict([(default_name, d) if len(data) == 1
                     else (f"_{i}_{default_name}", d)
                     for i, d in enumerate(data)])
    if not isinstance(data, dict):
        raise TypeError("Input must be NDArray, numpy.ndarray, list or dict")
    out = []
    for k, v in data.items():
        if not isinstance(v, NDArray):
            v = np.asarray(v)
            if v.dtype == np.float64:
                v = v.astype(np.float32)
            v = array(v)
        out.append((k, v))
    return out


class NDArrayIter(DataIter):
    def __init__(self, data, label=None, batch_size=1, shuffle=False,
                 last_batch_handle="pad", data_name="data",
                 label_name="softmax_label"):
        super().__init__(batch_size)
        self.data = _init_data(data, False, data_name)
        self.label = _init_data(label, True, label_name)
        self.num_data = self.data[0][1].shape[0]
        self.shuffle = shuffle
        self.last_batch_handle = last_batch_handle
        self.cursor = -batch_size
        self._order = np.arange(self.num_data)
        if shuffle:
            np.random.shuffle(self._order)

    @property
    def provide_data(self):
        return [DataDesc(k, (self.batch_size,) + tuple(v.shape[1:]), v.dtype)
                for k, v in self.data]

    @property
    def provide_label(self):
        return [DataDesc(k, (self.batch_size,) + tuple(v.shape[1:]), v.dtype)
                for k, v in self.label]

    def reset(self):
        self.cursor = -self.batch_size
        if self.shuffle:
            np.random.shuffle(self._order)

    def next(self):
        self.cursor += self.batch_size
        if self.cursor >= self.num_data:
            raise StopIteration
        idx = self._order[self.cursor:self.cursor + self.batch_size]
        pad = 0
        if len(idx) < self.batch_size:
            if self.last_batch_handle == "discard":
                raise StopIteration
            pad = self.batch_size - len(idx)
            idx = np.concatenate([idx, self._order[:pad]])
        data = [NDArray(v._t[idx]) for _, v in self.data]
        label = [NDArray(v._t[idx]) for _, v in self.label]
        return DataBatch(data=data, label=label, pad=pad,
                         provide_data=self.provide_data,
                         provide_label=self.provide_label)


class ResizeIter(DataIter):
    def __init__(self, data_iter, size, reset_internal=True):
        super().__init__(data_iter.batch_size)
        self.data_iter = data_iter
        self.size = size
        self.reset_internal = reset_internal
        self.cur = 0

    @property
    def provide_data(self):
        return self.data_iter.provide_data

    @property
    def provide_label(self):
        return self.data_iter.provide_label

    def reset(self):
        self.cur = 0
        if self.reset_internal:
            self.data_iter.reset()

    def next(self):
        if self.cur == self.size:
            raise StopIteration
        try:
            batch = self.data_iter.next()
        except StopIteration:
            self.data_iter.reset()
            batch = self.data_iter.next()
        self.cur += 1
        return batch


class PrefetchingIter(DataIter):
    """Double-buffered prefetch on a worker thread (reference
    src/io/iter_prefetcher.h PrefetcherIter)."""

    def __init__(self, iters, rename_data=None, rename_label=None,
                 prefetch_depth=2):
        if not isinstance(iters, list):
            iters = [iters]
        assert len(iters) == 1, "single backing iter supported"
        super().__init__(iters[0].batch_size)
        self.iter = iters[0]
        self._queue = queue.Queue(maxsize=prefetch_depth)
        self._stop = threading.Event()
        self._thread = None
        self._start()

    @property
    def provide_data(self):
        return self.iter.provide_data

    @property
    def provide_label(self):
        return self.iter.provide_label

    def _start(self):
        def worker():
            while not self._stop.is_set():
                try:
                    batch = self.iter.next()
                except StopIteration:
                    self._queue.put(None)
                    return
                self._queue.put(batch)
        self._thread = threading.Thread(target=worker, daemon=True)
        self._thread.start()

    def reset(self):
        self._stop.set()
        try:
            while True:
                self._queue.get_nowait()
        except queue.Empty:
            pass
        if self._thread is not None:
            self._thread.join(timeout=5)
        self._stop.clear()
        self.iter.reset()
        self._start()

    def next(self):
        batch = self._queue.get()
        if batch is None:
            raise StopIteration
        return batch


class MNISTIter(NDArrayIter):
    """Reads raw MNIST idx files (reference src/io/iter_mnist.cc)."""

    def __init__(self, image, label, batch_size=128, shuffle=True,
                 flat=False, **kwargs):
        from .gluon.data.vision.datasets import _read_idx_images, \
            _read_idx_labels
        imgs = _read_idx_images(image).astype(np.float32) / 255.0
        labs = _read_idx_labels(label).astype(np.float32)
        if flat:
            imgs = imgs.reshape(imgs.shape[0], -1)
        else:
            imgs = imgs.transpose(0, 3, 1, 2)  # NCHW
        super().__init__(imgs, labs, batch_size, shuffle,
                         label_name="softmax_label")


class CSVIter(NDArrayIter):
    def __init__(self, data_csv, data_shape, label_csv=None, label_shape=(1,),
                 batch_size=1, **kwargs):
        data = np.loadtxt(data_csv, delimiter=",", dtype=np.float32)
        data = data.reshape((-1,) + tuple(data_shape))
        label = None
        if label_csv is not None:
            label = np.loadtxt(label_csv, delimiter=",", dtype=np.float32)
            label = label.reshape((-1,) + tuple(label_shape))
        super().__init__(data, label, batch_size, **kwargs)


class LibSVMIter(NDArrayIter):
    """LibSVM-format reader (reference: src/io/iter_libsvm.cc).

    The reference yields CSRNDArray batches; CSR math is out of scope
    here (docs/PARITY.md), so rows are densified — same iterator
    contract, dense storage."""

    def __init__(self, data_libsvm, data_shape, label_shape=(1,),
                 batch_size=1, **kwargs):
        dim = int(np.prod(data_shape))
        rows, labels = [], []
        with open(data_libsvm) as f:
            for line in f:
                parts = line.split()
                if not parts:
                    continue
                labels.append(float(parts[0]))
                row = np.zeros(dim, dtype=np.float32)
                for tok in parts[1:]:
                    k, v = tok.split(":")
                    row[int(k)] = float(v)
                rows.append(row)
        data = np.stack(rows).reshape((-1,) + tuple(data_shape))
        label = np.asarray(labels, dtype=np.float32) \
            .reshape((-1,) + tuple(label_shape))
        super().__init__(data, label, batch_size, **kwargs)


def ImageRecordIter(path_imgrec, data_shape, batch_size, shuffle=False,
                    preprocess_threads=None, rand_crop=False,
                    rand_mirror=False, mean_r=0.0, mean_g=0.0, mean_b=0.0,
                    label_width=1, **kwargs):
    """RecordIO image iterator (reference API: mx.io.ImageRecordIter from
    src/io/iter_image_recordio_2.cc).  Maps the reference's flat kwargs
    onto the Python ImageIter pipeline over the same .rec format."""
    from .image import ImageIter, CreateAugmenter
    aug = CreateAugmenter(data_shape, rand_crop=rand_crop,
                          rand_mirror=rand_mirror,
                          mean=np.array([mean_r, mean_g, mean_b])
                          if (mean_r or mean_g or mean_b) else None)
    return ImageIter(batch_size, data_shape, path_imgrec=path_imgrec,
                     shuffle=shuffle, aug_list=aug)
