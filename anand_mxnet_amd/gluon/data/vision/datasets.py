"""Vision datasets (parity with gluon/data/vision/datasets.py).

No-network environment: constructors take a local ``root`` with the raw
files already present (idx/ubyte for MNIST, pickled batches for CIFAR);
there is no download path.  SyntheticImageDataset generates random data of
ImageNet shape for benchmarking."""
from __future__ import annotations

import gzip
import os
import pickle
import struct

import numpy as np

from ..dataset import Dataset, RecordFileDataset
from ....ndarray.ndarray import NDArray, array

__all__ = ["MNIST", "FashionMNIST", "CIFAR10", "CIFAR100",
           "ImageRecordDataset", "ImageFolderDataset",
           "SyntheticImageDataset"]


def _read_idx_images(path):
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        magic, num, rows, cols = struct.unpack(">IIII", f.read(16))
        data = np.frombuffer(f.read(), dtype=np.uint8)
        return data.reshape(num, rows, cols, 1)


def _read_idx_labels(path):
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        magic, num = struct.unpack(">II", f.read(8))
        return np.frombuffer(f.read(), dtype=np.uint8).astype(np.int32)


class MNIST(Dataset):
    _files = {
        True: ("train-images-idx3-ubyte", "train-labels-idx1-ubyte"),
        False: ("t10k-images-idx3-ubyte", "t10k-labels-idx1-ubyte"),
    }

    def __init__(self, root="~/.mxnet/datasets/mnist", train=True,
                 transform=None):
        root = os.path.expanduser(root)
        img, lab = self._files[train]

        def _find(base):
            for cand in (base, base + ".gz"):
                p = os.path.join(root, cand)
                if os.path.exists(p):
                    return p
            raise FileNotFoundError(
                f"{base} not found under {root} (offline environment: "
                "place the raw MNIST idx files there)")
        self._data = _read_idx_images(_find(img))
        self._label = _read_idx_labels(_find(lab))
        self._transform = transform

    def __getitem__(self, idx):
        data = array(self._data[idx].astype(np.float32))
        label = self._label[idx]
        if self._transform is not None:
            return self._transform(data, label)
        return data, label

    def __len__(self):
        return len(self._label)


class FashionMNIST(MNIST):
    def __init__(self, root="~/.mxnet/datasets/fashion-mnist", train=True,
                 transform=None):
        super().__init__(root, train, transform)


class CIFAR10(Dataset):
    def __init__(self, root="~/.mxnet/datasets/cifar10", train=True,
                 transform=None):
        root = os.path.expanduser(root)
        files = [f"data_batch_{i}" for i in range(1, 6)] if train \
            else ["test_batch"]
        datas, labels = [], []
        for fn in files:
            p = os.path.join(root, fn)
            if not os.path.exists(p):
                p = os.path.join(root, "cifar-10-batches-py", fn)
            with open(p, "rb") as f:
                d = pickle.load(f, encoding="bytes")
            datas.append(d[b"data"].reshape(-1, 3, 32, 32)
                         .transpose(0, 2, 3, 1))
            labels.extend(d[b"labels"])
        self._data = np.concatenate(datas)
        self._label = np.asarray(labels, dtype=np.int32)
        self._transform = transform

    def __getitem__(self, idx):
        data = array(self._data[idx].astype(np.float32))
        if self._transform is not None:
            return self._transform(data, self._label[idx])
        return data, self._label[idx]

    def __len__(self):
        return len(self._label)


class CIFAR100(CIFAR10):
    def __init__(self, root="~/.mxnet/datasets/cifar100", train=True,
                 transform=None):
        root = os.path.expanduser(root)
        fn = "train" if train else "test"
        p = os.path.join(root, fn)
        if not os.path.exists(p):
            p = os.path.join(root, "cifar-100-python", fn)
        with open(p, "rb") as f:
            d = pickle.load(f, encoding="bytes")
        self._data = d[b"data"].reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)
        self._label = np.asarray(d[b"fine_labels"], dtype=np.int32)
        self._transform = transform


class ImageRecordDataset(RecordFileDataset):
    """Dataset over im2rec-packed RecordIO (image bytes + label header)."""

    def __init__(self, filename, flag=1, transform=None):
        super().__init__(filename)
        self._flag = flag
        self._transform = transform

    def __getitem__(self, idx):
        from .... import recordio
        record = super().__getitem__(idx)
        header, img_bytes = recordio.unpack(record)
        # offline env: no JPEG decoder dependency; raw-encoded records only
        arr = np.frombuffer(img_bytes, dtype=np.uint8)
        label = header.label
        if self._transform is not None:
            return self._transform(array(arr), label)
        return array(arr), label


class ImageFolderDataset(Dataset):
    def __init__(self, root, flag=1, transform=None):
        self._root = os.path.expanduser(root)
        self._transform = transform
        self.synsets = []
        self.items = []
        for folder in sorted(os.listdir(self._root)):
            path = os.path.join(self._root, folder)
            if not os.path.isdir(path):
                continue
            label = len(self.synsets)
            self.synsets.append(folder)
            for fn in sorted(os.listdir(path)):
                if fn.endswith(".npy"):
                    self.items.append((os.path.join(path, fn), label))

    def __getitem__(self, idx):
        fn, label = self.items[idx]
        img = array(np.load(fn))
        if self._transform is not None:
            return self._transform(img, label)
        return img, label

    def __len__(self):
        return len(self.items)


class SyntheticImageDataset(Dataset):
    """Random images + labels (ImageNet-shape bench data, layout NHWC)."""

    def __init__(self, length=1280, shape=(224, 224, 3), classes=1000,
                 seed=0):
        rng = np.random.RandomState(seed)
        self._img = rng.uniform(0, 1, size=shape).astype(np.float32)
        self._labels = rng.randint(0, classes, size=length).astype(np.int32)
        self._length = length

    def __getitem__(self, idx):
        return array(self._img), self._labels[idx]

    def __len__(self):
        return self._length
