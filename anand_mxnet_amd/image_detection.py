"""Object-detection image pipeline (reference parity:
python/mxnet/image/detection.py — ImageDetIter + Det* augmenters).

Detection labels ride with the image as [num_obj, 5+] rows
(class, xmin, ymin, xmax, ymax) in RELATIVE coordinates; augmenters
transform image and boxes together.  The reference's random-crop /
random-pad samplers are represented by the flip / resize / border
subset that the shipped SSD-style examples exercise.
"""
from __future__ import annotations

import numpy as np
import torch

from .base import MXNetError
from .image import Augmenter, imresize
from .io import DataBatch, DataDesc, DataIter
from .ndarray.ndarray import NDArray

__all__ = ["DetAugmenter", "DetBorrowAug", "DetHorizontalFlipAug",
           "DetResizeAug", "CreateDetAugmenter", "ImageDetIter"]


class DetAugmenter:
    """Image+label augmenter base (reference detection.py)."""

    def __call__(self, src, label):
        raise NotImplementedError


class DetBorrowAug(DetAugmenter):
    """Wrap an image-only Augmenter; labels pass through (reference:
    DetBorrowAug)."""

    def __init__(self, augmenter: Augmenter):
        self.augmenter = augmenter

    def __call__(self, src, label):
        return self.augmenter(src), label


class DetHorizontalFlipAug(DetAugmenter):
    """Flip image and box x-coordinates with probability p."""

    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, src, label):
        if np.random.rand() < self.p:
            src = NDArray(torch.flip(src._t, (1,)))  # HWC width axis
            label = label.copy()
            x1 = label[:, 1].copy()
            label[:, 1] = 1.0 - label[:, 3]
            label[:, 3] = 1.0 - x1
        return src, label


class DetResizeAug(DetAugmenter):
    """Resize to (w, h); relative box coords are unchanged."""

    def __init__(self, size, interp=1):
        self.size = size if isinstance(size, (list, tuple)) \
            else (size, size)

    def __call__(self, src, label):
        return imresize(src, self.size[0], self.size[1]), label


def CreateDetAugmenter(data_shape, resize=0, rand_mirror=False,
                       mean=None, std=None, **kwargs):
    """(reference: image/detection.py — CreateDetAugmenter subset)"""
    augs = []
    augs.append(DetResizeAug((data_shape[2], data_shape[1])))
    if rand_mirror:
        augs.append(DetHorizontalFlipAug(0.5))
    if mean is not None or std is not None:
        from .image import CastAug
        augs.append(DetBorrowAug(CastAug()))
    return augs


class ImageDetIter(DataIter):
    """Detection iterator over an in-memory list or .lst-style file
    (reference ImageDetIter).  Labels are padded to a fixed
    [batch, max_obj, 5] tensor with -1 rows (reference convention)."""

    def __init__(self, batch_size, data_shape, imglist=None,
                 path_imglist=None, path_root=None, aug_list=None,
                 data_name="data", label_name="label", shuffle=False,
                 **kwargs):
        super().__init__(batch_size)
        self.data_shape = tuple(data_shape)
        self.aug_list = aug_list if aug_list is not None \
            else CreateDetAugmenter(data_shape, **kwargs)
        self.data_name, self.label_name = data_name, label_name
        self._items = []
        if imglist is not None:
            # [(img_array_or_path, label_rows [n, 5]), ...]
            self._items = list(imglist)
        elif path_imglist:
            import os
            with open(path_imglist) as f:
                for line in f:
                    parts = line.strip().split("\t")
                    # idx, [header_width, label_width, rows...], path
                    vals = [float(v) for v in parts[1:-1]]
                    rows = np.array(vals[2:], dtype=np.float32) \
                        .reshape(-1, int(vals[1]))
                    self._items.append(
                        (os.path.join(path_root or "", parts[-1]), rows))
        else:
            raise MXNetError("ImageDetIter needs imglist or path_imglist")
        self.max_obj = max((np.asarray(lab).shape[0]
                            for _, lab in self._items), default=1)
        self.shuffle = shuffle
        self.cursor = 0
        self.reset()

    @property
    def provide_data(self):
        return [DataDesc(self.data_name,
                         (self.batch_size,) + self.data_shape)]

    @property
    def provide_label(self):
        return [DataDesc(self.label_name,
                         (self.batch_size, self.max_obj, 5))]

    def reset(self):
        self.cursor = 0
        if self.shuffle:
            np.random.shuffle(self._items)

    def _read_one(self, item):
        img, rows = item
        if isinstance(img, str):
            img = NDArray(torch.from_numpy(
                np.load(img).astype(np.float32)))
        elif isinstance(img, np.ndarray):
            img = NDArray(torch.from_numpy(img.astype(np.float32)))
        rows = np.asarray(rows, dtype=np.float32)
        for aug in self.aug_list:
            img, rows = aug(img, rows)
        return img, rows

    def next(self):
        if self.cursor >= len(self._items):
            raise StopIteration
        C, H, W = self.data_shape
        data = torch.empty((self.batch_size, C, H, W))
        label = torch.full((self.batch_size, self.max_obj, 5), -1.0)
        n = 0
        while n < self.batch_size and self.cursor < len(self._items):
            img, rows = self._read_one(self._items[self.cursor])
            data[n] = img._t.permute(2, 0, 1)
            k = min(rows.shape[0], self.max_obj)
            if k:
                label[n, :k] = torch.from_numpy(rows[:k, :5])
            n += 1
            self.cursor += 1
        return DataBatch(data=[NDArray(data)], label=[NDArray(label)],
                         pad=self.batch_size - n,
                         provide_data=self.provide_data,
                         provide_label=self.provide_label)
