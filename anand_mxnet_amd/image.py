"""Image pipeline (parity with python/mxnet/image/image.py — ImageIter +
augmenter classes).

This environment has no OpenCV/PIL; decoding covers the raw-array records
written by tools/im2rec.py (``MXARAW00``) and .npy files.  Augmenters are
tensor ops (torch interpolate / flips / crops)."""
from __future__ import annotations

import os

import numpy as np
import torch

from .base import MXNetError
from .ndarray.ndarray import NDArray, array
from .io import DataIter, DataBatch, DataDesc
from . import recordio

__all__ = ["ImageIter", "imdecode", "imresize", "CreateAugmenter",
           "Augmenter", "ResizeAug", "ForceResizeAug", "RandomCropAug",
           "CenterCropAug", "HorizontalFlipAug", "CastAug"]


def imdecode(buf, to_rgb=True, flag=1):
    """Decode an image buffer (raw MXARAW00 records or .npy bytes)."""
    if isinstance(buf, (bytes, bytearray)):
        b = bytes(buf)
        if b[:8] == b"MXARAW00":
            import struct
            h, w, c = struct.unpack("<III", b[8:20])
            arr = np.frombuffer(b[20:], dtype=np.uint8)
            shape = [d for d in (h, w, c) if d != 0]
            return array(arr.reshape(shape).astype(np.float32))
        if b[:6] == b"\x93NUMPY":
            import io as _io
            return array(np.load(_io.BytesIO(b)).astype(np.float32))
        if b[:2] == b"\xff\xd8":
            # native baseline JPEG decoder (src/kernels/jpeg.hip)
            from .base import lib
            C = lib()
            if C is not None and hasattr(C, "imdecode_jpeg"):
                t = C.imdecode_jpeg(b)  # HWC uint8 RGB
                a = t.numpy().astype(np.float32)
                if not to_rgb:
                    a = a[..., ::-1].copy()  # BGR (reference cv2 default)
                if flag == 0:
                    a = a.mean(axis=2, keepdims=True)
                return array(a)
        raise MXNetError("imdecode: unsupported image format; this build "
                         "decodes baseline JPEG natively plus raw/npy "
                         "records (docs/PARITY.md)")
    return array(np.asarray(buf, dtype=np.float32))


def imresize(src, w, h, interp=1):
    t = src._t if isinstance(src, NDArray) else torch.as_tensor(src)
    chw = t.permute(2, 0, 1).unsqueeze(0).float()
    out = torch.nn.functional.interpolate(chw, size=(h, w), mode="bilinear",
                                          align_corners=False)
    return NDArray(out.squeeze(0).permute(1, 2, 0).contiguous())


class Augmenter:
    def __init__(self, **kwargs):
        self._kwargs = kwargs

    def __call__(self, src):
        raise NotImplementedError


class ResizeAug(Augmenter):
    def __init__(self, size, interp=2):
        super().__init__(size=size)
        self.size = size

    def __call__(self, src):
        h, w = src.shape[0], src.shape[1]
        if min(h, w) == self.size:
            return src
        if h < w:
            return imresize(src, int(w * self.size / h), self.size)
        return imresize(src, self.size, int(h * self.size / w))


class ForceResizeAug(Augmenter):
    def __init__(self, size, interp=2):
        super().__init__(size=size)
        self.size = size  # (w, h)

    def __call__(self, src):
        return imresize(src, self.size[0], self.size[1])


class RandomCropAug(Augmenter):
    def __init__(self, size, interp=2):
        super().__init__(size=size)
        self.size = size if isinstance(size, tuple) else (size, size)

    def __call__(self, src):
        h, w = src.shape[0], src.shape[1]
        cw, ch = self.size
        x = np.random.randint(0, max(w - cw, 0) + 1)
        y = np.random.randint(0, max(h - ch, 0) + 1)
        return NDArray(src._t[y:y + ch, x:x + cw].contiguous())


class CenterCropAug(Augmenter):
    def __init__(self, size, interp=2):
        super().__init__(size=size)
        self.size = size if isinstance(size, tuple) else (size, size)

    def __call__(self, src):
        h, w = src.shape[0], src.shape[1]
        cw, ch = self.size
        x, y = (w - cw) // 2, (h - ch) // 2
        return NDArray(src._t[y:y + ch, x:x + cw].contiguous())


class HorizontalFlipAug(Augmenter):
    def __init__(self, p=0.5):
        super().__init__(p=p)
        self.p = p

    def __call__(self, src):
        if np.random.rand() < self.p:
            return NDArray(torch.flip(src._t, (1,)))
        return src


class CastAug(Augmenter):
    def __call__(self, src):
        return src.astype("float32")


class ColorJitterAug(Augmenter):
    def __init__(self, brightness=0, contrast=0, saturation=0):
        super().__init__(brightness=brightness, contrast=contrast,
                         saturation=saturation)
        self.b, self.c, self.s = brightness, contrast, saturation

    def __call__(self, src):
        t = src._t
        if self.b > 0:
            t = t * (1.0 + np.random.uniform(-self.b, self.b))
        if self.c > 0:
            mean = t.mean()
            t = (t - mean) * (1.0 + np.random.uniform(-self.c, self.c)) + mean
        return NDArray(t)


def CreateAugmenter(data_shape, resize=0, rand_crop=False, rand_resize=False,
                    rand_mirror=False, mean=None, std=None, brightness=0,
                    contrast=0, saturation=0, inter_method=2, **kwargs):
    """Build the standard augmenter list (reference CreateAugmenter)."""
    auglist = []
    if resize > 0:
        auglist.append(ResizeAug(resize, inter_method))
    crop_size = (data_shape[2], data_shape[1])
    if rand_crop:
        auglist.append(RandomCropAug(crop_size, inter_method))
    else:
        auglist.append(CenterCropAug(crop_size, inter_method))
    if rand_mirror:
        auglist.append(HorizontalFlipAug(0.5))
    if brightness or contrast or saturation:
        auglist.append(ColorJitterAug(brightness, contrast, saturation))
    auglist.append(CastAug())
    return auglist


class ImageIter(DataIter):
    """Image iterator over RecordIO (im2rec output) or a path list
    (reference mx.image.ImageIter)."""

    def __init__(self, batch_size, data_shape, path_imgrec=None,
                 path_imglist=None, path_root=None, shuffle=False,
                 aug_list=None, data_name="data", label_name="softmax_label",
                 **kwargs):
        super().__init__(batch_size)
        self.data_shape = tuple(data_shape)  # (C, H, W)
        self.aug_list = aug_list if aug_list is not None else \
            CreateAugmenter(data_shape, **kwargs)
        self.data_name, self.label_name = data_name, label_name
        self._items = []
        if path_imgrec:
            idx = path_imgrec[:-4] + ".idx"
            self._rec = recordio.MXIndexedRecordIO(idx, path_imgrec, "r")
            self._items = list(self._rec.keys)
            self._mode = "rec"
        elif path_imglist:
            with open(path_imglist) as f:
                for line in f:
                    parts = line.strip().split("\t")
                    self._items.append((float(parts[1]),
                                        os.path.join(path_root or "",
                                                     parts[-1])))
            self._mode = "list"
        else:
            raise MXNetError("need path_imgrec or path_imglist")
        self.shuffle = shuffle
        self.cursor = 0
        self.reset()

    @property
    def provide_data(self):
        return [DataDesc(self.data_name,
                         (self.batch_size,) + self.data_shape)]

    @property
    def provide_label(self):
        return [DataDesc(self.label_name, (self.batch_size,))]

    def reset(self):
        self.cursor = 0
        if self.shuffle:
            np.random.shuffle(self._items)

    def _read_one(self, item):
        if self._mode == "rec":
            hdr, img_bytes = recordio.unpack(self._rec.read_idx(item))
            img = imdecode(img_bytes)
            label = hdr.label if np.isscalar(hdr.label) else hdr.label[0]
        else:
            label, path = item
            img = array(np.load(path).astype(np.float32))
        for aug in self.aug_list:
            img = aug(img)
        return img, float(label)

    def next(self):
        if self.cursor >= len(self._items):
            raise StopIteration
        C, H, W = self.data_shape
        data = torch.empty((self.batch_size, C, H, W))
        label = torch.empty((self.batch_size,))
        n = 0
        while n < self.batch_size and self.cursor < len(self._items):
            img, lab = self._read_one(self._items[self.cursor])
            data[n] = img._t.permute(2, 0, 1)
            label[n] = lab
            n += 1
            self.cursor += 1
        pad = self.batch_size - n
        return DataBatch(data=[NDArray(data)], label=[NDArray(label)],
                         pad=pad, provide_data=self.provide_data,
                         provide_label=self.provide_label)
