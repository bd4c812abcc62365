"""RecordIO (parity with python/mxnet/recordio.py + dmlc-core recordio.h:
MXRecordIO, MXIndexedRecordIO, IRHeader, pack/unpack — byte-compatible
format: per record [kMagic u32][cflag<<29|len u32][data][pad to 4B])."""
from __future__ import annotations

import os
import struct
from collections import namedtuple

import numpy as np

from .base import MXNetError

__all__ = ["MXRecordIO", "MXIndexedRecordIO", "IRHeader", "pack", "unpack",
           "pack_img", "unpack_img"]

_kMagic = 0xCED7230A

IRHeader = namedtuple("HEADER", ["flag", "label", "id", "id2"])
_IR_FORMAT = "IfQQ"
_IR_SIZE = struct.calcsize(_IR_FORMAT)


class MXRecordIO:
    """Sequential RecordIO reader/writer."""

    def __init__(self, uri, flag):
        self.uri = uri
        self.flag = flag
        self.record = None
        self.open()

    def open(self):
        if self.flag == "w":
            self.record = open(self.uri, "wb")
            self.writable = True
        elif self.flag == "r":
            self.record = open(self.uri, "rb")
            self.writable = False
        else:
            raise ValueError("Invalid flag %s" % self.flag)
        self.pid = os.getpid()

    def close(self):
        if self.record is not None:
            self.record.close()
            self.record = None

    def __del__(self):
        self.close()

    def __getstate__(self):
        d = dict(self.__dict__)
        d["record"] = None
        return d

    def __setstate__(self, d):
        self.__dict__.update(d)
        self.open()

    def reset(self):
        self.close()
        self.open()

    def tell(self):
        return self.record.tell()

    def write(self, buf):
        assert self.writable
        ln = len(buf)
        self.record.write(struct.pack("<I", _kMagic))
        self.record.write(struct.pack("<I", ln & ((1 << 29) - 1)))
        self.record.write(buf)
        pad = (4 - ln % 4) % 4
        if pad:
            self.record.write(b"\x00" * pad)

    def read(self):
        assert not self.writable
        hdr = self.record.read(4)
        if len(hdr) < 4:
            return None
        magic, = struct.unpack("<I", hdr)
        if magic != _kMagic:
            raise MXNetError("invalid record magic")
        lrec, = struct.unpack("<I", self.record.read(4))
        ln = lrec & ((1 << 29) - 1)
        cflag = lrec >> 29
        data = self.record.read(ln)
        pad = (4 - ln % 4) % 4
        if pad:
            self.record.read(pad)
        if cflag != 0:
            raise MXNetError("multi-part records not supported")
        return data


class MXIndexedRecordIO(MXRecordIO):
    """RecordIO with a .idx sidecar for random access."""

    def __init__(self, idx_path, uri, flag, key_type=int):
        self.idx_path = idx_path
        self.idx = {}
        self.keys = []
        self.key_type = key_type
        self.fidx = None
        super().__init__(uri, flag)

    def open(self):
        super().open()
        self.idx = {}
        self.keys = []
        if self.flag == "r" and os.path.exists(self.idx_path):
            with open(self.idx_path) as f:
                for line in f:
                    parts = line.strip().split("\t")
                    key = self.key_type(parts[0])
                    self.idx[key] = int(parts[1])
                    self.keys.append(key)
        elif self.flag == "w":
            self.fidx = open(self.idx_path, "w")

    def close(self):
        if self.fidx is not None:
            self.fidx.close()
            self.fidx = None
        super().close()

    def seek(self, idx):
        assert not self.writable
        self.record.seek(self.idx[idx])

    def read_idx(self, idx):
        self.seek(idx)
        return self.read()

    def write_idx(self, idx, buf):
        key = self.key_type(idx)
        pos = self.tell()
        self.write(buf)
        self.fidx.write(f"{key}\t{pos}\n")
        self.idx[key] = pos
        self.keys.append(key)


def pack(header, s):
    header = IRHeader(*header)
    if isinstance(header.label, (int, float)):
        hdr = struct.pack(_IR_FORMAT, 0, float(header.label), header.id,
                          header.id2)
    else:
        label = np.asarray(header.label, dtype=np.float32)
        hdr = struct.pack(_IR_FORMAT, label.size, 0.0, header.id, header.id2)
        hdr += label.tobytes()
    return hdr + s


def unpack(s):
    flag, label, id_, id2 = struct.unpack(_IR_FORMAT, s[:_IR_SIZE])
    s = s[_IR_SIZE:]
    if flag > 0:
        label = np.frombuffer(s[:flag * 4], dtype=np.float32).copy()
        s = s[flag * 4:]
    header = IRHeader(flag, label, id_, id2)
    return header, s


def pack_img(header, img, quality=95, img_fmt=".jpg"):
    """Pack an image array; offline environment (no OpenCV): raw npy bytes."""
    arr = np.ascontiguousarray(img, dtype=np.uint8)
    meta = struct.pack("<III", *(arr.shape + (0,) * (3 - arr.ndim))[:3])
    return pack(header, b"MXARAW00" + meta + arr.tobytes())


def unpack_img(s, iscolor=-1):
    header, buf = unpack(s)
    if buf[:8] != b"MXARAW00":
        raise MXNetError("compressed image records need an image decoder; "
                         "this build packs raw arrays (no OpenCV, offline)")
    h, w, c = struct.unpack("<III", buf[8:20])
    arr = np.frombuffer(buf[20:], dtype=np.uint8)
    shape = [d for d in (h, w, c) if d != 0]
    return header, arr.reshape(shape)
