"""Byte-compatible ``.params`` serializer.

Reference wire format (src/ndarray/ndarray.cc — NDArray::Save/Load and
src/c_api/c_api.cc — MXNDArraySave/MXNDArrayLoad):

file := uint64 kMXAPINDArrayListMagic(0x112) | uint64 reserved(0)
        | uint64 n_arrays | n_arrays * ndarray
        | uint64 n_names  | n_names  * (uint64 len | bytes)
ndarray (V2) := uint32 0xF993FAC9 | int32 stype(-1 dense... see note)
        | shape(uint32 ndim | int64*ndim) | int32 dev_type | int32 dev_id
        | int32 dtype | raw data bytes (C-order, prod(shape)*sizeof(dtype))

TShape dims are int64 for V1/V2 (V1 magic 0xF993FAC8 is "version 1, with
int64_t TShape"); only the legacy V0 layout (magic==ndim) used uint32 dims.
ndim==0 on the wire means an EMPTY/none array (upstream semantics), so
saving a genuine 0-dim scalar raises instead of silently dropping its value.

stype: NDArrayStorageType (include/mxnet/ndarray.h): kUndefinedStorage=-1,
kDefaultStorage=0 (dense), kRowSparseStorage=1, kCSRStorage=2.  Dense writes 0.

Module checkpoints prefix names with ``arg:``/``aux:``; Gluon
``save_parameters`` uses bare names.  Loader accepts both (and V1 arrays).
"""
from __future__ import annotations

import struct

import numpy as np
import torch

from .base import MXNetError

MAGIC_LIST = 0x112
NDARRAY_V1_MAGIC = 0xF993FAC8
NDARRAY_V2_MAGIC = 0xF993FAC9

# mshadow type codes
_DTYPE_CODE = {
    torch.float32: 0, torch.float64: 1, torch.float16: 2, torch.uint8: 3,
    torch.int32: 4, torch.int8: 5, torch.int64: 6, torch.bool: 7,
    torch.bfloat16: 12,
}
_CODE_DTYPE = {v: k for k, v in _DTYPE_CODE.items()}
_NP_OF_CODE = {0: np.float32, 1: np.float64, 2: np.float16, 3: np.uint8,
               4: np.int32, 5: np.int8, 6: np.int64, 7: np.bool_}


def _tensor_of(x):
    from .ndarray.ndarray import NDArray
    if isinstance(x, NDArray):
        return x._t
    if isinstance(x, torch.Tensor):
        return x
    raise TypeError(type(x))


def _write_ndarray(out: bytearray, t: torch.Tensor):
    t = t.detach().contiguous().cpu()
    if t.dim() == 0:
        raise MXNetError(
            "cannot save a 0-dim array: on the wire ndim==0 means an "
            "empty/none array (upstream semantics) — reshape to (1,) first")
    out += struct.pack("<I", NDARRAY_V2_MAGIC)
    out += struct.pack("<i", 0)  # stype: dense (kDefaultStorage)
    out += struct.pack("<I", t.dim())
    for s in t.shape:
        out += struct.pack("<q", s)  # int64 TShape dims (V1+ layout)
    out += struct.pack("<ii", 1, 0)  # ctx: cpu(0) (reference saves via cpu copy)
    out += struct.pack("<i", _DTYPE_CODE[t.dtype])
    if t.dtype == torch.bfloat16:
        out += t.view(torch.int16).numpy().tobytes()
    elif t.dtype == torch.bool:
        out += t.numpy().tobytes()
    else:
        out += t.numpy().tobytes()


class _Reader:
    def __init__(self, buf: bytes):
        self.buf = buf
        self.pos = 0

    def take(self, n):
        b = self.buf[self.pos:self.pos + n]
        if len(b) != n:
            raise MXNetError("truncated .params file")
        self.pos += n
        return b

    def u32(self):
        return struct.unpack("<I", self.take(4))[0]

    def i32(self):
        return struct.unpack("<i", self.take(4))[0]

    def u64(self):
        return struct.unpack("<Q", self.take(8))[0]


def _read_ndarray(r: _Reader) -> torch.Tensor:
    magic = r.u32()
    legacy_v0 = False
    if magic == NDARRAY_V2_MAGIC or magic == NDARRAY_V1_MAGIC + 2:  # V2/V3
        stype = r.i32()
        if stype not in (0, -1):
            raise MXNetError("sparse arrays in .params are not supported "
                             "(dense fallback framework)")
        ndim = r.u32()
    elif magic == NDARRAY_V1_MAGIC:
        ndim = r.u32()
    else:
        # legacy V0: magic itself was ndim (TShape saved directly, uint32)
        legacy_v0 = True
        ndim = magic
        if ndim > 32:
            raise MXNetError(f"bad .params magic 0x{magic:x}")
    if legacy_v0:
        shape = [r.u32() for _ in range(ndim)]
    else:
        shape = [struct.unpack("<q", r.take(8))[0] for _ in range(ndim)]
    if ndim == 0:
        return torch.zeros(())
    r.i32()  # dev_type (ignored: arrays load to cpu, reference parity)
    r.i32()  # dev_id
    code = r.i32()
    n = 1
    for s in shape:
        n *= s
    if code == 12:  # bfloat16
        raw = np.frombuffer(r.take(n * 2), dtype=np.int16).copy()
        return torch.from_numpy(raw).view(torch.bfloat16).reshape(shape)
    npdt = _NP_OF_CODE[code]
    itemsize = np.dtype(npdt).itemsize
    raw = np.frombuffer(r.take(n * itemsize), dtype=npdt).copy()
    return torch.from_numpy(raw).reshape(shape)


def save_ndarrays(fname: str, data):
    """data: dict name->NDArray, list of NDArray, or single NDArray."""
    from .ndarray.ndarray import NDArray
    names, arrays = [], []
    if isinstance(data, dict):
        for k in data:
            names.append(k)
            arrays.append(_tensor_of(data[k]))
    elif isinstance(data, (list, tuple)):
        arrays = [_tensor_of(a) for a in data]
    elif isinstance(data, (NDArray, torch.Tensor)):
        arrays = [_tensor_of(data)]
    else:
        raise TypeError(type(data))

    out = bytearray()
    out += struct.pack("<QQ", MAGIC_LIST, 0)
    out += struct.pack("<Q", len(arrays))
    for t in arrays:
        _write_ndarray(out, t)
    out += struct.pack("<Q", len(names))
    for nm in names:
        b = nm.encode()
        out += struct.pack("<Q", len(b))
        out += b
    with open(fname, "wb") as f:
        f.write(bytes(out))


def load_ndarrays(fname: str):
    """Returns dict name->NDArray if names present, else list of NDArray."""
    from .ndarray.ndarray import NDArray
    with open(fname, "rb") as f:
        buf = f.read()
    r = _Reader(buf)
    magic = r.u64()
    if magic != MAGIC_LIST:
        raise MXNetError(f"not a .params file (magic 0x{magic:x})")
    r.u64()  # reserved
    n = r.u64()
    arrays = [_read_ndarray(r) for _ in range(n)]
    n_names = r.u64()
    names = []
    for _ in range(n_names):
        ln = r.u64()
        names.append(r.take(ln).decode())
    if names:
        if len(names) != len(arrays):
            raise MXNetError("corrupt .params: name/array count mismatch")
        return {nm: NDArray(t) for nm, t in zip(names, arrays)}
    return [NDArray(t) for t in arrays]
