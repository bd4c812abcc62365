"""Golden-byte tests for the .params wire format (reference
NDArray::Save/Load — survey hard requirement: byte-exact layout)."""
import struct

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd
from anand_mxnet_amd.serializer import (MAGIC_LIST, NDARRAY_V2_MAGIC,
                                        save_ndarrays, load_ndarrays)


def test_file_header_bytes(tmp_path):
    f = str(tmp_path / "x.params")
    nd.save(f, {"w": nd.zeros((2, 3))})
    raw = open(f, "rb").read()
    # uint64 list magic 0x112, uint64 reserved 0
    assert struct.unpack("<Q", raw[0:8])[0] == 0x112
    assert struct.unpack("<Q", raw[8:16])[0] == 0
    # uint64 count
    assert struct.unpack("<Q", raw[16:24])[0] == 1
    # ndarray v2 magic
    assert struct.unpack("<I", raw[24:28])[0] == 0xF993FAC9
    # stype dense = 0
    assert struct.unpack("<i", raw[28:32])[0] == 0
    # ndim=2 (uint32), then dims 2,3 as int64 (V1+ TShape is int64_t)
    assert struct.unpack("<I", raw[32:36])[0] == 2
    assert struct.unpack("<qq", raw[36:52]) == (2, 3)
    # ctx cpu(1,0), dtype f32 (0)
    assert struct.unpack("<iii", raw[52:64]) == (1, 0, 0)
    # 24 bytes of zeros data
    assert raw[64:88] == b"\x00" * 24


def test_zero_dim_save_raises(tmp_path):
    """ndim==0 on the wire means an empty array upstream; saving a genuine
    scalar must raise rather than silently drop its value."""
    import torch
    from anand_mxnet_amd.serializer import save_ndarrays
    f = str(tmp_path / "z.params")
    with pytest.raises(mx.MXNetError):
        save_ndarrays(f, {"s": nd.NDArray(torch.tensor(3.5))})


def test_roundtrip_dict(tmp_path):
    f = str(tmp_path / "d.params")
    arrays = {
        "a": nd.array(np.random.uniform(-1, 1, (3, 4)).astype(np.float32)),
        "b": nd.array(np.arange(5).astype(np.int64), dtype="int64"),
        "c": nd.ones((2,), dtype="float16"),
        "arg:fc_weight": nd.array(np.random.randn(4, 4).astype(np.float64),
                                  dtype="float64"),
    }
    save_ndarrays(f, arrays)
    loaded = load_ndarrays(f)
    assert set(loaded.keys()) == set(arrays.keys())
    for k in arrays:
        np.testing.assert_allclose(loaded[k].asnumpy(), arrays[k].asnumpy())
        assert loaded[k].dtype == arrays[k].dtype


def test_roundtrip_list(tmp_path):
    f = str(tmp_path / "l.params")
    arrays = [nd.ones((2, 2)), nd.zeros((3,))]
    save_ndarrays(f, arrays)
    loaded = load_ndarrays(f)
    assert isinstance(loaded, list) and len(loaded) == 2
    np.testing.assert_allclose(loaded[0].asnumpy(), np.ones((2, 2)))


def test_roundtrip_bf16(tmp_path):
    f = str(tmp_path / "bf.params")
    a = nd.array(np.random.uniform(-2, 2, (4, 4)).astype(np.float32))
    a16 = a.astype("bfloat16")
    save_ndarrays(f, {"w": a16})
    loaded = load_ndarrays(f)
    assert loaded["w"].dtype == "bfloat16"
    np.testing.assert_allclose(loaded["w"].asnumpy(), a16.asnumpy())


def test_scalar_and_empty(tmp_path):
    f = str(tmp_path / "s.params")
    save_ndarrays(f, {"s": nd.array(np.float32(3.5).reshape(1))})
    loaded = load_ndarrays(f)
    assert loaded["s"].asscalar() == 3.5


def test_checkpoint_prefixes(tmp_path):
    from anand_mxnet_amd.model import save_checkpoint, load_checkpoint
    prefix = str(tmp_path / "model")
    arg = {"fc1_weight": nd.ones((2, 2))}
    aux = {"bn_moving_mean": nd.zeros((2,))}
    save_checkpoint(prefix, 3, None, arg, aux)
    sym, a, x = load_checkpoint(prefix, 3)
    assert "fc1_weight" in a and "bn_moving_mean" in x
    np.testing.assert_allclose(a["fc1_weight"].asnumpy(), np.ones((2, 2)))


def test_bad_magic(tmp_path):
    f = str(tmp_path / "bad.params")
    open(f, "wb").write(b"\x00" * 32)
    with pytest.raises(mx.MXNetError):
        load_ndarrays(f)


def test_roundtrip_dtype_matrix(tmp_path):
    """Every dtype the reference serializer carries (TShape + type flag +
    raw blob) must round-trip byte-faithfully."""
    from anand_mxnet_amd.serializer import save_ndarrays, load_ndarrays
    path = str(tmp_path / "dt.params")
    arrays = {}
    for i, dt in enumerate(["float32", "float64", "float16", "bfloat16",
                            "uint8", "int32", "int64", "int8"]):
        arr = nd.array(np.arange(6).reshape(2, 3), dtype=dt)
        arrays[f"arg:w{i}_{dt}"] = arr
    save_ndarrays(path, arrays)
    back = load_ndarrays(path)
    assert set(back) == set(arrays)
    for k, v in arrays.items():
        got = back[k]
        assert got._t.dtype == v._t.dtype, k
        np.testing.assert_array_equal(
            got.astype("float64").asnumpy(),
            v.astype("float64").asnumpy())


def test_large_name_and_many_arrays(tmp_path):
    from anand_mxnet_amd.serializer import save_ndarrays, load_ndarrays
    path = str(tmp_path / "many.params")
    arrays = {f"arg:{'x' * 200}_{i}": nd.ones((1,)) for i in range(64)}
    save_ndarrays(path, arrays)
    back = load_ndarrays(path)
    assert len(back) == 64


def test_legacy_v0_read(tmp_path):
    """Legacy V0 records (magic == ndim, uint32 TShape dims) still load
    (reference ndarray.cc NDArray::Load's oldest branch)."""
    import struct
    import numpy as np
    from anand_mxnet_amd.serializer import load_ndarrays

    data = np.arange(6, dtype=np.float32).reshape(2, 3)
    buf = b""
    buf += struct.pack("<Q", 0x112)      # list magic
    buf += struct.pack("<Q", 0)          # reserved
    buf += struct.pack("<Q", 1)          # n arrays
    # V0 ndarray: ndim(=magic) | uint32 dims | dev_type | dev_id | dtype
    buf += struct.pack("<I", 2)          # ndim as magic
    buf += struct.pack("<II", 2, 3)      # uint32 dims
    buf += struct.pack("<ii", 1, 0)      # dev_type, dev_id
    buf += struct.pack("<i", 0)          # float32
    buf += data.tobytes()
    buf += struct.pack("<Q", 1)          # n names
    name = b"arg:w"
    buf += struct.pack("<Q", len(name)) + name
    p = tmp_path / "v0.params"
    p.write_bytes(buf)
    out = load_ndarrays(str(p))
    assert "arg:w" in out or "w" in out
    arr = out.get("arg:w", out.get("w"))
    assert np.allclose(arr.asnumpy(), data)


def test_serializer_hypothesis_fuzz(tmp_path):
    """Property fuzz: save/load round-trips arbitrary shapes/dtypes
    bit-exactly (hypothesis, bounded examples for suite speed)."""
    from hypothesis import given, settings, strategies as st
    import numpy as np
    import torch
    from anand_mxnet_amd.serializer import save_ndarrays, load_ndarrays
    from anand_mxnet_amd.ndarray.ndarray import NDArray

    dtypes = [np.float32, np.float64, np.float16, np.uint8, np.int32,
              np.int8, np.int64]

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.integers(1, 7), min_size=1, max_size=4),
           st.integers(0, len(dtypes) - 1),
           st.integers(0, 2 ** 31 - 1))
    def roundtrip(shape, di, seed):
        rs = np.random.RandomState(seed)
        a = (rs.rand(*shape) * 100 - 50).astype(dtypes[di])
        f = str(tmp_path / "h.params")
        save_ndarrays(f, {"x": NDArray(torch.from_numpy(a.copy()))})
        out = load_ndarrays(f)["x"].asnumpy()
        assert out.dtype == a.dtype
        assert out.shape == a.shape
        assert np.array_equal(out, a)

    roundtrip()
