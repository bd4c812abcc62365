"""IO tests: NDArrayIter, RecordIO byte format, DataLoader (incl. worker
processes), datasets/transforms (reference: test_io.py, test_recordio.py,
test_gluon_data.py)."""
import os
import struct

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, recordio
from anand_mxnet_amd.io import NDArrayIter, ResizeIter, PrefetchingIter, \
    DataDesc
from anand_mxnet_amd.gluon.data import (ArrayDataset, DataLoader,
                                        SimpleDataset, SyntheticDataset)


def test_ndarray_iter():
    x = np.arange(40).reshape(10, 4).astype(np.float32)
    y = np.arange(10).astype(np.float32)
    it = NDArrayIter(x, y, batch_size=3, last_batch_handle="pad")
    batches = list(it)
    assert len(batches) == 4
    assert batches[0].data[0].shape == (3, 4)
    assert batches[-1].pad == 2
    it.reset()
    assert len(list(it)) == 4
    descs = it.provide_data
    assert isinstance(descs[0], DataDesc)
    assert descs[0].shape == (3, 4)


def test_ndarray_iter_discard_shuffle():
    x = np.arange(10).astype(np.float32)
    it = NDArrayIter(x, None, batch_size=3, shuffle=True,
                     last_batch_handle="discard")
    batches = list(it)
    assert len(batches) == 3


def test_resize_prefetch():
    x = np.arange(8).astype(np.float32)
    base = NDArrayIter(x, None, batch_size=4)
    r = ResizeIter(base, 5)
    assert len(list(r)) == 5
    base.reset()
    p = PrefetchingIter(NDArrayIter(x, None, batch_size=4))
    assert len(list(p)) == 2
    p.reset()
    assert len(list(p)) == 2


def test_recordio_roundtrip(tmp_path):
    f = str(tmp_path / "t.rec")
    w = recordio.MXRecordIO(f, "w")
    for i in range(5):
        w.write(f"record{i}".encode() * (i + 1))
    w.close()
    r = recordio.MXRecordIO(f, "r")
    for i in range(5):
        assert r.read() == f"record{i}".encode() * (i + 1)
    assert r.read() is None
    # byte-format check: magic
    raw = open(f, "rb").read()
    assert struct.unpack("<I", raw[:4])[0] == 0xCED7230A


def test_indexed_recordio(tmp_path):
    f = str(tmp_path / "t.rec")
    idx = str(tmp_path / "t.idx")
    w = recordio.MXIndexedRecordIO(idx, f, "w")
    for i in range(5):
        w.write_idx(i, f"rec{i}".encode())
    w.close()
    r = recordio.MXIndexedRecordIO(idx, f, "r")
    assert r.read_idx(3) == b"rec3"
    assert r.read_idx(0) == b"rec0"
    assert r.keys == [0, 1, 2, 3, 4]


def test_recordio_pack_unpack():
    hdr = recordio.IRHeader(0, 3.0, 7, 0)
    s = recordio.pack(hdr, b"payload")
    h2, data = recordio.unpack(s)
    assert h2.label == 3.0 and h2.id == 7 and data == b"payload"
    # float-array label
    hdr = recordio.IRHeader(0, np.array([1.0, 2.0], np.float32), 1, 0)
    s = recordio.pack(hdr, b"x")
    h2, data = recordio.unpack(s)
    assert h2.flag == 2 and list(h2.label) == [1.0, 2.0]


def test_pack_img_raw(tmp_path):
    img = np.random.randint(0, 255, (4, 6, 3)).astype(np.uint8)
    s = recordio.pack_img(recordio.IRHeader(0, 1.0, 0, 0), img)
    h, img2 = recordio.unpack_img(s)
    np.testing.assert_array_equal(img, img2)


def test_dataloader_basic():
    ds = ArrayDataset(np.arange(20).reshape(10, 2).astype(np.float32),
                      np.arange(10).astype(np.int32))
    dl = DataLoader(ds, batch_size=4, shuffle=False, last_batch="keep")
    batches = list(dl)
    assert len(batches) == 3
    data, label = batches[0]
    assert data.shape == (4, 2) and label.shape == (4,)
    assert label.asnumpy().tolist() == [0, 1, 2, 3]


def test_dataloader_workers():
    ds = ArrayDataset(np.arange(64).reshape(32, 2).astype(np.float32))
    dl = DataLoader(ds, batch_size=8, num_workers=2)
    batches = list(dl)
    assert len(batches) == 4
    total = np.concatenate([b.asnumpy() for b in batches])
    np.testing.assert_array_equal(total, np.arange(64).reshape(32, 2))


def test_dataset_transform():
    ds = SimpleDataset(list(range(8))).transform(lambda x: x * 2)
    assert ds[3] == 6
    ds2 = ArrayDataset(np.ones((4, 2), np.float32),
                       np.zeros(4, np.int32)).transform_first(
        lambda x: x + 1)
    d, l = ds2[0]
    assert float(d[0]) == 2.0


def test_synthetic_dataset():
    ds = SyntheticDataset([(3, 8, 8), ()], [np.float32, np.int64],
                          length=16, classes=10)
    img, lbl = ds[0]
    assert img.shape == (3, 8, 8)
    assert 0 <= int(lbl) < 10
    assert len(ds) == 16


def test_mnist_iter_from_raw(tmp_path):
    # synthesize tiny idx files (reference MNISTIter reads raw idx)
    imgs = np.random.randint(0, 255, (6, 28, 28)).astype(np.uint8)
    labels = np.arange(6).astype(np.uint8)
    ipath = str(tmp_path / "img-idx3-ubyte")
    lpath = str(tmp_path / "lab-idx1-ubyte")
    with open(ipath, "wb") as f:
        f.write(struct.pack(">IIII", 2051, 6, 28, 28))
        f.write(imgs.tobytes())
    with open(lpath, "wb") as f:
        f.write(struct.pack(">II", 2049, 6))
        f.write(labels.tobytes())
    from anand_mxnet_amd.io import MNISTIter
    it = MNISTIter(ipath, lpath, batch_size=2, shuffle=False)
    b = next(it)
    assert b.data[0].shape == (2, 1, 28, 28)
    assert b.label[0].asnumpy().tolist() == [0, 1]


def test_libsvm_and_imagerecord_iters(tmp_path):
    import numpy as np
    import anand_mxnet_amd as mx
    p = tmp_path / "d.libsvm"
    p.write_text("1 0:0.5 3:1.5\n0 1:2.0\n1 2:0.25 3:0.75\n")
    it = mx.io.LibSVMIter(str(p), data_shape=(4,), batch_size=3)
    b = next(iter(it))
    assert b.data[0].shape == (3, 4)
    assert float(b.data[0].asnumpy()[1, 1]) == 2.0
    # ImageRecordIter over an im2rec-style .rec
    from anand_mxnet_amd import recordio
    rec = str(tmp_path / "data.rec")
    idx = str(tmp_path / "data.idx")
    w = recordio.MXIndexedRecordIO(idx, rec, "w")
    import io as _io
    for i in range(4):
        buf = _io.BytesIO()
        np.save(buf, (np.random.rand(8, 8, 3) * 255).astype("uint8"))
        hdr = recordio.IRHeader(0, float(i % 2), i, 0)
        w.write_idx(i, recordio.pack(hdr, buf.getvalue()))
    w.close()
    rit = mx.io.ImageRecordIter(path_imgrec=rec, data_shape=(3, 8, 8),
                                batch_size=2)
    batch = next(iter(rit))
    assert batch.data[0].shape == (2, 3, 8, 8)


def test_color_transforms():
    import numpy as np
    from anand_mxnet_amd import nd
    from anand_mxnet_amd.gluon.data.vision import transforms as T
    np.random.seed(0)
    img = nd.array(np.random.rand(8, 8, 3).astype("float32"))
    for t in [T.RandomBrightness(0.5), T.RandomContrast(0.5),
              T.RandomSaturation(0.5), T.RandomHue(0.2),
              T.RandomColorJitter(0.3, 0.3, 0.3, 0.1),
              T.RandomLighting(0.1)]:
        out = t(img)
        assert out.shape == img.shape
    # hue with alpha=0 is identity
    h = T.RandomHue(0.0)
    np.testing.assert_allclose(h(img).asnumpy(), img.asnumpy(), atol=1e-5)
    # saturation of a gray image is identity
    gray = nd.array(np.full((4, 4, 3), 0.5, dtype="float32"))
    s = T.RandomSaturation(0.9)
    np.testing.assert_allclose(s(gray).asnumpy(), gray.asnumpy(),
                               atol=1e-6)


def test_image_det_iter():
    import numpy as np
    from anand_mxnet_amd.image_detection import (ImageDetIter,
                                                 DetHorizontalFlipAug)
    from anand_mxnet_amd.ndarray.ndarray import NDArray
    import torch
    rs = np.random.RandomState(0)
    items = []
    for i in range(5):
        img = rs.rand(8, 8, 3).astype("float32")
        boxes = np.array([[1, 0.1, 0.2, 0.4, 0.6],
                          [2, 0.5, 0.5, 0.9, 0.8]][:1 + i % 2],
                         dtype="float32")
        items.append((img, boxes))
    it = ImageDetIter(batch_size=2, data_shape=(3, 8, 8), imglist=items)
    b = next(iter(it))
    assert b.data[0].shape == (2, 3, 8, 8)
    assert b.label[0].shape[0] == 2 and b.label[0].shape[2] == 5
    # flip aug maps x coords correctly and deterministically at p=1
    aug = DetHorizontalFlipAug(p=1.1)
    img = NDArray(torch.rand(4, 4, 3))
    lab = np.array([[0, 0.1, 0.2, 0.4, 0.6]], dtype="float32")
    img2, lab2 = aug(img, lab)
    np.testing.assert_allclose(lab2[0], [0, 0.6, 0.2, 0.9, 0.6],
                               rtol=1e-6)


def test_interval_filter_samplers():
    from anand_mxnet_amd.gluon.data.sampler import (IntervalSampler,
                                                    FilterSampler)
    s = IntervalSampler(6, 2)
    assert list(s) == [0, 2, 4, 1, 3, 5] and len(s) == 6
    s2 = IntervalSampler(6, 2, rollover=False)
    assert list(s2) == [0, 2, 4] and len(s2) == 3
    from anand_mxnet_amd.gluon.data.dataset import SimpleDataset
    ds = SimpleDataset([1, 2, 3, 4, 5])
    f = FilterSampler(lambda v: v % 2 == 1, ds)
    assert list(f) == [0, 2, 4]
