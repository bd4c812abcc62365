"""Native baseline JPEG decoder (src/kernels/jpeg.hip) vs a from-scratch
encoder built here from the JPEG spec (Annex K tables, scipy DCT) — the
environment ships no image library, so the test constructs its own
bitstreams.  Reference parity: the OpenCV decode inside
iter_image_recordio_2.cc."""
import struct

import numpy as np
import pytest
from scipy.fftpack import dct

from anand_mxnet_amd.base import lib

_C = lib()
pytestmark = pytest.mark.skipif(
    _C is None or not hasattr(_C, "imdecode_jpeg"),
    reason="native extension not built")

# Annex K quantization tables (quality ~50)
QY = np.array([
    16, 11, 10, 16, 24, 40, 51, 61, 12, 12, 14, 19, 26, 58, 60, 55,
    14, 13, 16, 24, 40, 57, 69, 56, 14, 17, 22, 29, 51, 87, 80, 62,
    18, 22, 37, 56, 68, 109, 103, 77, 24, 35, 55, 64, 81, 104, 113, 92,
    49, 64, 78, 87, 103, 121, 120, 101, 72, 92, 95, 98, 112, 100, 103, 99,
]).reshape(8, 8)
QC = np.array([
    17, 18, 24, 47, 99, 99, 99, 99, 18, 21, 26, 66, 99, 99, 99, 99,
    24, 26, 56, 99, 99, 99, 99, 99, 47, 66, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99,
]).reshape(8, 8)

# Annex K Huffman table specs: (bits[1..16], values)
DC_LUM = ([0, 1, 5, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0],
          list(range(12)))
DC_CHR = ([0, 3, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0],
          list(range(12)))
AC_LUM_BITS = [0, 2, 1, 3, 3, 2, 4, 3, 5, 5, 4, 4, 0, 0, 1, 0x7D]
AC_LUM_VALS = [
    0x01, 0x02, 0x03, 0x00, 0x04, 0x11, 0x05, 0x12, 0x21, 0x31, 0x41,
    0x06, 0x13, 0x51, 0x61, 0x07, 0x22, 0x71, 0x14, 0x32, 0x81, 0x91,
    0xA1, 0x08, 0x23, 0x42, 0xB1, 0xC1, 0x15, 0x52, 0xD1, 0xF0, 0x24,
    0x33, 0x62, 0x72, 0x82, 0x09, 0x0A, 0x16, 0x17, 0x18, 0x19, 0x1A,
    0x25, 0x26, 0x27, 0x28, 0x29, 0x2A, 0x34, 0x35, 0x36, 0x37, 0x38,
    0x39, 0x3A, 0x43, 0x44, 0x45, 0x46, 0x47, 0x48, 0x49, 0x4A, 0x53,
    0x54, 0x55, 0x56, 0x57, 0x58, 0x59, 0x5A, 0x63, 0x64, 0x65, 0x66,
    0x67, 0x68, 0x69, 0x6A, 0x73, 0x74, 0x75, 0x76, 0x77, 0x78, 0x79,
    0x7A, 0x83, 0x84, 0x85, 0x86, 0x87, 0x88, 0x89, 0x8A, 0x92, 0x93,
    0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9A, 0xA2, 0xA3, 0xA4, 0xA5,
    0xA6, 0xA7, 0xA8, 0xA9, 0xAA, 0xB2, 0xB3, 0xB4, 0xB5, 0xB6, 0xB7,
    0xB8, 0xB9, 0xBA, 0xC2, 0xC3, 0xC4, 0xC5, 0xC6, 0xC7, 0xC8, 0xC9,
    0xCA, 0xD2, 0xD3, 0xD4, 0xD5, 0xD6, 0xD7, 0xD8, 0xD9, 0xDA, 0xE1,
    0xE2, 0xE3, 0xE4, 0xE5, 0xE6, 0xE7, 0xE8, 0xE9, 0xEA, 0xF1, 0xF2,
    0xF3, 0xF4, 0xF5, 0xF6, 0xF7, 0xF8, 0xF9, 0xFA,
]
ZIG = np.array([
    0, 1, 8, 16, 9, 2, 3, 10, 17, 24, 32, 25, 18, 11, 4, 5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6, 7, 14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63])


def _huff_codes(bits, vals):
    codes = {}
    code = 0
    k = 0
    for l in range(1, 17):
        for _ in range(bits[l - 1]):
            codes[vals[k]] = (code, l)
            code += 1
            k += 1
        code <<= 1
    return codes


class _BitWriter:
    def __init__(self):
        self.out = bytearray()
        self.acc = 0
        self.n = 0

    def put(self, code, length):
        for i in range(length - 1, -1, -1):
            self.acc = (self.acc << 1) | ((code >> i) & 1)
            self.n += 1
            if self.n == 8:
                self.out.append(self.acc)
                if self.acc == 0xFF:
                    self.out.append(0x00)
                self.acc = 0
                self.n = 0

    def flush(self):
        while self.n:
            self.put(1, 1)  # pad with 1s


def _magnitude(v):
    if v == 0:
        return 0, 0
    s = int(v).bit_length() if v > 0 else int(-v).bit_length()
    if v < 0:
        v = v + (1 << s) - 1
    return s, v


def _encode_block(bw, block, q, dc_codes, ac_codes, pred):
    coef = dct(dct(block.astype(np.float64) - 128.0, axis=0, norm="ortho"),
               axis=1, norm="ortho")
    qc = np.rint(coef / q).astype(np.int64)
    z = qc.ravel()[ZIG]
    diff = int(z[0]) - pred
    s, bitsv = _magnitude(diff)
    c, l = dc_codes[s]
    bw.put(c, l)
    if s:
        bw.put(bitsv, s)
    run = 0
    for k in range(1, 64):
        v = int(z[k])
        if v == 0:
            run += 1
            continue
        while run > 15:
            c, l = ac_codes[0xF0]
            bw.put(c, l)
            run -= 16
        s, bitsv = _magnitude(v)
        c, l = ac_codes[(run << 4) | s]
        bw.put(c, l)
        bw.put(bitsv, s)
        run = 0
    if run:
        c, l = ac_codes[0x00]
        bw.put(c, l)
    return int(z[0])


def encode_jpeg_gray(img):
    """Minimal baseline JPEG: grayscale 4:4:4, Annex K tables."""
    H, W = img.shape
    out = bytearray(b"\xff\xd8")
    # DQT
    out += b"\xff\xdb" + struct.pack(">H", 67) + b"\x00" + \
        bytes(QY.ravel()[ZIG].astype(np.uint8))
    # SOF0
    out += b"\xff\xc0" + struct.pack(">HBHHB", 11, 8, H, W, 1) + \
        bytes([1, 0x11, 0])
    # DHT (DC lum, AC lum)
    for tc, (bits, vals) in ((0, DC_LUM), (1, (AC_LUM_BITS, AC_LUM_VALS))):
        out += b"\xff\xc4" + struct.pack(">H", 19 + len(vals)) + \
            bytes([tc << 4]) + bytes(bits) + bytes(vals)
    # SOS
    out += b"\xff\xda" + struct.pack(">HB", 8, 1) + bytes([1, 0x00]) + \
        bytes([0, 63, 0])
    bw = _BitWriter()
    dc_codes = _huff_codes(*DC_LUM)
    ac_codes = _huff_codes(AC_LUM_BITS, AC_LUM_VALS)
    ph = (H + 7) // 8 * 8
    pw = (W + 7) // 8 * 8
    padded = np.pad(img, ((0, ph - H), (0, pw - W)), mode="edge")
    pred = 0
    for by in range(0, ph, 8):
        for bx in range(0, pw, 8):
            pred = _encode_block(bw, padded[by:by + 8, bx:bx + 8], QY,
                                 dc_codes, ac_codes, pred)
    bw.flush()
    out += bw.out
    out += b"\xff\xd9"
    return bytes(out)


def test_decode_gray_roundtrip():
    rng = np.random.RandomState(0)
    # smooth image (JPEG is lossy; smooth content round-trips tightly)
    x = np.linspace(0, 4 * np.pi, 80)
    img = (127 + 100 * np.outer(np.sin(x[:56]), np.cos(x[:72] / 2)))
    img = np.clip(img + rng.uniform(-3, 3, img.shape), 0, 255) \
        .astype(np.uint8)
    data = encode_jpeg_gray(img)
    t = _C.imdecode_jpeg(data)
    assert tuple(t.shape) == (56, 72, 3)
    dec = t.numpy()[:, :, 0].astype(np.float64)
    err = np.abs(dec - img.astype(np.float64)).mean()
    assert err < 6.0, f"mean abs err {err}"


def test_decode_flat_blocks_exact():
    # flat 8x8 blocks quantize exactly (DC only)
    img = np.zeros((16, 16), dtype=np.uint8)
    img[:8, :8] = 50
    img[:8, 8:] = 120
    img[8:, :8] = 200
    img[8:, 8:] = 255
    data = encode_jpeg_gray(img)
    dec = _C.imdecode_jpeg(data).numpy()[:, :, 0]
    for (sy, sx, v) in [(0, 0, 50), (0, 8, 120), (8, 0, 200), (8, 8, 255)]:
        blk = dec[sy:sy + 8, sx:sx + 8].astype(np.int32)
        assert np.abs(blk - v).max() <= 8, (v, blk[0, 0])


def test_progressive_rejected():
    import anand_mxnet_amd  # noqa
    bad = b"\xff\xd8\xff\xc2" + struct.pack(">H", 11) + b"\x08" + \
        struct.pack(">HH", 8, 8) + b"\x01" + bytes([1, 0x11, 0]) + \
        b"\xff\xd9"
    with pytest.raises(RuntimeError, match="progressive"):
        _C.imdecode_jpeg(bad)


def test_image_imdecode_integration():
    from anand_mxnet_amd import image
    img = np.full((24, 24), 99, dtype=np.uint8)
    data = encode_jpeg_gray(img)
    nd_img = image.imdecode(data)
    a = nd_img.asnumpy()
    assert a.shape == (24, 24, 3)
    assert abs(a.mean() - 99) < 4


def _rgb_to_ycbcr(rgb):
    r, g, b = rgb[..., 0], rgb[..., 1], rgb[..., 2]
    y = 0.299 * r + 0.587 * g + 0.114 * b
    cb = -0.168736 * r - 0.331264 * g + 0.5 * b + 128
    cr = 0.5 * r - 0.418688 * g - 0.081312 * b + 128
    return y, cb, cr


def encode_jpeg_color(rgb, subsample=False):
    """Baseline color JPEG: 4:4:4 or 4:2:0, Annex K tables."""
    H, W, _ = rgb.shape
    y, cb, cr = _rgb_to_ycbcr(rgb.astype(np.float64))
    out = bytearray(b"\xff\xd8")
    out += b"\xff\xdb" + struct.pack(">H", 67) + b"\x00" + \
        bytes(QY.ravel()[ZIG].astype(np.uint8))
    out += b"\xff\xdb" + struct.pack(">H", 67) + b"\x01" + \
        bytes(QC.ravel()[ZIG].astype(np.uint8))
    hv_y = 0x22 if subsample else 0x11
    out += b"\xff\xc0" + struct.pack(">HBHHB", 17, 8, H, W, 3) + \
        bytes([1, hv_y, 0]) + bytes([2, 0x11, 1]) + bytes([3, 0x11, 1])
    for tc_th, (bits, vals) in (
            (0x00, DC_LUM), (0x10, (AC_LUM_BITS, AC_LUM_VALS)),
            (0x01, DC_CHR), (0x11, (AC_LUM_BITS, AC_LUM_VALS))):
        out += b"\xff\xc4" + struct.pack(">H", 19 + len(vals)) + \
            bytes([tc_th]) + bytes(bits) + bytes(vals)
    out += b"\xff\xda" + struct.pack(">HB", 12, 3) + \
        bytes([1, 0x00, 2, 0x11, 3, 0x11]) + bytes([0, 63, 0])

    dcl = _huff_codes(*DC_LUM)
    acl = _huff_codes(AC_LUM_BITS, AC_LUM_VALS)
    dcc = _huff_codes(*DC_CHR)
    acc = _huff_codes(AC_LUM_BITS, AC_LUM_VALS)
    bw = _BitWriter()
    if not subsample:
        ph = (H + 7) // 8 * 8
        pw = (W + 7) // 8 * 8
        planes = [np.pad(p, ((0, ph - H), (0, pw - W)), mode="edge")
                  for p in (y, cb, cr)]
        preds = [0, 0, 0]
        for by in range(0, ph, 8):
            for bx in range(0, pw, 8):
                for ci, (pl, q, dc, ac) in enumerate(
                        [(planes[0], QY, dcl, acl),
                         (planes[1], QC, dcc, acc),
                         (planes[2], QC, dcc, acc)]):
                    preds[ci] = _encode_block(
                        bw, pl[by:by + 8, bx:bx + 8], q, dc, ac, preds[ci])
    else:
        # 4:2:0: MCU = 16x16 luma (4 blocks) + 8x8 Cb + 8x8 Cr
        ph = (H + 15) // 16 * 16
        pw = (W + 15) // 16 * 16
        yp = np.pad(y, ((0, ph - H), (0, pw - W)), mode="edge")
        cbp = np.pad(cb, ((0, ph - H), (0, pw - W)), mode="edge")
        crp = np.pad(cr, ((0, ph - H), (0, pw - W)), mode="edge")
        cbs = cbp.reshape(ph // 2, 2, pw // 2, 2).mean(axis=(1, 3))
        crs = crp.reshape(ph // 2, 2, pw // 2, 2).mean(axis=(1, 3))
        preds = [0, 0, 0]
        for my in range(0, ph, 16):
            for mx in range(0, pw, 16):
                for (dy, dx) in [(0, 0), (0, 8), (8, 0), (8, 8)]:
                    preds[0] = _encode_block(
                        bw, yp[my + dy:my + dy + 8, mx + dx:mx + dx + 8],
                        QY, dcl, acl, preds[0])
                sy, sx = my // 2, mx // 2
                preds[1] = _encode_block(bw, cbs[sy:sy + 8, sx:sx + 8],
                                         QC, dcc, acc, preds[1])
                preds[2] = _encode_block(bw, crs[sy:sy + 8, sx:sx + 8],
                                         QC, dcc, acc, preds[2])
    bw.flush()
    out += bw.out + b"\xff\xd9"
    return bytes(out)


@pytest.mark.parametrize("subsample", [False, True])
def test_decode_color(subsample):
    rng = np.random.RandomState(2)
    H, W = 40, 48
    xx, yy = np.meshgrid(np.linspace(0, 1, W), np.linspace(0, 1, H))
    rgb = np.stack([
        np.clip(255 * xx, 0, 255),
        np.clip(255 * yy, 0, 255),
        np.clip(128 + 90 * np.sin(6 * xx) * np.cos(5 * yy), 0, 255),
    ], axis=2).astype(np.uint8)
    data = encode_jpeg_color(rgb, subsample=subsample)
    dec = _C.imdecode_jpeg(data).numpy().astype(np.float64)
    err = np.abs(dec - rgb.astype(np.float64)).mean()
    assert dec.shape == (H, W, 3)
    assert err < (9.0 if subsample else 6.0), f"mean abs err {err}"


def test_image_record_iter_with_jpeg_records(tmp_path):
    """End-to-end ImageRecordIter over a .rec of real JPEG payloads
    (reference: iter_image_recordio_2.cc decode path) — the records are
    encoded by the in-test spec encoder, decoded by the native decoder."""
    from anand_mxnet_amd import recordio, io as mxio

    rec_path = str(tmp_path / "data.rec")
    idx_path = str(tmp_path / "data.idx")
    rec = recordio.MXIndexedRecordIO(idx_path, rec_path, "w")
    rng = np.random.RandomState(3)
    means = []
    for i in range(6):
        xx, yy = np.meshgrid(np.linspace(0, 1, 32), np.linspace(0, 1, 32))
        base = rng.randint(40, 200)
        rgb = np.stack([np.clip(base + 40 * xx, 0, 255),
                        np.clip(base + 40 * yy, 0, 255),
                        np.full((32, 32), base)], axis=2).astype(np.uint8)
        means.append(rgb.astype(np.float64).mean())
        header = recordio.IRHeader(0, float(i % 3), i, 0)
        rec.write_idx(i, recordio.pack(header,
                                       encode_jpeg_color(rgb)))
    rec.close()

    it = mxio.ImageRecordIter(path_imgrec=rec_path, path_imgidx=idx_path,
                              data_shape=(3, 32, 32), batch_size=3,
                              shuffle=False)
    batches = list(it)
    assert len(batches) == 2
    data = batches[0].data[0].asnumpy()
    assert data.shape == (3, 3, 32, 32)
    # decoded content matches the source image statistics (lossy codec)
    for j in range(3):
        assert abs(data[j].mean() - means[j]) < 6.0, (j, data[j].mean())
    labels = batches[0].label[0].asnumpy()
    np.testing.assert_allclose(labels, [0.0, 1.0, 2.0])
