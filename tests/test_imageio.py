"""Image pipeline: own baseline JPEG decoder + threaded RecordIO batcher
(reference ImageRecordIter, iter_image_recordio_2.cc).

Ground truth: a test-side numpy JPEG ENCODER (written here from the spec,
standard quant/huffman tables, 4:4:4) — an independent implementation, so
decode(encode(x)) ~ x is a real check of the C++ decoder."""
import struct

import numpy as np
import pytest

from mxnet_amd import _imageio

# ---------------------------------------------------------------------------
# minimal baseline JPEG encoder (test oracle; spec tables)
# ---------------------------------------------------------------------------
QY = np.array([  # Annex K luminance table
    16, 11, 10, 16, 24, 40, 51, 61, 12, 12, 14, 19, 26, 58, 60, 55,
    14, 13, 16, 24, 40, 57, 69, 56, 14, 17, 22, 29, 51, 87, 80, 62,
    18, 22, 37, 56, 68, 109, 103, 77, 24, 35, 55, 64, 81, 104, 113, 92,
    49, 64, 78, 87, 103, 121, 120, 101, 72, 92, 95, 98, 112, 100, 103, 99])
QC = np.array([
    17, 18, 24, 47, 99, 99, 99, 99, 18, 21, 26, 66, 99, 99, 99, 99,
    24, 26, 56, 99, 99, 99, 99, 99, 47, 66, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99])
ZZ = np.array([
    0, 1, 8, 16, 9, 2, 3, 10, 17, 24, 32, 25, 18, 11, 4, 5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6, 7, 14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63])
# Annex K huffman specs: (bits[16], vals[])
DC_LUM = ([0, 1, 5, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0],
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
    0xF3, 0xF4, 0xF5, 0xF6, 0xF7, 0xF8, 0xF9, 0xFA]
AC_LUM = (AC_LUM_BITS, AC_LUM_VALS)


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
        self.acc = (self.acc << length) | code
        self.n += length
        while self.n >= 8:
            b = (self.acc >> (self.n - 8)) & 0xFF
            self.out.append(b)
            if b == 0xFF:
                self.out.append(0x00)
            self.n -= 8

    def flush(self):
        if self.n:
            b = (self.acc << (8 - self.n)) & 0xFF
            b |= (1 << (8 - self.n)) - 1  # pad with 1s
            self.out.append(b)
            if b == 0xFF:
                self.out.append(0x00)
            self.n = 0


def _dct2(block):
    n = 8
    c = np.array([np.sqrt(1 / n) if u == 0 else np.sqrt(2 / n)
                  for u in range(n)])
    m = np.array([[c[u] * np.cos((2 * x + 1) * u * np.pi / (2 * n))
                   for x in range(n)] for u in range(n)])
    return m @ block @ m.T


def encode_jpeg_gray(img):
    """Baseline JFIF, single (luma) component, no subsampling."""
    h, w = img.shape
    assert h % 8 == 0 and w % 8 == 0
    dc_codes = _huff_codes(*DC_LUM)
    ac_codes = _huff_codes(*AC_LUM)
    bw = _BitWriter()
    pred = 0
    for by in range(0, h, 8):
        for bx in range(0, w, 8):
            blk = img[by:by + 8, bx:bx + 8].astype(np.float64) - 128.0
            coef = _dct2(blk)
            q = np.round(coef.flatten()[ZZ] / QY).astype(int)
            # NOTE the DQT segment below emits QY verbatim, and the
            # division above indexes QY by zigzag position: encoder and
            # decoder agree on the (permuted) table, which the spec
            # allows — any 64-entry table is a valid quantizer.
            dc = int(q[0])
            diff = dc - pred
            pred = dc

            def cat(v):
                return 0 if v == 0 else int(v).bit_length() if v > 0 \
                    else int(-v).bit_length()

            s = cat(diff)
            code, ln = dc_codes[s]
            bw.put(code, ln)
            if s:
                val = diff if diff > 0 else diff + (1 << s) - 1
                bw.put(val & ((1 << s) - 1), s)
            # AC
            run = 0
            last_nz = 0
            for k in range(1, 64):
                if q[k] != 0:
                    last_nz = k
            for k in range(1, last_nz + 1):
                v = int(q[k])
                if v == 0:
                    run += 1
                    continue
                while run > 15:
                    code, ln = ac_codes[0xF0]
                    bw.put(code, ln)
                    run -= 16
                s = cat(v)
                code, ln = ac_codes[(run << 4) | s]
                bw.put(code, ln)
                val = v if v > 0 else v + (1 << s) - 1
                bw.put(val & ((1 << s) - 1), s)
                run = 0
            if last_nz != 63:
                code, ln = ac_codes[0x00]
                bw.put(code, ln)
    bw.flush()

    def seg(marker, payload):
        return bytes([0xFF, marker]) + struct.pack('>H', len(payload) + 2) \
            + payload

    out = b'\xff\xd8'
    out += seg(0xDB, bytes([0]) + bytes(int(v) for v in QY))
    out += seg(0xC0, bytes([8]) + struct.pack('>HH', h, w) +
               bytes([1, 1, 0x11, 0]))
    dcb, dcv = DC_LUM
    out += seg(0xC4, bytes([0x00]) + bytes(dcb) + bytes(dcv))
    out += seg(0xC4, bytes([0x10]) + bytes(AC_LUM_BITS) +
               bytes(AC_LUM_VALS))
    out += seg(0xDA, bytes([1, 1, 0x00, 0, 63, 0]))
    out += bytes(bw.out)
    out += b'\xff\xd9'
    return out


# ---------------------------------------------------------------------------
# tests
# ---------------------------------------------------------------------------
def test_jpeg_roundtrip_gray():
    rs = np.random.RandomState(0)
    # smooth image: JPEG is lossy, smooth content survives quantization
    yy, xx = np.mgrid[0:32, 0:48]
    img = (128 + 80 * np.sin(yy / 7.0) * np.cos(xx / 9.0)).astype('uint8')
    blob = encode_jpeg_gray(img)
    got = _imageio.decode_jpeg(blob)
    assert got.shape == (32, 48, 3)
    err = np.abs(got[:, :, 0].astype(int) - img.astype(int))
    assert err.mean() < 4.0 and err.max() < 40, (err.mean(), err.max())
    # gray replicated across channels
    np.testing.assert_array_equal(got[:, :, 0], got[:, :, 1])


def test_jpeg_sharp_content():
    rs = np.random.RandomState(1)
    img = (rs.rand(16, 16) * 255).astype('uint8')
    blob = encode_jpeg_gray(img)
    got = _imageio.decode_jpeg(blob)[:, :, 0].astype(int)
    # heavy quantization on noise: loose bound, but structure must hold
    assert np.abs(got - img.astype(int)).mean() < 30


def _write_rec(path, items, raw=True):
    """items: list of (label, uint8 HxWx3)."""
    MAGIC = 0xced7230a
    with open(path, 'wb') as f:
        for label, img in items:
            if raw:
                payload = struct.pack('<II', img.shape[0], img.shape[1]) \
                    + img.tobytes()
            else:
                payload = bytes(img)  # already-encoded bytes
            body = struct.pack('<IfQQ', 0, float(label), 0, 0) + payload
            f.write(struct.pack('<II', MAGIC, len(body)))
            f.write(body)
            pad = (-len(body)) % 4
            f.write(b'\x00' * pad)


def test_record_iter_raw_batches(tmp_path):
    rs = np.random.RandomState(2)
    items = [(i % 7, (rs.rand(40, 40, 3) * 255).astype('uint8'))
             for i in range(10)]
    rec = str(tmp_path / 'data.rec')
    _write_rec(rec, items, raw=True)
    it = _imageio.ImageRecordIter(rec, batch_size=4, out_h=32, out_w=32,
                                  threads=2, shuffle=False, rand_crop=False,
                                  rand_mirror=False, resize_shorter=0)
    assert it.size == 10
    got, data, labels = it.next_batch(4, 32, 32)
    assert got == 4 and data.shape == (4, 32, 32, 3)
    # center crop of item 0: rows/cols [4:36)
    np.testing.assert_array_equal(data[0], items[0][1][4:36, 4:36])
    np.testing.assert_allclose(labels[:4], [0, 1, 2, 3])
    got, _, _ = it.next_batch(4, 32, 32)
    assert got == 4
    got, _, _ = it.next_batch(4, 32, 32)
    assert got == 2  # tail
    got, _, _ = it.next_batch(4, 32, 32)
    assert got == 0  # exhausted
    it.reset()
    got, _, _ = it.next_batch(4, 32, 32)
    assert got == 4


def test_record_iter_jpeg_and_resize(tmp_path):
    yy, xx = np.mgrid[0:64, 0:64]
    base = (128 + 60 * np.sin(yy / 9.0) * np.cos(xx / 7.0)).astype('uint8')
    blob = encode_jpeg_gray(base)
    rec = str(tmp_path / 'j.rec')
    _write_rec(rec, [(3, np.frombuffer(blob, dtype='uint8'))], raw=False)
    it = _imageio.ImageRecordIter(rec, batch_size=1, out_h=32, out_w=32,
                                  threads=1, shuffle=False, rand_crop=False,
                                  rand_mirror=False, resize_shorter=32)
    got, data, labels = it.next_batch(1, 32, 32)
    assert got == 1 and labels[0] == 3
    # decoded+resized content should still correlate with the original
    small = data[0, :, :, 0].astype(float)
    ref = base[::2, ::2].astype(float)
    corr = np.corrcoef(small.ravel(), ref.ravel())[0, 1]
    assert corr > 0.95, corr


def test_record_iter_augment(tmp_path):
    rs = np.random.RandomState(3)
    items = [(0, (rs.rand(48, 48, 3) * 255).astype('uint8'))]
    rec = str(tmp_path / 'a.rec')
    _write_rec(rec, items)
    it = _imageio.ImageRecordIter(rec, batch_size=1, out_h=32, out_w=32,
                                  threads=1, shuffle=True, rand_crop=True,
                                  rand_mirror=True, resize_shorter=0,
                                  seed=7)
    got, d1, _ = it.next_batch(1, 32, 32)
    it.reset()
    got, d2, _ = it.next_batch(1, 32, 32)
    assert got == 1
    # crops come from the source image (every pixel must exist in it)
    assert d1.min() >= items[0][1].min() and d1.max() <= items[0][1].max()


def test_csv_iter():
    import os
    """CSVIter (reference src/io/iter_csv.cc): C++ CSV parse, separate
    label file, padding on the last batch, reset."""
    import tempfile
    import mxnet_amd as mx
    from mxnet_amd.io import CSVIter
    d = tempfile.mkdtemp()
    data = np.arange(20.).reshape(5, 4)
    lab = np.arange(5.).reshape(5, 1) * 10
    np.savetxt(os.path.join(d, 'x.csv'), data, delimiter=',')
    np.savetxt(os.path.join(d, 'y.csv'), lab, delimiter=',')
    it = CSVIter(os.path.join(d, 'x.csv'), (2, 2),
                 label_csv=os.path.join(d, 'y.csv'), batch_size=2)
    assert it.num_records == 5
    b = next(it)
    np.testing.assert_array_equal(b.data[0].asnumpy(),
                                  data[:2].reshape(2, 2, 2))
    np.testing.assert_array_equal(b.label[0].asnumpy(), lab[:2])
    next(it)
    assert next(it).pad == 1
    it.reset()
    assert next(it).pad == 0
    # no-label variant defaults labels to zero
    it2 = CSVIter(os.path.join(d, 'x.csv'), (4,), batch_size=5)
    b = next(it2)
    np.testing.assert_array_equal(b.label[0].asnumpy(), np.zeros((5, 1)))
