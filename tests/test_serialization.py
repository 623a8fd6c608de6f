"""Checkpoint byte-format tests."""
import struct

import numpy as np

import mxnet_amd as mx
from mxnet_amd import nd
from mxnet_amd.utils import serialization as ser


def test_params_roundtrip_dict(tmp_path):
    f = str(tmp_path / 'x.params')
    data = {'a': nd.array(np.arange(6, dtype='float32').reshape(2, 3)),
            'b': nd.array(np.arange(4, dtype='float16').reshape(4,)),
            'c': nd.array(np.arange(3, dtype='int64'))}
    ser.save_ndarrays(f, data)
    loaded = ser.load_ndarrays(f)
    assert set(loaded) == {'a', 'b', 'c'}
    for k in data:
        assert loaded[k].shape == data[k].shape
        assert loaded[k].dtype == data[k].dtype
        assert np.allclose(loaded[k].asnumpy(), data[k].asnumpy())


def test_params_roundtrip_list(tmp_path):
    f = str(tmp_path / 'y.params')
    arrays = [nd.ones((3,)), nd.zeros((2, 2))]
    ser.save_ndarrays(f, arrays)
    loaded = ser.load_ndarrays(f)
    assert isinstance(loaded, list) and len(loaded) == 2


def test_params_exact_bytes(tmp_path):
    """The on-disk layout must match the reference format exactly:
    magics, TShape i32+i64 dims, Context, type flag (ndarray.cc:1729-1990)."""
    f = str(tmp_path / 'z.params')
    ser.save_ndarrays(f, {'w': nd.array(np.array([1.0, 2.0], dtype='float32'))})
    raw = open(f, 'rb').read()
    off = 0
    magic, reserved, count = struct.unpack_from('<QQQ', raw, off)
    assert magic == 0x112          # kMXAPINDArrayListMagic
    assert reserved == 0
    assert count == 1
    off += 24
    (nd_magic,) = struct.unpack_from('<I', raw, off)
    assert nd_magic == 0xF993fac9  # NDARRAY_V2_MAGIC
    off += 4
    (stype,) = struct.unpack_from('<i', raw, off)
    assert stype == 0
    off += 4
    (ndim,) = struct.unpack_from('<i', raw, off)
    assert ndim == 1
    off += 4
    (dim0,) = struct.unpack_from('<q', raw, off)
    assert dim0 == 2
    off += 8
    dev_type, dev_id = struct.unpack_from('<ii', raw, off)
    assert dev_type == 1  # cpu
    off += 8
    (tflag,) = struct.unpack_from('<i', raw, off)
    assert tflag == 0     # float32
    off += 4
    vals = struct.unpack_from('<2f', raw, off)
    assert vals == (1.0, 2.0)
    off += 8
    (name_count,) = struct.unpack_from('<Q', raw, off)
    assert name_count == 1
    off += 8
    (name_len,) = struct.unpack_from('<Q', raw, off)
    assert name_len == 1
    off += 8
    assert raw[off:off + 1] == b'w'


def test_nd_save_load(tmp_path):
    f = str(tmp_path / 'nd.params')
    nd.save(f, {'k': nd.ones((2,))})
    out = nd.load(f)
    assert np.allclose(out['k'].asnumpy(), 1)


def test_bfloat16_roundtrip(tmp_path):
    import torch
    f = str(tmp_path / 'bf.params')
    t = torch.randn(4, 4).to(torch.bfloat16)
    ser.save_ndarrays(f, {'x': nd.from_torch(t)})
    out = ser.load_ndarrays(f)['x']
    assert out._t.dtype == torch.bfloat16
    assert torch.equal(out._t, t)


def test_npy_npz_roundtrip(tmp_path):
    """.npy/.npz save/load (reference serialization/cnpy.cc)."""
    import numpy as np
    import torch
    import mxnet_amd as mx
    a = mx.nd.from_torch(torch.randn(3, 4))
    mx.nd.save(str(tmp_path / 'a.npy'), a)
    b = mx.nd.load(str(tmp_path / 'a.npy'))[0]
    np.testing.assert_allclose(b.asnumpy(), a.asnumpy())
    d = {'x': a, 'y': mx.nd.from_torch(torch.arange(5))}
    mx.nd.save(str(tmp_path / 'd.npz'), d)
    z = mx.nd.load(str(tmp_path / 'd.npz'))
    assert set(z) == {'x', 'y'}
    np.testing.assert_allclose(z['x'].asnumpy(), a.asnumpy())
