"""Golden-file checkpoint compatibility.

The fixtures below are written BY HAND from the reference byte format
(writer NDArray::Save /root/reference/src/ndarray/
ndarray.cc:1729-1990 and MXNDArraySave list format :1962) — an
independent encoder, so a bug shared between our writer and reader
cannot hide.  Covers V2 records, V1 legacy records, empty ("none")
arrays, fp16/int64 payloads, and the arg:/aux: export naming that
HybridBlock.export writes (reference gluon/block.py:1546-1572).
"""
import struct

import numpy as np
import pytest

import mxnet_amd as mx
from mxnet_amd.utils import serialization as ser

V1 = 0xF993fac8
V2 = 0xF993fac9
LIST_MAGIC = 0x112


def _golden_v2_record(arr, type_flag):
    """Hand-encoded V2 NDArray record per Appendix A."""
    out = struct.pack('<I', V2)
    out += struct.pack('<i', 0)                       # stype dense
    out += struct.pack('<i', arr.ndim)                # TShape ndim (i32)
    for d in arr.shape:
        out += struct.pack('<q', d)                   # dims (i64)
    out += struct.pack('<ii', 1, 0)                   # Context cpu(0)
    out += struct.pack('<i', type_flag)
    out += arr.tobytes()
    return out


def _golden_params(records, names):
    out = struct.pack('<QQQ', LIST_MAGIC, 0, len(records))
    for r in records:
        out += r
    out += struct.pack('<Q', len(names))
    for n in names:
        b = n.encode()
        out += struct.pack('<Q', len(b)) + b
    return out


def test_golden_v2_params_load(tmp_path):
    w = np.arange(6, dtype='float32').reshape(2, 3)
    b = np.array([1.5, -2.5], dtype='float16')
    i = np.array([7, -9], dtype='int64')
    blob = _golden_params(
        [_golden_v2_record(w, 0), _golden_v2_record(b, 2),
         _golden_v2_record(i, 6)],
        ['arg:fc_weight', 'arg:fc_bias', 'aux:step'])
    f = tmp_path / 'golden.params'
    f.write_bytes(blob)
    loaded = ser.load_ndarrays(str(f))
    assert set(loaded) == {'arg:fc_weight', 'arg:fc_bias', 'aux:step'}
    np.testing.assert_array_equal(loaded['arg:fc_weight'].asnumpy(), w)
    np.testing.assert_array_equal(loaded['arg:fc_bias'].asnumpy(), b)
    np.testing.assert_array_equal(loaded['aux:step'].asnumpy(), i)


def test_golden_none_array_does_not_desync(tmp_path):
    """A V2 record with ndim==0 is an empty 'none' array: the reference
    writer stops right after the shape (ndarray.cc is_none()); every
    record after it must still parse (ADVICE round-1 low finding)."""
    w = np.array([3.0], dtype='float32')
    none_rec = struct.pack('<I', V2) + struct.pack('<i', 0) + \
        struct.pack('<i', 0)  # ndim 0, record ENDS here
    blob = _golden_params([none_rec, _golden_v2_record(w, 0)],
                          ['arg:none', 'arg:w'])
    f = tmp_path / 'none.params'
    f.write_bytes(blob)
    loaded = ser.load_ndarrays(str(f))
    assert loaded['arg:none'].size == 0
    np.testing.assert_array_equal(loaded['arg:w'].asnumpy(), w)


def test_golden_v1_legacy_load(tmp_path):
    """V1 records: magic then TShape, no storage type field."""
    w = np.array([[9.0, 8.0]], dtype='float32')
    rec = struct.pack('<I', V1)
    rec += struct.pack('<i', 2)
    rec += struct.pack('<qq', 1, 2)
    rec += struct.pack('<ii', 1, 0)
    rec += struct.pack('<i', 0)
    rec += w.tobytes()
    blob = _golden_params([rec], ['w'])
    f = tmp_path / 'v1.params'
    f.write_bytes(blob)
    loaded = ser.load_ndarrays(str(f))
    np.testing.assert_array_equal(loaded['w'].asnumpy(), w)


def test_golden_pre_v1_u32_dims(tmp_path):
    """Pre-V1 legacy: the 'magic' field is actually ndim, dims are u32
    (reference LegacyTShapeLoad ndarray.cc:1804)."""
    w = np.array([4.0, 5.0, 6.0], dtype='float32')
    rec = struct.pack('<I', 1)          # ndim = 1 (doubles as magic)
    rec += struct.pack('<I', 3)         # u32 dim
    rec += struct.pack('<ii', 1, 0)
    rec += struct.pack('<i', 0)
    rec += w.tobytes()
    blob = _golden_params([rec], ['w'])
    f = tmp_path / 'prev1.params'
    f.write_bytes(blob)
    loaded = ser.load_ndarrays(str(f))
    np.testing.assert_array_equal(loaded['w'].asnumpy(), w)


def test_export_arg_aux_naming(tmp_path):
    """HybridBlock.export writes arg:<name> / aux:<name> (weights vs
    running statistics) — reference gluon/block.py:1546-1572; the
    exported pair must round-trip through SymbolBlock-style import."""
    from mxnet_amd.gluon import nn
    net = nn.HybridSequential()
    net.add(nn.Conv2D(4, kernel_size=3, padding=1, layout='NHWC'),
            nn.BatchNorm(axis=-1),
            nn.Dense(3))
    net.initialize()
    x = mx.nd.array(np.random.RandomState(0).randn(2, 5, 5, 2))
    net(x)  # materialize deferred shapes
    path = str(tmp_path / 'model')
    net.hybridize()
    net(x)
    net.export(path, epoch=0)
    params = ser.load_ndarrays(path + '-0000.params')
    args = {k for k in params if k.startswith('arg:')}
    auxs = {k for k in params if k.startswith('aux:')}
    assert any('weight' in k for k in args)
    assert any('running_mean' in k for k in auxs), auxs
    assert any('running_var' in k for k in auxs), auxs
    # and the exported symbol JSON parses with nodes/heads/arg_nodes
    import json
    sym = json.load(open(path + '-symbol.json'))
    for key in ('nodes', 'arg_nodes', 'heads'):
        assert key in sym, sym.keys()
