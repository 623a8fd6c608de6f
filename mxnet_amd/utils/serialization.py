"""`.params` NDArray-list file format — byte-compatible with the reference.

Format (writer NDArray::Save /root/reference/src/ndarray/
ndarray.cc:1729-1990):

    u64 0x112 (kMXAPINDArrayListMagic)
    u64 0 (reserved)
    u64 count, then per-array:
        u32 magic 0xF993fac9 (V2)   [V3 = 0xF993faca also accepted on load]
        i32 stype (dense = 0... actually V2/V3 carry stype as i32; dense=0)
        TShape: i32 ndim, ndim x i64
        Context: i32 dev_type, i32 dev_id
        i32 type_flag (mshadow enum)
        raw data bytes
    u64 count, then per-name: u64 len, bytes
"""
import struct

import numpy as _np
import torch

from ..base import NP_TO_TYPE_FLAG, TYPE_FLAG_TO_NP, TORCH_TO_TYPE_FLAG, \
    TYPE_FLAG_TO_TORCH
from ..ndarray.ndarray import NDArray

LIST_MAGIC = 0x112
V2_MAGIC = 0xF993fac9
V3_MAGIC = 0xF993faca
V1_MAGIC = 0xF993fac8


def _write_ndarray(f, nd):
    if getattr(nd, 'is_native', False):
        # native runtime: bytes via asnumpy (bf16 comes back as its u16
        # bit pattern, matching the on-disk representation)
        arr = nd.asnumpy()
        if str(nd.dtype) == 'bfloat16':
            type_flag = 11
        else:
            type_flag = NP_TO_TYPE_FLAG[arr.dtype]
        shape = tuple(arr.shape)
        f.write(struct.pack('<I', V3_MAGIC if len(shape) == 0 else V2_MAGIC))
        f.write(struct.pack('<i', 0))
        f.write(struct.pack('<i', len(shape)))
        for s in shape:
            f.write(struct.pack('<q', s))
        f.write(struct.pack('<ii', 1, 0))
        f.write(struct.pack('<i', type_flag))
        f.write(_np.ascontiguousarray(arr).tobytes())
        return
    t = nd._t.detach().cpu().contiguous()
    type_flag = TORCH_TO_TYPE_FLAG[t.dtype]
    shape = tuple(t.shape)
    # 0-dim scalars only exist under np-shape semantics: V3 magic
    # (reference NDArray::Save ndarray.cc:1730-1738; in V2 ndim==0 means
    # an empty "none" array and the record stops right after the shape).
    f.write(struct.pack('<I', V3_MAGIC if len(shape) == 0 else V2_MAGIC))
    f.write(struct.pack('<i', 0))                      # stype dense
    f.write(struct.pack('<i', len(shape)))
    for s in shape:
        f.write(struct.pack('<q', s))
    f.write(struct.pack('<ii', 1, 0))                  # Context cpu(0)
    f.write(struct.pack('<i', type_flag))
    if t.dtype is torch.bfloat16:
        raw = t.view(torch.uint16).numpy().tobytes()
    else:
        raw = t.numpy().tobytes()
    f.write(raw)


def _read_exact(f, n):
    b = f.read(n)
    if len(b) != n:
        raise EOFError('truncated .params file')
    return b


def _read_ndarray(f):
    magic = struct.unpack('<I', _read_exact(f, 4))[0]
    if magic in (V2_MAGIC, V3_MAGIC):
        stype = struct.unpack('<i', _read_exact(f, 4))[0]
        if stype != 0:
            raise NotImplementedError('sparse NDArray load: deferred')
        ndim = struct.unpack('<i', _read_exact(f, 4))[0]
        # "none" arrays stop right after the shape (reference
        # NDArray::Save `if (is_none()) return` ndarray.cc:1753):
        # V2 ndim==0 and V3 ndim==-1 carry no ctx/dtype/data — reading
        # further would desync every subsequent record in the file.
        if (magic == V2_MAGIC and ndim == 0) or ndim < 0:
            return NDArray(torch.empty(0))
        shape = struct.unpack('<%dq' % ndim, _read_exact(f, 8 * ndim)) if ndim else ()
    elif magic == V1_MAGIC:
        ndim = struct.unpack('<i', _read_exact(f, 4))[0]
        if ndim == 0:
            return NDArray(torch.empty(0))
        shape = struct.unpack('<%dq' % ndim, _read_exact(f, 8 * ndim)) if ndim else ()
    else:
        # pre-V1 legacy: magic was actually ndim (u32 dims format)
        ndim = magic
        shape = struct.unpack('<%dI' % ndim, _read_exact(f, 4 * ndim)) if ndim else ()
    _dev_type, _dev_id = struct.unpack('<ii', _read_exact(f, 8))
    type_flag = struct.unpack('<i', _read_exact(f, 4))[0]
    if type_flag == 11:  # bfloat16
        n = int(_np.prod(shape)) if shape else 1
        raw = _read_exact(f, 2 * n)
        arr = _np.frombuffer(raw, dtype=_np.uint16).reshape(shape).copy()
        t = torch.from_numpy(arr).view(torch.bfloat16)
        return NDArray(t)
    dtype = TYPE_FLAG_TO_NP[type_flag]
    n = int(_np.prod(shape)) if shape else 1
    raw = _read_exact(f, dtype.itemsize * n)
    arr = _np.frombuffer(raw, dtype=dtype).reshape(shape).copy()
    from ..base import native_mode
    if native_mode():
        from ..ndarray.ndarray import array as _mk
        return _mk(arr, dtype=str(dtype))
    return NDArray(torch.from_numpy(arr))


def save_ndarrays(fname, data):
    """data: dict[str, NDArray] or list[NDArray]."""
    if isinstance(data, dict):
        names = list(data.keys())
        arrays = [data[k] for k in names]
    else:
        names = []
        arrays = list(data)
    with open(fname, 'wb') as f:
        f.write(struct.pack('<Q', LIST_MAGIC))
        f.write(struct.pack('<Q', 0))
        f.write(struct.pack('<Q', len(arrays)))
        for nd in arrays:
            _write_ndarray(f, nd)
        f.write(struct.pack('<Q', len(names)))
        for name in names:
            b = name.encode('utf-8')
            f.write(struct.pack('<Q', len(b)))
            f.write(b)


def load_ndarrays(fname):
    """Returns dict[str, NDArray] if names present, else list[NDArray]."""
    with open(fname, 'rb') as f:
        magic = struct.unpack('<Q', _read_exact(f, 8))[0]
        if magic != LIST_MAGIC:
            raise ValueError(f'{fname}: not an NDArray list file (magic {magic:#x})')
        struct.unpack('<Q', _read_exact(f, 8))  # reserved
        count = struct.unpack('<Q', _read_exact(f, 8))[0]
        arrays = [_read_ndarray(f) for _ in range(count)]
        names = []
        rest = f.read(8)
        if len(rest) == 8:
            ncount = struct.unpack('<Q', rest)[0]
            for _ in range(ncount):
                ln = struct.unpack('<Q', _read_exact(f, 8))[0]
                names.append(_read_exact(f, ln).decode('utf-8'))
    if names:
        return dict(zip(names, arrays))
    return arrays
