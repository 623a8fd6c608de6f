"""Shared dtype tables and small helpers.

Reference parity: mxnet/python/mxnet/base.py + mshadow dtype enum
(/root/reference/3rdparty/mshadow/mshadow/base.h) — the integer type
flags below must match mshadow's so `.params` files are byte-compatible
(SURVEY.md Appendix A).
"""
import numpy as _np
import torch

# mshadow type_flag -> numpy dtype (ndarray.cc:1729 serialization order)
TYPE_FLAG_TO_NP = {
    0: _np.dtype('float32'),
    1: _np.dtype('float64'),
    2: _np.dtype('float16'),
    3: _np.dtype('uint8'),
    4: _np.dtype('int32'),
    5: _np.dtype('int8'),
    6: _np.dtype('int64'),
    7: _np.dtype('bool'),
    # 11 = bfloat16 in later mxnet; numpy has no bf16 — keep raw uint16 view
    11: _np.dtype('uint16'),
}
NP_TO_TYPE_FLAG = {v: k for k, v in TYPE_FLAG_TO_NP.items() if k != 11}

TORCH_TO_NP = {
    torch.float32: _np.dtype('float32'),
    torch.float64: _np.dtype('float64'),
    torch.float16: _np.dtype('float16'),
    torch.uint8: _np.dtype('uint8'),
    torch.int32: _np.dtype('int32'),
    torch.int8: _np.dtype('int8'),
    torch.int64: _np.dtype('int64'),
    torch.bool: _np.dtype('bool'),
}
NP_TO_TORCH = {v: k for k, v in TORCH_TO_NP.items()}
TORCH_TO_TYPE_FLAG = {t: NP_TO_TYPE_FLAG[d] for t, d in TORCH_TO_NP.items()}
TORCH_TO_TYPE_FLAG[torch.bfloat16] = 11
TYPE_FLAG_TO_TORCH = {f: NP_TO_TORCH[d] for f, d in TYPE_FLAG_TO_NP.items()
                      if d in NP_TO_TORCH}
TYPE_FLAG_TO_TORCH[11] = torch.bfloat16

_DTYPE_ALIASES = {
    'float': 'float32', 'double': 'float64', 'half': 'float16',
    'bfloat16': 'bfloat16', 'bool': 'bool',
}


def np_dtype(dtype):
    """Normalize any dtype spec (str, np.dtype, torch.dtype) to np.dtype."""
    if dtype is None:
        return _np.dtype('float32')
    if isinstance(dtype, torch.dtype):
        return TORCH_TO_NP[dtype]
    if isinstance(dtype, str):
        dtype = _DTYPE_ALIASES.get(dtype, dtype)
        if dtype == 'bfloat16':
            raise TypeError('bfloat16 has no numpy dtype; use torch_dtype')
    return _np.dtype(dtype)


def torch_dtype(dtype):
    """Normalize any dtype spec to a torch.dtype."""
    if dtype is None:
        return torch.float32
    if isinstance(dtype, torch.dtype):
        return dtype
    if isinstance(dtype, str):
        dtype = _DTYPE_ALIASES.get(dtype, dtype)
        if dtype == 'bfloat16':
            return torch.bfloat16
    return NP_TO_TORCH[_np.dtype(dtype)]


def dtype_name(dtype):
    """Canonical string name ('float32', 'bfloat16', ...)."""
    if isinstance(dtype, torch.dtype):
        return str(dtype).replace('torch.', '')
    return _np.dtype(dtype).name


class MXNetError(RuntimeError):
    """Error type mirroring the reference's base.MXNetError."""


def check_sanity():
    return True


# ---------------------------------------------------------------------------
# native runtime (mxnet_amd._core) interop
# ---------------------------------------------------------------------------
# _core DTypeFlag values (src/core/base.h) — NOTE: these differ from the
# .params serialization flags above for bfloat16 (12 vs the file format's 11)
CORE_FLAG_TO_NP = {
    0: _np.dtype('float32'), 1: _np.dtype('float64'), 2: _np.dtype('float16'),
    3: _np.dtype('uint8'), 4: _np.dtype('int32'), 5: _np.dtype('int8'),
    6: _np.dtype('int64'), 7: _np.dtype('bool'),
}
NP_TO_CORE_FLAG = {v: k for k, v in CORE_FLAG_TO_NP.items()}
CORE_FLAG_BF16 = 12


def core_flag(dtype):
    """dtype spec -> _core DTypeFlag."""
    if isinstance(dtype, str) and _DTYPE_ALIASES.get(dtype, dtype) == 'bfloat16':
        return CORE_FLAG_BF16
    if isinstance(dtype, torch.dtype):
        if dtype is torch.bfloat16:
            return CORE_FLAG_BF16
        return NP_TO_CORE_FLAG[TORCH_TO_NP[dtype]]
    return NP_TO_CORE_FLAG[np_dtype(dtype)]


def core_flag_name(flag):
    if flag == CORE_FLAG_BF16:
        return 'bfloat16'
    return CORE_FLAG_TO_NP[flag].name


_NATIVE = [None]  # None = follow MXNET_NATIVE_RUNTIME env; else forced bool


def native_mode():
    """True when new NDArrays should be backed by the native C++ runtime
    (own storage pool + engine + kernels) instead of torch tensors."""
    if _NATIVE[0] is not None:
        return _NATIVE[0]
    import os
    return os.environ.get('MXNET_NATIVE_RUNTIME', '0') == '1'


def set_native(flag):
    prev = _NATIVE[0]
    _NATIVE[0] = bool(flag) if flag is not None else None
    return prev
