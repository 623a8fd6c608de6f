"""Numpy-mode switches (reference python/mxnet/util.py).

The torch-backed NDArray is always numpy-semantics (zero-dim arrays,
boolean indexing), so these are recorded flags for API parity.
"""
import functools

_NP_SHAPE = True
_NP_ARRAY = True


def is_np_shape():
    return _NP_SHAPE


def is_np_array():
    return _NP_ARRAY


def set_np_shape(active=True):
    global _NP_SHAPE
    prev = _NP_SHAPE
    _NP_SHAPE = active
    return prev


def set_np(shape=True, array=True):
    global _NP_SHAPE, _NP_ARRAY
    _NP_SHAPE, _NP_ARRAY = shape, array


def reset_np():
    set_np(True, True)


def use_np(func):
    """Decorator for parity; numpy semantics are always on."""
    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        return func(*args, **kwargs)
    return wrapper


use_np_array = use_np
use_np_shape = use_np


def get_gpu_count():
    import torch
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def getenv(name):
    import os
    return os.environ.get(name)


def setenv(name, value):
    import os
    os.environ[name] = value
