"""Batchify functions (reference gluon/data/batchify.py + src/io/batchify.cc)."""
import numpy as _np
import torch

from ...ndarray.ndarray import NDArray


def _to_tensor(x):
    if isinstance(x, NDArray):
        if x.is_native:
            # native samples collate host-side (same as numpy inputs);
            # the batch goes back through nd.array on the active runtime
            return torch.as_tensor(x.asnumpy())
        return x._t
    if isinstance(x, torch.Tensor):
        return x
    a = _np.asarray(x)
    if a.dtype == _np.float64:
        a = a.astype(_np.float32)
    return torch.as_tensor(a)


def _wrap_batch(t):
    """Return the batch on the ACTIVE runtime (native_mode -> native)."""
    from ...base import native_mode
    if native_mode():
        from ...ndarray.ndarray import array as _arr
        return _arr(t.numpy())
    return NDArray(t)


class Stack:
    """Stack samples along a new batch axis (batchify.cc StackBatchify)."""

    def __call__(self, data):
        if isinstance(data[0], tuple):
            return tuple(Stack()([d[i] for d in data])
                         for i in range(len(data[0])))
        if isinstance(data[0], (int, float)):
            return _wrap_batch(torch.tensor(data))
        ts = [_to_tensor(d) for d in data]
        return _wrap_batch(torch.stack(ts, dim=0))


class Pad:
    """Pad ragged samples to the max length (batchify.cc PadBatchify)."""

    def __init__(self, axis=0, pad_val=0, dtype=None):
        self._axis = axis
        self._pad_val = pad_val

    def __call__(self, data):
        ts = [_to_tensor(d) for d in data]
        max_len = max(t.shape[self._axis] for t in ts)
        padded = []
        for t in ts:
            if t.shape[self._axis] < max_len:
                pad_shape = list(t.shape)
                pad_shape[self._axis] = max_len - t.shape[self._axis]
                filler = torch.full(pad_shape, self._pad_val, dtype=t.dtype)
                t = torch.cat([t, filler], dim=self._axis)
            padded.append(t)
        return _wrap_batch(torch.stack(padded, dim=0))


class Group:
    """Apply one batchify per field (batchify.cc GroupBatchify)."""

    def __init__(self, *fns):
        self._fns = fns

    def __call__(self, data):
        assert len(data[0]) == len(self._fns)
        return tuple(fn([d[i] for d in data])
                     for i, fn in enumerate(self._fns))
