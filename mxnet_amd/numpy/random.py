"""mx.np.random (reference python/mxnet/numpy/random.py)."""
import torch

from ..ndarray.ndarray import NDArray
from ..base import torch_dtype
from ..context import current_context


def _dev(ctx=None, device=None):
    c = device or ctx or current_context()
    return c.torch_device


def seed(s):
    torch.manual_seed(s)
    import numpy as _np
    import random as _random
    _np.random.seed(s)
    _random.seed(s)


def uniform(low=0.0, high=1.0, size=None, dtype=None, ctx=None, device=None):
    size = size if size is not None else ()
    if isinstance(size, int):
        size = (size,)
    t = torch.empty(size, dtype=torch_dtype(dtype), device=_dev(ctx, device))
    return NDArray(t.uniform_(low, high))


def normal(loc=0.0, scale=1.0, size=None, dtype=None, ctx=None, device=None):
    size = size if size is not None else ()
    if isinstance(size, int):
        size = (size,)
    t = torch.empty(size, dtype=torch_dtype(dtype), device=_dev(ctx, device))
    return NDArray(t.normal_(loc, scale))


randn_like = None


def randint(low, high=None, size=None, dtype='int64', ctx=None, device=None):
    if high is None:
        low, high = 0, low
    size = size if size is not None else ()
    if isinstance(size, int):
        size = (size,)
    return NDArray(torch.randint(low, high, size, dtype=torch_dtype(dtype),
                                 device=_dev(ctx, device)))


def rand(*size):
    return uniform(size=size or ())


def randn(*size):
    return normal(size=size or ())


def choice(a, size=None, replace=True, p=None, ctx=None):
    n = a if isinstance(a, int) else len(a)
    size = size if size is not None else ()
    if isinstance(size, int):
        size = (size,)
    count = 1
    for s in size:
        count *= s
    if p is not None:
        probs = p._t if isinstance(p, NDArray) else torch.as_tensor(p)
        idx = torch.multinomial(probs.float(), count, replacement=replace)
    elif replace:
        idx = torch.randint(0, n, (count,))
    else:
        idx = torch.randperm(n)[:count]
    idx = idx.reshape(size)
    if isinstance(a, int):
        return NDArray(idx)
    return NDArray((a._t if isinstance(a, NDArray) else torch.as_tensor(a))[idx])


def shuffle(x):
    perm = torch.randperm(x._t.shape[0], device=x._t.device)
    with torch.no_grad():
        x._t.copy_(x._t[perm])


def multinomial(n, pvals, size=None):
    probs = pvals._t if isinstance(pvals, NDArray) else torch.as_tensor(pvals)
    counts = torch.multinomial(probs.float().repeat(1, 1), n, replacement=True)
    out = torch.zeros(probs.shape[-1], dtype=torch.int64)
    for i in counts.flatten():
        out[i] += 1
    return NDArray(out)
