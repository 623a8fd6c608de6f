"""mx.np — numpy-compatible array namespace (reference python/mxnet/numpy).

Arrays are the same torch-backed NDArray; this namespace provides
numpy-style creation/op names (the reference generates these from the
_npi_* registry; here they are direct functions).
"""
import builtins

import numpy as _onp
import torch

from ..ndarray.ndarray import NDArray, waitall  # noqa: F401
from ..ndarray import ops as _ops
from ..base import torch_dtype
from ..context import current_context

ndarray = NDArray
pi = _onp.pi
e = _onp.e
inf = _onp.inf
nan = _onp.nan
newaxis = None

float32 = _onp.float32
float64 = _onp.float64
float16 = _onp.float16
int32 = _onp.int32
int64 = _onp.int64
int8 = _onp.int8
uint8 = _onp.uint8
bool_ = _onp.bool_


def _dev(ctx=None, device=None):
    c = device or ctx or current_context()
    return c.torch_device


def array(object, dtype=None, ctx=None, device=None):
    if isinstance(object, NDArray):
        t = object._t
        if dtype is not None:
            t = t.to(torch_dtype(dtype))
        return NDArray(t.to(_dev(ctx, device)))
    from_list = not isinstance(object, _onp.ndarray)
    a = _onp.asarray(object)
    if dtype is None and (from_list or a.dtype == _onp.float64):
        a = a.astype(_onp.float32)
    t = torch.as_tensor(a, dtype=torch_dtype(dtype) if dtype else None)
    return NDArray(t.to(_dev(ctx, device)))


asarray = array


def zeros(shape, dtype=None, ctx=None, device=None, order='C'):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.zeros(shape, dtype=torch_dtype(dtype), device=_dev(ctx, device)))


def ones(shape, dtype=None, ctx=None, device=None, order='C'):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.ones(shape, dtype=torch_dtype(dtype), device=_dev(ctx, device)))


def empty(shape, dtype=None, ctx=None, device=None, order='C'):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.empty(shape, dtype=torch_dtype(dtype), device=_dev(ctx, device)))


def full(shape, fill_value, dtype=None, ctx=None, device=None):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.full(shape, fill_value, dtype=torch_dtype(dtype),
                              device=_dev(ctx, device)))


def zeros_like(a, dtype=None):
    return NDArray(torch.zeros_like(a._t, dtype=torch_dtype(dtype) if dtype else None))


def ones_like(a, dtype=None):
    return NDArray(torch.ones_like(a._t, dtype=torch_dtype(dtype) if dtype else None))


def arange(start, stop=None, step=1, dtype=None, ctx=None, device=None):
    if stop is None:
        start, stop = 0, start
    return NDArray(torch.arange(start, stop, step,
                                dtype=torch_dtype(dtype) if dtype else None,
                                device=_dev(ctx, device)).to(
        torch_dtype(dtype) if dtype else torch.float32))


def linspace(start, stop, num=50, endpoint=True, dtype=None, ctx=None, device=None):
    if not endpoint:
        stop = start + (stop - start) * (num - 1) / num
    return NDArray(torch.linspace(start, stop, num,
                                  dtype=torch_dtype(dtype) if dtype else None,
                                  device=_dev(ctx, device)))


def eye(N, M=None, k=0, dtype=None, ctx=None, device=None):
    t = torch.eye(N, M or N, dtype=torch_dtype(dtype), device=_dev(ctx, device))
    if k:
        t = torch.diag_embed(torch.diagonal(t, offset=0), offset=k)[:N, :(M or N)]
    return NDArray(t)


def _w(fn):
    def op(*args, **kwargs):
        ts = [a._t if isinstance(a, NDArray) else a for a in args]
        out = fn(*ts, **kwargs)
        return NDArray(out) if isinstance(out, torch.Tensor) else out
    return op


add = _w(torch.add)
subtract = _w(torch.sub)
multiply = _w(torch.mul)
divide = _w(torch.div)
true_divide = divide
mod = _w(torch.remainder)
power = _w(torch.pow)
matmul = _w(torch.matmul)
maximum = _w(torch.maximum)
minimum = _w(torch.minimum)
exp = _w(torch.exp)
expm1 = _w(torch.expm1)
log = _w(torch.log)
log2 = _w(torch.log2)
log10 = _w(torch.log10)
log1p = _w(torch.log1p)
sqrt = _w(torch.sqrt)
cbrt = _w(lambda x: torch.sign(x) * torch.abs(x).pow(1 / 3))
square = _w(torch.square)
absolute = _w(torch.abs)
abs = absolute
sign = _w(torch.sign)
sin = _w(torch.sin)
cos = _w(torch.cos)
tan = _w(torch.tan)
arcsin = _w(torch.asin)
arccos = _w(torch.acos)
arctan = _w(torch.atan)
arctan2 = _w(torch.atan2)
sinh = _w(torch.sinh)
cosh = _w(torch.cosh)
tanh = _w(torch.tanh)
arcsinh = _w(torch.asinh)
arccosh = _w(torch.acosh)
arctanh = _w(torch.atanh)
floor = _w(torch.floor)
ceil = _w(torch.ceil)
trunc = _w(torch.trunc)
rint = _w(torch.round)
around = _w(torch.round)
round = around
round_ = around
reciprocal = _w(torch.reciprocal)
negative = _w(torch.neg)
logical_not = _w(lambda x: ~x.bool())
isnan = _w(torch.isnan)
isinf = _w(torch.isinf)
isfinite = _w(torch.isfinite)
clip = _w(torch.clamp)
where = _w(lambda c, x, y: torch.where(c.bool(), x, y))
dot = _w(torch.matmul)
tensordot = _w(torch.tensordot)
einsum = _w(torch.einsum)
outer = _w(torch.outer)


def _red(fn):
    def op(a, axis=None, dtype=None, keepdims=False, **kw):
        t = a._t if isinstance(a, NDArray) else a
        if axis is None:
            out = fn(t)
        else:
            out = fn(t, dim=axis, keepdim=keepdims)
            if not isinstance(out, torch.Tensor):   # (values, indices)
                out = out[0]
        if dtype is not None:
            out = out.to(torch_dtype(dtype))
        return NDArray(out)
    return op


sum = _red(torch.sum)
prod = _red(torch.prod)
mean = _red(torch.mean)
std = _red(torch.std)
var = _red(torch.var)
max = _red(torch.amax)
min = _red(torch.amin)
amax = max
amin = min


def argmax(a, axis=None, out=None):
    t = a._t
    return NDArray(t.argmax() if axis is None else t.argmax(dim=axis))


def argmin(a, axis=None, out=None):
    t = a._t
    return NDArray(t.argmin() if axis is None else t.argmin(dim=axis))


def concatenate(seq, axis=0, out=None):
    return NDArray(torch.cat([s._t for s in seq], dim=axis or 0))


def stack(arrays, axis=0, out=None):
    return NDArray(torch.stack([a._t for a in arrays], dim=axis))


def split(ary, indices_or_sections, axis=0):
    t = ary._t
    if isinstance(indices_or_sections, int):
        outs = torch.chunk(t, indices_or_sections, dim=axis)
    else:
        sizes = []
        prev = 0
        for i in indices_or_sections:
            sizes.append(i - prev)
            prev = i
        sizes.append(t.shape[axis] - prev)
        outs = torch.split(t, sizes, dim=axis)
    return [NDArray(o) for o in outs]


def reshape(a, newshape, order='C'):
    return a.reshape(newshape)


def transpose(a, axes=None):
    return a.transpose(axes)


def swapaxes(a, axis1, axis2):
    return NDArray(a._t.transpose(axis1, axis2))


def expand_dims(a, axis):
    return NDArray(a._t.unsqueeze(axis))


def squeeze(a, axis=None):
    return NDArray(a._t.squeeze() if axis is None else a._t.squeeze(axis))


def tile(a, reps):
    if isinstance(reps, int):
        reps = (reps,)
    return NDArray(a._t.repeat(*reps))


def repeat(a, repeats, axis=None):
    return NDArray(torch.repeat_interleave(a._t, repeats, dim=axis))


def broadcast_to(a, shape):
    return NDArray(a._t.broadcast_to(shape))


def unique(a, return_index=False, return_inverse=False, return_counts=False,
           axis=None):
    out = torch.unique(a._t, return_inverse=return_inverse,
                       return_counts=return_counts, dim=axis)
    if isinstance(out, tuple):
        return tuple(NDArray(o) for o in out)
    return NDArray(out)


def sort(a, axis=-1):
    return NDArray(torch.sort(a._t, dim=axis).values)


def argsort(a, axis=-1):
    return NDArray(torch.argsort(a._t, dim=axis))


def cumsum(a, axis=None, dtype=None):
    t = a._t
    if axis is None:
        t = t.reshape(-1)
        axis = 0
    return NDArray(torch.cumsum(t, dim=axis))


def allclose(a, b, rtol=1e-5, atol=1e-8, equal_nan=False):
    return bool(torch.allclose(a._t, b._t if isinstance(b, NDArray) else
                               torch.as_tensor(b), rtol=rtol, atol=atol,
                               equal_nan=equal_nan))


def array_equal(a, b):
    return bool(torch.equal(a._t, b._t if isinstance(b, NDArray) else
                            torch.as_tensor(b)))


from . import random  # noqa: E402
from . import linalg  # noqa: E402
