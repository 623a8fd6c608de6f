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




# ---------------------------------------------------------------------------
# numpy-API tail (reference src/operator/numpy/* np_* op registrations):
# torch-backed, matching numpy semantics; _w unwraps NDArray operands
# ---------------------------------------------------------------------------

def _w(x):
    return x._t if isinstance(x, NDArray) else torch.as_tensor(x)


def vstack(tup):
    return NDArray(torch.vstack([_w(a) for a in tup]))


def hstack(tup):
    return NDArray(torch.hstack([_w(a) for a in tup]))


def dstack(tup):
    return NDArray(torch.dstack([_w(a) for a in tup]))


def diff(a, n=1, axis=-1):
    return NDArray(torch.diff(_w(a), n=n, dim=axis))


def flip(m, axis=None):
    t = _w(m)
    dims = list(range(t.dim())) if axis is None else         ([axis] if isinstance(axis, int) else list(axis))
    return NDArray(torch.flip(t, dims))


def roll(a, shift, axis=None):
    t = _w(a)
    if axis is None:
        return NDArray(torch.roll(t.reshape(-1), shift).reshape(t.shape))
    return NDArray(torch.roll(t, shift, axis))


def rot90(m, k=1, axes=(0, 1)):
    return NDArray(torch.rot90(_w(m), k, axes))


def pad(array, pad_width, mode='constant', constant_values=0):
    t = _w(array)
    if isinstance(pad_width, int):
        pad_width = [(pad_width, pad_width)] * t.dim()
    elif len(pad_width) == 2 and isinstance(pad_width[0], int):
        pad_width = [tuple(pad_width)] * t.dim()
    flat = []
    for b, e in reversed(list(pad_width)):
        flat += [b, e]
    md = {'constant': 'constant', 'edge': 'replicate',
          'reflect': 'reflect'}[mode]
    kw = {'value': constant_values} if md == 'constant' else {}
    if md != 'constant' and t.dim() < 3:
        sq = t[None, None]
        out = torch.nn.functional.pad(sq, flat, mode=md, **kw)[0, 0]
    else:
        out = torch.nn.functional.pad(t, flat, mode=md, **kw)
    return NDArray(out)


def tril(m, k=0):
    return NDArray(torch.tril(_w(m), k))


def triu(m, k=0):
    return NDArray(torch.triu(_w(m), k))


def inner(a, b):
    return NDArray(torch.inner(_w(a), _w(b)))


def kron(a, b):
    return NDArray(torch.kron(_w(a), _w(b)))


def trace(a, offset=0, axis1=0, axis2=1):
    return NDArray(torch.diagonal(_w(a), offset, axis1, axis2).sum(-1))


def diag(v, k=0):
    return NDArray(torch.diag(_w(v), k))


def diagonal(a, offset=0, axis1=0, axis2=1):
    return NDArray(torch.diagonal(_w(a), offset, axis1, axis2))


def ravel(a, order='C'):
    return NDArray(_w(a).reshape(-1))


def moveaxis(a, source, destination):
    return NDArray(torch.movedim(_w(a), source, destination))


def atleast_2d(*arys):
    outs = [NDArray(torch.atleast_2d(_w(a))) for a in arys]
    return outs[0] if len(outs) == 1 else outs


def nan_to_num(x, nan=0.0, posinf=None, neginf=None):
    return NDArray(torch.nan_to_num(_w(x), nan=nan, posinf=posinf,
                                    neginf=neginf))


def logical_and(x1, x2):
    return NDArray(torch.logical_and(_w(x1), _w(x2)))


def logical_or(x1, x2):
    return NDArray(torch.logical_or(_w(x1), _w(x2)))


def logical_not(x):
    return NDArray(torch.logical_not(_w(x)))


def bitwise_and(x1, x2):
    return NDArray(torch.bitwise_and(_w(x1), _w(x2)))


def bitwise_or(x1, x2):
    return NDArray(torch.bitwise_or(_w(x1), _w(x2)))


def bitwise_xor(x1, x2):
    return NDArray(torch.bitwise_xor(_w(x1), _w(x2)))


def histogram(a, bins=10, range=None):
    t = _w(a).float()
    lo, hi = (range if range is not None
              else (float(t.min()), float(t.max())))
    hist = torch.histc(t, bins=bins, min=lo, max=hi)
    edges = torch.linspace(lo, hi, bins + 1)
    return NDArray(hist), NDArray(edges)


def percentile(a, q, axis=None):
    t = _w(a).float()
    qq = torch.as_tensor(q, dtype=torch.float64) / 100.0
    return NDArray(torch.quantile(t, qq.to(t.dtype), dim=axis))


def quantile(a, q, axis=None):
    t = _w(a).float()
    return NDArray(torch.quantile(t, torch.as_tensor(q, dtype=t.dtype),
                                  dim=axis))


def median(a, axis=None):
    # torch.median picks the lower middle for even counts; numpy
    # averages -- use the 0.5 quantile for numpy semantics
    t = _w(a).float()
    q = torch.tensor(0.5, dtype=t.dtype)
    return NDArray(torch.quantile(t, q) if axis is None
                   else torch.quantile(t, q, dim=axis))


def average(a, axis=None, weights=None):
    t = _w(a).float()
    if weights is None:
        return NDArray(t.mean() if axis is None else t.mean(dim=axis))
    w = _w(weights).float()
    if axis is None:
        return NDArray((t * w).sum() / w.sum())
    return NDArray((t * w).sum(dim=axis) / w.sum(dim=axis))


def cov(m, rowvar=True):
    t = _w(m).float()
    if not rowvar:
        t = t.t()
    return NDArray(torch.cov(t))


def corrcoef(x, rowvar=True):
    t = _w(x).float()
    if not rowvar:
        t = t.t()
    return NDArray(torch.corrcoef(t))


def cross(a, b, axis=-1):
    return NDArray(torch.cross(_w(a), _w(b), dim=axis))


def interp(x, xp, fp):
    xt, xpt, fpt = _w(x).float(), _w(xp).float(), _w(fp).float()
    idx = torch.searchsorted(xpt, xt).clamp(1, xpt.numel() - 1)
    x0, x1 = xpt[idx - 1], xpt[idx]
    y0, y1 = fpt[idx - 1], fpt[idx]
    t = (xt - x0) / (x1 - x0)
    return NDArray((y0 + t * (y1 - y0)).clamp(float(fpt[0]) if False
                                              else -float('inf'),
                                              float('inf')))


def gradient(f, *varargs, axis=None):
    t = _w(f).float()
    axes = (range(t.dim()) if axis is None
            else ([axis] if isinstance(axis, int) else axis))
    outs = [NDArray(torch.gradient(t, dim=ax)[0]) for ax in axes]
    return outs[0] if len(outs) == 1 else outs


def polyval(p, x):
    pt, xt = _w(p).float(), _w(x).float()
    out = torch.zeros_like(xt)
    for c in pt:
        out = out * xt + c
    return NDArray(out)


def bincount(x, weights=None, minlength=0):
    return NDArray(torch.bincount(_w(x).long(),
                                  _w(weights) if weights is not None
                                  else None, minlength))


def digitize(x, bins, right=False):
    return NDArray(torch.bucketize(_w(x), _w(bins), right=not right))


def nanmean(a, axis=None):
    t = _w(a).float()
    return NDArray(t.nanmean() if axis is None else t.nanmean(dim=axis))


def nansum(a, axis=None):
    t = _w(a).float()
    return NDArray(t.nansum() if axis is None else t.nansum(dim=axis))


def floor_divide(x1, x2):
    return NDArray(torch.floor_divide(_w(x1), _w(x2)))


def fmod(x1, x2):
    return NDArray(torch.fmod(_w(x1), _w(x2)))


def divmod(x1, x2):
    q = torch.div(_w(x1), _w(x2), rounding_mode='floor')
    return NDArray(q), NDArray(_w(x1) - q * _w(x2))


def heaviside(x1, x2):
    return NDArray(torch.heaviside(_w(x1), _w(x2).to(_w(x1).dtype)))


def copysign(x1, x2):
    return NDArray(torch.copysign(_w(x1), _w(x2)))


def ldexp(x1, x2):
    return NDArray(torch.ldexp(_w(x1), _w(x2)))


def frexp(x):
    m, e = torch.frexp(_w(x))
    return NDArray(m), NDArray(e)


def fix(x):
    return NDArray(torch.trunc(_w(x)))


def ediff1d(ary):
    return NDArray(torch.diff(_w(ary).reshape(-1)))


def i0(x):
    return NDArray(torch.special.i0(_w(x)))


def sinc(x):
    return NDArray(torch.sinc(_w(x)))


from . import random  # noqa: E402
from . import linalg  # noqa: E402
