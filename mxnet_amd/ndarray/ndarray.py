"""NDArray — the imperative n-dimensional array.

Reference parity: mxnet.ndarray.NDArray (/root/reference/python/mxnet/ndarray/ndarray.py,
include/mxnet/ndarray.h:82).  MI355X-native design: storage is a
``torch.Tensor`` living in the ROCm caching allocator (HBM3E-backed on GPU);
asynchronous execution comes from HIP streams (every GPU op is an async
launch on the device's compute stream) instead of the reference's
ThreadedEngine worker threads.  ``wait_to_read``/``waitall`` map to HIP
stream/event synchronisation.  Autograd state is carried by torch.autograd,
which our ``mxnet_amd.autograd`` module drives with MXNet semantics.
"""
import numpy as _np
import torch

from ..base import torch_dtype, np_dtype, TORCH_TO_NP
from ..context import Context, current_context

__all__ = ['NDArray', 'array', 'zeros', 'ones', 'full', 'empty', 'arange',
           'from_torch', 'waitall', 'concat', 'stack', 'save', 'load']


class NDArray:
    """An n-dimensional array on a device, torch.Tensor-backed."""

    __slots__ = ('_t',)

    def __init__(self, data):
        assert isinstance(data, torch.Tensor), type(data)
        self._t = data

    # -- properties ------------------------------------------------------
    @property
    def handle(self):
        return self._t

    @property
    def shape(self):
        return tuple(self._t.shape)

    @property
    def size(self):
        return self._t.numel()

    @property
    def ndim(self):
        return self._t.dim()

    @property
    def dtype(self):
        d = self._t.dtype
        if d is torch.bfloat16:
            return 'bfloat16'
        return TORCH_TO_NP[d]

    @property
    def context(self):
        return Context.from_torch(self._t.device)

    ctx = context

    @property
    def stype(self):
        return 'default'

    @property
    def grad(self):
        g = self._t.grad
        if g is None:
            return None
        return NDArray(g)

    @property
    def T(self):
        return NDArray(self._t.t().contiguous()) if self._t.dim() == 2 \
            else NDArray(self._t.permute(*reversed(range(self._t.dim()))).contiguous())

    # -- sync points -----------------------------------------------------
    def wait_to_read(self):
        # HIP-stream model: reading on host requires draining the device
        # stream that produced this tensor (reference: WaitForVar,
        # threaded_engine.cc:379).
        if self._t.is_cuda:
            torch.cuda.synchronize(self._t.device)

    def asnumpy(self):
        t = self._t.detach()
        if t.dtype is torch.bfloat16:
            t = t.float()
        return t.cpu().numpy()

    def asscalar(self):
        return self.asnumpy().item()

    def item(self):
        return self._t.item()

    def __float__(self):
        return float(self._t.item())

    def __int__(self):
        return int(self._t.item())

    def __bool__(self):
        return bool(self._t.item()) if self.size == 1 else self.size > 0

    def __len__(self):
        return self._t.shape[0]

    # -- autograd --------------------------------------------------------
    def attach_grad(self, grad_req='write', stype=None):
        """Allocate gradient buffer and request autograd recording.

        Reference: ndarray.py attach_grad -> MXAutogradMarkVariables.
        """
        self._t.requires_grad_(True)
        if self._t.grad is None:
            self._t.grad = torch.zeros_like(self._t)
        self._t._mx_grad_req = grad_req

    def detach(self):
        return NDArray(self._t.detach())

    def backward(self, out_grad=None, retain_graph=False, train_mode=True):
        from .. import autograd
        autograd.backward([self], [out_grad] if out_grad is not None else None,
                          retain_graph=retain_graph, train_mode=train_mode)

    # -- conversion / movement -------------------------------------------
    def astype(self, dtype, copy=True):
        td = torch_dtype(dtype)
        out = self._t.to(td)
        if copy and out is self._t:
            out = out.clone()
        return NDArray(out)

    def as_in_context(self, ctx):
        dev = ctx.torch_device
        if self._t.device == dev:
            return self
        return NDArray(self._t.to(dev, non_blocking=True))

    as_in_ctx = as_in_context

    def copyto(self, other):
        """Copy to another NDArray or a Context (reference CopyFromTo,
        ndarray.cc — on MI355X this is hipMemcpyAsync on the copy stream)."""
        if isinstance(other, Context):
            return NDArray(self._t.to(other.torch_device, non_blocking=True).clone()
                           if self._t.device == other.torch_device
                           else self._t.to(other.torch_device, non_blocking=True))
        assert isinstance(other, NDArray)
        with torch.no_grad():
            other._t.copy_(self._t, non_blocking=True)
        return other

    def copy(self):
        return NDArray(self._t.clone())

    def to_torch(self):
        return self._t

    def tolist(self):
        return self._t.tolist()

    # -- shape ops ---------------------------------------------------------
    def reshape(self, *shape, **kwargs):
        if len(shape) == 1 and isinstance(shape[0], (list, tuple)):
            shape = tuple(shape[0])
        shape = tuple(-1 if s in (-1, 0) and s == -1 else s for s in shape)
        # mxnet reshape treats 0 as "copy this dim"
        new = []
        for i, s in enumerate(shape):
            if s == 0:
                new.append(self._t.shape[i])
            else:
                new.append(s)
        return NDArray(self._t.reshape(new))

    def flatten(self):
        return NDArray(self._t.reshape(self._t.shape[0], -1))

    def expand_dims(self, axis):
        return NDArray(self._t.unsqueeze(axis))

    def squeeze(self, axis=None):
        return NDArray(self._t.squeeze() if axis is None else self._t.squeeze(axis))

    def transpose(self, axes=None):
        if axes is None:
            axes = tuple(reversed(range(self.ndim)))
        return NDArray(self._t.permute(*axes).contiguous())

    def broadcast_to(self, shape):
        return NDArray(self._t.broadcast_to(shape).contiguous())

    def swapaxes(self, a, b):
        return NDArray(self._t.transpose(a, b).contiguous())

    def split(self, num_outputs, axis=0):
        outs = torch.chunk(self._t, num_outputs, dim=axis)
        return [NDArray(o) for o in outs]

    # -- indexing ----------------------------------------------------------
    @staticmethod
    def _unwrap_index(key):
        if isinstance(key, NDArray):
            return key._t
        if isinstance(key, tuple):
            return tuple(NDArray._unwrap_index(k) for k in key)
        return key

    def __getitem__(self, key):
        out = self._t[NDArray._unwrap_index(key)]
        if not isinstance(out, torch.Tensor):
            out = torch.tensor(out)
        return NDArray(out)

    def __setitem__(self, key, value):
        key = NDArray._unwrap_index(key)
        with torch.no_grad():
            if isinstance(value, NDArray):
                self._t[key] = value._t
            elif isinstance(value, (int, float)):
                self._t[key] = value
            else:
                self._t[key] = torch.as_tensor(value, dtype=self._t.dtype,
                                               device=self._t.device)

    def slice_axis(self, axis, begin, end):
        sl = [slice(None)] * self.ndim
        sl[axis] = slice(begin, end)
        return NDArray(self._t[tuple(sl)])

    # -- arithmetic --------------------------------------------------------
    @staticmethod
    def _rhs(other, like):
        if isinstance(other, NDArray):
            return other._t
        return other

    def __add__(self, o): return NDArray(self._t + NDArray._rhs(o, self))
    def __radd__(self, o): return NDArray(NDArray._rhs(o, self) + self._t)
    def __sub__(self, o): return NDArray(self._t - NDArray._rhs(o, self))
    def __rsub__(self, o): return NDArray(NDArray._rhs(o, self) - self._t)
    def __mul__(self, o): return NDArray(self._t * NDArray._rhs(o, self))
    def __rmul__(self, o): return NDArray(NDArray._rhs(o, self) * self._t)
    def __truediv__(self, o): return NDArray(self._t / NDArray._rhs(o, self))
    def __rtruediv__(self, o): return NDArray(NDArray._rhs(o, self) / self._t)
    def __pow__(self, o): return NDArray(self._t ** NDArray._rhs(o, self))
    def __mod__(self, o): return NDArray(self._t % NDArray._rhs(o, self))
    def __neg__(self): return NDArray(-self._t)
    def __abs__(self): return NDArray(self._t.abs())

    def __iadd__(self, o):
        with torch.no_grad():
            self._t += NDArray._rhs(o, self)
        return self

    def __isub__(self, o):
        with torch.no_grad():
            self._t -= NDArray._rhs(o, self)
        return self

    def __imul__(self, o):
        with torch.no_grad():
            self._t *= NDArray._rhs(o, self)
        return self

    def __itruediv__(self, o):
        with torch.no_grad():
            self._t /= NDArray._rhs(o, self)
        return self

    def __eq__(self, o): return NDArray((self._t == NDArray._rhs(o, self)).to(self._t.dtype)) if isinstance(o, (NDArray, int, float, torch.Tensor)) else NotImplemented
    def __ne__(self, o): return NDArray((self._t != NDArray._rhs(o, self)).to(self._t.dtype)) if isinstance(o, (NDArray, int, float, torch.Tensor)) else NotImplemented
    def __gt__(self, o): return NDArray((self._t > NDArray._rhs(o, self)).to(self._t.dtype))
    def __ge__(self, o): return NDArray((self._t >= NDArray._rhs(o, self)).to(self._t.dtype))
    def __lt__(self, o): return NDArray((self._t < NDArray._rhs(o, self)).to(self._t.dtype))
    def __le__(self, o): return NDArray((self._t <= NDArray._rhs(o, self)).to(self._t.dtype))

    def __hash__(self):
        return id(self)

    # -- reductions ----------------------------------------------------------
    def sum(self, axis=None, keepdims=False):
        if axis is None:
            return NDArray(self._t.sum())
        return NDArray(self._t.sum(dim=axis, keepdim=keepdims))

    def mean(self, axis=None, keepdims=False):
        if axis is None:
            return NDArray(self._t.float().mean().to(self._t.dtype) if not self._t.is_floating_point() else self._t.mean())
        return NDArray(self._t.mean(dim=axis, keepdim=keepdims))

    def max(self, axis=None, keepdims=False):
        if axis is None:
            return NDArray(self._t.max())
        return NDArray(self._t.max(dim=axis, keepdim=keepdims).values)

    def min(self, axis=None, keepdims=False):
        if axis is None:
            return NDArray(self._t.min())
        return NDArray(self._t.min(dim=axis, keepdim=keepdims).values)

    def argmax(self, axis=None):
        return NDArray(self._t.argmax() if axis is None else self._t.argmax(dim=axis))

    def argmin(self, axis=None):
        return NDArray(self._t.argmin() if axis is None else self._t.argmin(dim=axis))

    def norm(self):
        return NDArray(self._t.float().norm().to(self._t.dtype))

    def abs(self):
        return NDArray(self._t.abs())

    def clip(self, a_min, a_max):
        return NDArray(self._t.clamp(a_min, a_max))

    def __repr__(self):
        return '%s\n<NDArray %s @%s>' % (
            str(self.asnumpy()), 'x'.join(map(str, self.shape)), self.context)


# ---------------------------------------------------------------------------
# creation functions
# ---------------------------------------------------------------------------

def _device(ctx):
    return (ctx or current_context()).torch_device


def array(source_array, ctx=None, dtype=None):
    if isinstance(source_array, NDArray):
        t = source_array._t
        t = t.to(_device(ctx))
        if dtype is not None:
            t = t.to(torch_dtype(dtype))
        return NDArray(t.clone() if t is source_array._t else t)
    if isinstance(source_array, torch.Tensor):
        t = source_array.to(_device(ctx))
        if dtype is not None:
            t = t.to(torch_dtype(dtype))
        return NDArray(t)
    from_list = not isinstance(source_array, _np.ndarray)
    a = _np.asarray(source_array)
    if dtype is None and (from_list or a.dtype == _np.float64):
        a = a.astype(_np.float32)  # mxnet default dtype is fp32
    t = torch.as_tensor(a, dtype=torch_dtype(dtype) if dtype is not None else None)
    return NDArray(t.to(_device(ctx)))


def from_torch(t):
    return NDArray(t)


def empty(shape, ctx=None, dtype=None):
    return NDArray(torch.empty(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def zeros(shape, ctx=None, dtype=None, **kwargs):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.zeros(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def ones(shape, ctx=None, dtype=None, **kwargs):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.ones(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def full(shape, val, ctx=None, dtype=None):
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.full(shape, val, dtype=torch_dtype(dtype), device=_device(ctx)))


def arange(start, stop=None, step=1.0, ctx=None, dtype=None):
    if stop is None:
        start, stop = 0, start
    return NDArray(torch.arange(start, stop, step, dtype=torch_dtype(dtype),
                                device=_device(ctx)))


def zeros_like(a):
    return NDArray(torch.zeros_like(a._t))


def ones_like(a):
    return NDArray(torch.ones_like(a._t))


def concat(*arys, dim=1):
    if len(arys) == 1 and isinstance(arys[0], (list, tuple)):
        arys = arys[0]
    return NDArray(torch.cat([a._t for a in arys], dim=dim))


def stack(*arys, axis=0):
    if len(arys) == 1 and isinstance(arys[0], (list, tuple)):
        arys = arys[0]
    return NDArray(torch.stack([a._t for a in arys], dim=axis))


def waitall():
    """Block until all async GPU work completes (reference Engine::WaitForAll)."""
    if torch.cuda.is_available():
        for i in range(torch.cuda.device_count()):
            torch.cuda.synchronize(i)


def save(fname, data):
    """Save NDArray(s); ``.npy``/``.npz`` extensions use the numpy
    formats (reference src/serialization/cnpy.cc), anything else the
    reference ``.params`` byte format."""
    import numpy as _np
    if str(fname).endswith('.npy'):
        t = data[0] if isinstance(data, (list, tuple)) else data
        _np.save(fname, t.asnumpy())
        return
    if str(fname).endswith('.npz'):
        if isinstance(data, dict):
            _np.savez(fname, **{k: v.asnumpy() for k, v in data.items()})
        else:
            arrs = data if isinstance(data, (list, tuple)) else [data]
            _np.savez(fname, *[a.asnumpy() for a in arrs])
        return
    from ..utils import serialization
    serialization.save_ndarrays(fname, data)


def load(fname):
    import numpy as _np
    if str(fname).endswith('.npy'):
        return [NDArray(torch.from_numpy(_np.load(fname)))]
    if str(fname).endswith('.npz'):
        z = _np.load(fname)
        return {k: NDArray(torch.from_numpy(z[k])) for k in z.files}
    from ..utils import serialization
    return serialization.load_ndarrays(fname)
