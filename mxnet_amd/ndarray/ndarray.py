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

from .. import _core
from ..base import torch_dtype, np_dtype, TORCH_TO_NP, NP_TO_CORE_FLAG, \
    core_flag, core_flag_name, native_mode
from ..context import Context, current_context

__all__ = ['NDArray', 'array', 'zeros', 'ones', 'full', 'empty', 'arange',
           'from_torch', 'waitall', 'concat', 'stack', 'save', 'load']


class NDArray:
    """An n-dimensional array on a device.

    Dual-backend: ``_tt`` is a torch.Tensor (the torch-ROCm compute path)
    OR ``_h`` is a native ``_core.NDArray`` backed by the own C++ runtime
    (pooled HIP storage + threaded engine + CDNA4 kernels + own autograd
    tape).  Touching ``._t`` on a native array raises loudly — every op
    that supports the native path branches on ``is_native`` explicitly.
    """

    __slots__ = ('_tt', '_h', '_native_grad', '_bn_presums')

    def __init__(self, data):
        self._native_grad = None
        self._bn_presums = None
        if isinstance(data, _core.NDArray):
            self._h = data
            self._tt = None
        else:
            assert isinstance(data, torch.Tensor), type(data)
            self._tt = data
            self._h = None

    @property
    def is_native(self):
        return self._h is not None

    @property
    def _t(self):
        if self._tt is None:
            raise RuntimeError(
                'this NDArray is native-runtime-backed; the torch path is '
                'not available for it (op lacks a native branch)')
        return self._tt

    @_t.setter
    def _t(self, v):
        self._tt = v
        self._h = None

    def _invoke(self, name, inputs, attrs=None):
        """Run a native-registry op on this array (+ extra native inputs)."""
        hs = [a._h if isinstance(a, NDArray) else a for a in inputs]
        return NDArray(_core.invoke(name, hs, attrs or {})[0])

    # -- properties ------------------------------------------------------
    @property
    def handle(self):
        return self._h if self._h is not None else self._tt

    @property
    def shape(self):
        if self._h is not None:
            return tuple(self._h.shape)
        return tuple(self._tt.shape)

    @property
    def size(self):
        if self._h is not None:
            return self._h.size
        return self._tt.numel()

    @property
    def ndim(self):
        if self._h is not None:
            return len(self._h.shape)
        return self._tt.dim()

    @property
    def dtype(self):
        if self._h is not None:
            name = core_flag_name(self._h.dtype)
            return name if name == 'bfloat16' else _np.dtype(name)
        d = self._tt.dtype
        if d is torch.bfloat16:
            return 'bfloat16'
        return TORCH_TO_NP[d]

    @property
    def context(self):
        if self._h is not None:
            dev_type, dev_id = self._h.ctx
            return Context('cpu' if dev_type == 1 else 'gpu', dev_id)
        return Context.from_torch(self._tt.device)

    ctx = context

    @property
    def stype(self):
        return 'default'

    @property
    def grad(self):
        if self._h is not None:
            return self._native_grad
        g = self._tt.grad
        if g is None:
            return None
        return NDArray(g)

    @property
    def T(self):
        if self._h is not None:
            return self.transpose()
        return NDArray(self._t.t().contiguous()) if self._t.dim() == 2 \
            else NDArray(self._t.permute(*reversed(range(self._t.dim()))).contiguous())

    # -- sync points -----------------------------------------------------
    def wait_to_read(self):
        # native: engine WaitForVar (queue drained + execution event);
        # torch path: drain the producing device stream
        if self._h is not None:
            self._h.wait_to_read()
            return
        if self._tt.is_cuda:
            torch.cuda.synchronize(self._tt.device)

    def asnumpy(self):
        if self._h is not None:
            a = self._h.asnumpy()
            return a
        t = self._tt.detach()
        if t.dtype is torch.bfloat16:
            t = t.float()
        return t.cpu().numpy()

    def asscalar(self):
        return self.asnumpy().item()

    def item(self):
        if self._h is not None:
            return self.asnumpy().item()
        return self._tt.item()

    def __float__(self):
        return float(self.item())

    def __int__(self):
        return int(self.item())

    def __bool__(self):
        return bool(self._t.item()) if self.size == 1 else self.size > 0

    def __len__(self):
        return self.shape[0]

    # -- autograd --------------------------------------------------------
    def attach_grad(self, grad_req='write', stype=None):
        """Allocate gradient buffer and request autograd recording.

        Reference: ndarray.py attach_grad -> MXAutogradMarkVariables;
        native path: Imperative::MarkVariable over the own tape.
        """
        if self._h is not None:
            g = _core.invoke('zeros_like', [self._h], {})[0]
            self._native_grad = NDArray(g)
            _core.mark_variable(self._h, g, 2 if grad_req == 'add' else 1)
            return
        self._t.requires_grad_(True)
        if self._t.grad is None:
            self._t.grad = torch.zeros_like(self._t)
        self._t._mx_grad_req = grad_req

    def detach(self):
        if self._h is not None:
            return NDArray(self._h)
        return NDArray(self._t.detach())

    def backward(self, out_grad=None, retain_graph=False, train_mode=True):
        from .. import autograd
        autograd.backward([self], [out_grad] if out_grad is not None else None,
                          retain_graph=retain_graph, train_mode=train_mode)

    # -- conversion / movement -------------------------------------------
    def astype(self, dtype, copy=True):
        if self._h is not None:
            return self._invoke('cast', [self],
                                {'dtype': str(core_flag(dtype))})
        td = torch_dtype(dtype)
        out = self._t.to(td)
        if copy and out is self._t:
            out = out.clone()
        return NDArray(out)

    def as_in_context(self, ctx):
        if self._h is not None:
            want = (1, 0) if ctx.device_type == 'cpu' else (2, ctx.device_id)
            if tuple(self._h.ctx) == want:
                return self
            dst = _core.NDArray(list(self.shape), want[0], want[1],
                                self._h.dtype)
            self._h.copyto(dst)
            return NDArray(dst)
        dev = ctx.torch_device
        if self._t.device == dev:
            return self
        return NDArray(self._t.to(dev, non_blocking=True))

    as_in_ctx = as_in_context

    def copyto(self, other):
        """Copy to another NDArray or a Context (reference CopyFromTo,
        ndarray.cc — on MI355X this is hipMemcpyAsync on the copy stream)."""
        if self._h is not None:
            if isinstance(other, Context):
                return self.as_in_context(other)
            assert other._h is not None, 'cannot copy native -> torch NDArray'
            self._h.copyto(other._h)
            return other
        if isinstance(other, Context):
            return NDArray(self._t.to(other.torch_device, non_blocking=True).clone()
                           if self._t.device == other.torch_device
                           else self._t.to(other.torch_device, non_blocking=True))
        assert isinstance(other, NDArray)
        with torch.no_grad():
            other._t.copy_(self._t, non_blocking=True)
        return other

    def copy(self):
        if self._h is not None:
            return self._invoke('_copy', [self])
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
                new.append(self.shape[i])
            else:
                new.append(s)
        if self._h is not None:
            if _core.is_recording():
                return self._invoke(
                    'Reshape', [self],
                    {'shape': '(' + ','.join(str(s) for s in new) + ')'})
            total, known, at = self.size, 1, -1
            for i, s in enumerate(new):
                if s == -1:
                    at = i
                else:
                    known *= s
            if at >= 0:
                new[at] = total // known
            return NDArray(self._h.reshape(new))
        return NDArray(self._t.reshape(new))

    def flatten(self):
        return self.reshape(self.shape[0], -1)

    def expand_dims(self, axis):
        if self._h is not None:
            shp = list(self.shape)
            if axis < 0:
                axis += len(shp) + 1
            shp.insert(axis, 1)
            return self.reshape(shp)
        return NDArray(self._t.unsqueeze(axis))

    def squeeze(self, axis=None):
        if self._h is not None:
            shp = [d for i, d in enumerate(self.shape)
                   if not (d == 1 and (axis is None or i == axis
                                       or (isinstance(axis, (list, tuple))
                                           and i in axis)))]
            return self.reshape(shp or [1])
        return NDArray(self._t.squeeze() if axis is None else self._t.squeeze(axis))

    def transpose(self, axes=None):
        if axes is None:
            axes = tuple(reversed(range(self.ndim)))
        if self._h is not None:
            return self._invoke(
                'transpose', [self],
                {'axes': '(' + ','.join(str(a) for a in axes) + ')'})
        return NDArray(self._t.permute(*axes).contiguous())

    def broadcast_to(self, shape):
        if self._h is not None:
            return self._invoke(
                'broadcast_to', [self],
                {'shape': '(' + ','.join(str(int(d)) for d in shape) + ',)'})
        return NDArray(self._t.broadcast_to(shape).contiguous())

    def swapaxes(self, a, b):
        if self._h is not None:
            axes = list(range(self.ndim))
            axes[a], axes[b] = axes[b], axes[a]
            return self.transpose(tuple(axes))
        return NDArray(self._t.transpose(a, b).contiguous())

    def split(self, num_outputs, axis=0):
        if self._h is not None:
            n = self.shape[axis]
            step = (n + num_outputs - 1) // num_outputs
            outs = []
            for b in range(0, n, step):
                sl = [slice(None)] * self.ndim
                sl[axis] = slice(b, min(b + step, n))
                outs.append(self[tuple(sl)])
            return outs
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

    def _native_basic_index(self, key):
        """Basic int/slice indexing on a native array as a recorded
        `_strided_copy` op (gather fwd, scatter-into-zeros bwd) — views
        would share the chunk without a tape edge (reference
        ndarray.py:720 slicing semantics)."""
        if not isinstance(key, tuple):
            key = (key,)
        shp = self.shape
        # row-major element strides of the (contiguous) input
        istr = [1] * len(shp)
        for i in range(len(shp) - 2, -1, -1):
            istr[i] = istr[i + 1] * shp[i + 1]
        out_shape, out_strides, offset = [], [], 0
        dim = 0
        for k in key:
            if k is Ellipsis:
                skip = len(shp) - dim - sum(1 for kk in key
                                            if kk is not Ellipsis)
                dim += skip
                for d in range(dim - skip, dim):
                    out_shape.append(shp[d])
                    out_strides.append(istr[d])
                continue
            if isinstance(k, (int, _np.integer)):
                i = int(k)
                if i < 0:
                    i += shp[dim]
                if not 0 <= i < shp[dim]:
                    # IndexError also terminates python's iteration
                    # protocol (for d in arr) correctly
                    raise IndexError(
                        f'index {k} out of bounds for axis {dim} '
                        f'with size {shp[dim]}')
                offset += i * istr[dim]
            elif isinstance(k, slice):
                start, stop, step = k.indices(shp[dim])
                n = max(0, (stop - start + (step - (1 if step > 0 else -1)))
                        // step)
                out_shape.append(n)
                out_strides.append(istr[dim] * step)
                offset += start * istr[dim]
            else:
                return None  # fancy indexing: unsupported natively
            dim += 1
        for d in range(dim, len(shp)):
            out_shape.append(shp[d])
            out_strides.append(istr[d])
        if not out_shape:
            out_shape, out_strides = [1], [0]
        return self._invoke(
            '_strided_copy', [self],
            {'shape': '(' + ','.join(map(str, out_shape)) + ',)',
             'strides': '(' + ','.join(map(str, out_strides)) + ',)',
             'offset': str(offset)})

    def __getitem__(self, key):
        if self._h is not None:
            out = self._native_basic_index(key)
            if out is None:
                raise TypeError(
                    'native NDArray supports basic (int/slice) indexing '
                    'only; got %r' % (key,))
            return out
        out = self._t[NDArray._unwrap_index(key)]
        if not isinstance(out, torch.Tensor):
            out = torch.tensor(out)
        return NDArray(out)

    def __setitem__(self, key, value):
        if self._h is not None:
            # basic full-slice / scalar fill on native arrays (used by
            # init code paths); partial writes need the torch frontend
            if (key is None or key == slice(None) or
                    (isinstance(key, tuple)
                     and all(k == slice(None) for k in key))):
                if isinstance(value, (int, float)):
                    from .. import _core
                    _core.invoke_into(
                        '_full', [], [self._h],
                        {'value': str(float(value)),
                         'shape': '(' + ','.join(
                             str(d) for d in self.shape) + ',)',
                         'dtype': str(self._h.dtype)})
                    return
                if isinstance(value, NDArray) and value.is_native:
                    from .. import _core
                    _core.invoke_into('_copy_into', [value._h], [self._h], {})
                    return
            raise TypeError('native NDArray supports only full-slice '
                            'assignment (x[:] = v)')
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
        return self[tuple(sl)] if self._h is not None \
            else NDArray(self._t[tuple(sl)])

    # -- arithmetic --------------------------------------------------------
    @staticmethod
    def _rhs(other, like):
        if isinstance(other, NDArray):
            return other._t
        return other

    def __add__(self, o):
        if self._h is not None:
            return self._invoke('elemwise_add', [self, o]) \
                if isinstance(o, NDArray) else \
                self._invoke('_plus_scalar', [self], {'alpha': str(float(o))})
        return NDArray(self._t + NDArray._rhs(o, self))

    def __radd__(self, o):
        if self._h is not None:
            return self.__add__(o)
        return NDArray(NDArray._rhs(o, self) + self._t)

    def __sub__(self, o):
        if self._h is not None:
            return self._invoke('elemwise_sub', [self, o]) \
                if isinstance(o, NDArray) else \
                self._invoke('_plus_scalar', [self],
                             {'alpha': str(-float(o))})
        return NDArray(self._t - NDArray._rhs(o, self))

    def __rsub__(self, o):
        if self._h is not None:
            if isinstance(o, NDArray):
                return o.__sub__(self)
            return self._invoke('_rminus_scalar', [self],
                                {'alpha': str(float(o))})
        return NDArray(NDArray._rhs(o, self) - self._t)

    def __mul__(self, o):
        if self._h is not None:
            return self._invoke('elemwise_mul', [self, o]) \
                if isinstance(o, NDArray) else \
                self._invoke('_mul_scalar', [self], {'alpha': str(float(o))})
        return NDArray(self._t * NDArray._rhs(o, self))

    def __rmul__(self, o):
        if self._h is not None:
            return self.__mul__(o)
        return NDArray(NDArray._rhs(o, self) * self._t)

    def __truediv__(self, o):
        if self._h is not None:
            return self._invoke('elemwise_div', [self, o]) \
                if isinstance(o, NDArray) else \
                self._invoke('_mul_scalar', [self],
                             {'alpha': str(1.0 / float(o))})
        return NDArray(self._t / NDArray._rhs(o, self))

    def __rtruediv__(self, o):
        if self._h is not None:
            if isinstance(o, NDArray):
                return o.__truediv__(self)
            return self._invoke('_rdiv_scalar', [self],
                                {'alpha': str(float(o))})
        return NDArray(NDArray._rhs(o, self) / self._t)

    def __pow__(self, o):
        if self._h is not None:
            return self._invoke('power', [self, o]) \
                if isinstance(o, NDArray) else \
                self._invoke('_power_scalar', [self],
                             {'alpha': str(float(o))})
        return NDArray(self._t ** NDArray._rhs(o, self))

    def __mod__(self, o): return NDArray(self._t % NDArray._rhs(o, self))

    def __neg__(self):
        if self._h is not None:
            return self._invoke('negative', [self])
        return NDArray(-self._t)

    def __abs__(self):
        if self._h is not None:
            return self._invoke('abs', [self])
        return NDArray(self._t.abs())

    def __iadd__(self, o):
        if self._h is not None:
            if isinstance(o, NDArray):
                _core.invoke_into('_grad_add', [o._h], [self._h], {})
            else:
                t = self._invoke('_plus_scalar', [self],
                                 {'alpha': str(float(o))})
                _core.invoke_into('_copy_into', [t._h], [self._h], {})
            return self
        with torch.no_grad():
            self._t += NDArray._rhs(o, self)
        return self

    def __isub__(self, o):
        with torch.no_grad():
            self._t -= NDArray._rhs(o, self)
        return self

    def __imul__(self, o):
        if self._h is not None:
            t = self.__mul__(o)
            _core.invoke_into('_copy_into', [t._h], [self._h], {})
            return self
        with torch.no_grad():
            self._t *= NDArray._rhs(o, self)
        return self

    def __itruediv__(self, o):
        with torch.no_grad():
            self._t /= NDArray._rhs(o, self)
        return self

    def _ncmp(self, o, op):
        rhs = o if isinstance(o, NDArray) else self * 0.0 + float(o)
        return self._invoke(op, [self, rhs])

    def __eq__(self, o):
        if not isinstance(o, (NDArray, int, float, torch.Tensor)):
            return NotImplemented
        if self._h is not None:
            return self._ncmp(o, 'equal')
        return NDArray((self._t == NDArray._rhs(o, self)).to(self._t.dtype))

    def __ne__(self, o):
        if not isinstance(o, (NDArray, int, float, torch.Tensor)):
            return NotImplemented
        if self._h is not None:
            return self._ncmp(o, 'not_equal')
        return NDArray((self._t != NDArray._rhs(o, self)).to(self._t.dtype))

    def __gt__(self, o):
        if self._h is not None:
            return self._ncmp(o, 'greater')
        return NDArray((self._t > NDArray._rhs(o, self)).to(self._t.dtype))

    def __ge__(self, o):
        if self._h is not None:
            return self._ncmp(o, 'greater_equal')
        return NDArray((self._t >= NDArray._rhs(o, self)).to(self._t.dtype))

    def __lt__(self, o):
        if self._h is not None:
            return self._ncmp(o, 'less')
        return NDArray((self._t < NDArray._rhs(o, self)).to(self._t.dtype))

    def __le__(self, o):
        if self._h is not None:
            return self._ncmp(o, 'less_equal')
        return NDArray((self._t <= NDArray._rhs(o, self)).to(self._t.dtype))

    def __hash__(self):
        return id(self)

    # -- reductions ----------------------------------------------------------
    @staticmethod
    def _axis_attr(axis, keepdims):
        at = {'keepdims': '1' if keepdims else '0'}
        if axis is not None:
            if isinstance(axis, int):
                axis = (axis,)
            at['axis'] = '(' + ','.join(str(a) for a in axis) + ')'
        return at

    def sum(self, axis=None, keepdims=False):
        if self._h is not None:
            return self._invoke('sum', [self],
                                NDArray._axis_attr(axis, keepdims))
        if axis is None:
            return NDArray(self._t.sum())
        return NDArray(self._t.sum(dim=axis, keepdim=keepdims))

    def mean(self, axis=None, keepdims=False):
        if self._h is not None:
            return self._invoke('mean', [self],
                                NDArray._axis_attr(axis, keepdims))
        if axis is None:
            return NDArray(self._t.float().mean().to(self._t.dtype) if not self._t.is_floating_point() else self._t.mean())
        return NDArray(self._t.mean(dim=axis, keepdim=keepdims))

    def max(self, axis=None, keepdims=False):
        if self._h is not None:
            return self._invoke('max', [self],
                                NDArray._axis_attr(axis, keepdims))
        if axis is None:
            return NDArray(self._t.max())
        return NDArray(self._t.max(dim=axis, keepdim=keepdims).values)

    def min(self, axis=None, keepdims=False):
        if self._h is not None:
            return self._invoke('min', [self],
                                NDArray._axis_attr(axis, keepdims))
        if axis is None:
            return NDArray(self._t.min())
        return NDArray(self._t.min(dim=axis, keepdim=keepdims).values)

    def argmax(self, axis=None):
        if self._h is not None:
            from . import ops as _ops
            return _ops.argmax(self, axis=axis)
        return NDArray(self._t.argmax() if axis is None else self._t.argmax(dim=axis))

    def argmin(self, axis=None):
        if self._h is not None:
            from . import ops as _ops
            return _ops.argmin(self, axis=axis)
        return NDArray(self._t.argmin() if axis is None else self._t.argmin(dim=axis))

    def norm(self):
        if self._h is not None:
            return self.square().sum().sqrt()
        return NDArray(self._t.float().norm().to(self._t.dtype))

    def abs(self):
        if self._h is not None:
            return self._invoke('abs', [self])
        return NDArray(self._t.abs())

    def exp(self):
        if self._h is not None:
            return self._invoke('exp', [self])
        return NDArray(self._t.exp())

    def log(self):
        if self._h is not None:
            return self._invoke('log', [self])
        return NDArray(self._t.log())

    def sqrt(self):
        if self._h is not None:
            return self._invoke('sqrt', [self])
        return NDArray(self._t.sqrt())

    def square(self):
        if self._h is not None:
            return self._invoke('square', [self])
        return NDArray(self._t.square())

    def clip(self, a_min, a_max):
        if self._h is not None:
            return self._invoke('clip', [self],
                                {'alpha': str(float(a_min)),
                                 'beta': str(float(a_max))})
        return NDArray(self._t.clamp(a_min, a_max))

    # pickling (DataLoader workers / checkpoint helpers): numpy
    # round-trip onto the same backend and context
    def __reduce__(self):
        return (_rebuild_ndarray,
                (self.asnumpy(), str(self.dtype),
                 self.context.device_type, self.context.device_id,
                 self.is_native))

    def __repr__(self):
        return '%s\n<NDArray %s @%s>' % (
            str(self.asnumpy()), 'x'.join(map(str, self.shape)), self.context)


def _rebuild_ndarray(arr, dtype, dev_type, dev_id, was_native):
    from ..base import set_native
    prev = set_native(was_native)
    try:
        ctx = Context(dev_type, dev_id)
        return array(arr, ctx=ctx, dtype=dtype)
    finally:
        set_native(prev)


# ---------------------------------------------------------------------------
# creation functions
# ---------------------------------------------------------------------------

def _device(ctx):
    return (ctx or current_context()).torch_device


def _core_ctx(ctx):
    c = ctx or current_context()
    return (1, 0) if c.device_type == 'cpu' else (2, c.device_id)


def _native_full(shape, val, ctx, dtype):
    if isinstance(shape, int):
        shape = (shape,)
    shape = tuple(int(s) for s in shape)  # TypeError on non-int dims
    dt, di = _core_ctx(ctx)
    at = {'shape': '(' + ','.join(str(s) for s in shape) + ')',
          'value': str(float(val)), 'dtype': str(core_flag(dtype))}
    if dt == 2:
        at['__ctx_gpu__'] = str(di)
    else:
        at['__ctx_gpu__'] = '-1'
    return NDArray(_core.invoke('_full', [], at)[0])


def array(source_array, ctx=None, dtype=None):
    if native_mode():
        if isinstance(source_array, NDArray):
            source_array = source_array.asnumpy()
        from_list = not isinstance(source_array, _np.ndarray)
        a = _np.asarray(source_array)
        if dtype is not None:
            a = a.astype(np_dtype(dtype))
        elif from_list or a.dtype == _np.float64:
            a = a.astype(_np.float32)  # mxnet default dtype
        dt, di = _core_ctx(ctx)
        return NDArray(_core.from_numpy(_np.ascontiguousarray(a), dt, di))
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
    if native_mode():
        if isinstance(shape, int):
            shape = (shape,)
        dt, di = _core_ctx(ctx)
        return NDArray(_core.NDArray(list(shape), dt, di, core_flag(dtype)))
    return NDArray(torch.empty(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def zeros(shape, ctx=None, dtype=None, **kwargs):
    if native_mode():
        return _native_full(shape, 0, ctx, dtype)
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.zeros(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def ones(shape, ctx=None, dtype=None, **kwargs):
    if native_mode():
        return _native_full(shape, 1, ctx, dtype)
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.ones(shape, dtype=torch_dtype(dtype), device=_device(ctx)))


def full(shape, val, ctx=None, dtype=None):
    if native_mode():
        return _native_full(shape, val, ctx, dtype)
    if isinstance(shape, int):
        shape = (shape,)
    return NDArray(torch.full(shape, val, dtype=torch_dtype(dtype), device=_device(ctx)))


def arange(start, stop=None, step=1.0, ctx=None, dtype=None):
    if stop is None:
        start, stop = 0, start
    return NDArray(torch.arange(start, stop, step, dtype=torch_dtype(dtype),
                                device=_device(ctx)))


def zeros_like(a):
    if a.is_native:
        return a._invoke('zeros_like', [a])
    return NDArray(torch.zeros_like(a._t))


def ones_like(a):
    if a.is_native:
        return a._invoke('ones_like', [a])
    return NDArray(torch.ones_like(a._t))


def concat(*arys, dim=1):
    if len(arys) == 1 and isinstance(arys[0], (list, tuple)):
        arys = arys[0]
    if arys and arys[0].is_native:
        return arys[0]._invoke('concat', list(arys), {'dim': str(dim)})
    return NDArray(torch.cat([a._t for a in arys], dim=dim))


def stack(*arys, axis=0):
    if len(arys) == 1 and isinstance(arys[0], (list, tuple)):
        arys = arys[0]
    if arys and arys[0].is_native:
        parts = [a.expand_dims(axis) for a in arys]
        return parts[0]._invoke('concat', parts, {'dim': str(axis)})
    return NDArray(torch.stack([a._t for a in arys], dim=axis))


def waitall():
    """Block until all async work completes (reference Engine::WaitForAll:
    native engine queues drained + device streams synced; plus the torch
    streams when the torch path is in use)."""
    _core.wait_all()
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
