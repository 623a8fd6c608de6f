"""Gluon Parameter / ParameterDict.

Reference parity: python/mxnet/gluon/parameter.py (Parameter:47).  Supports
deferred shape inference, per-context replicated data for single-process
multi-GPU (the KVStore('device') model), grad_req write/add/null, and
fp16/bf16 storage with fp32 master copies handled by the optimizer
(multi-precision updates, reference optimizer_op.cc mp_* variants).
"""
from collections import OrderedDict

import torch

from ..ndarray.ndarray import NDArray, zeros, from_torch
from ..context import Context, cpu, current_context
from ..base import torch_dtype, MXNetError
from .. import initializer as _init_mod


class DeferredInitializationError(MXNetError):
    pass


class Parameter:
    """A Block parameter, possibly replicated over several contexts."""

    def __init__(self, name='weight', grad_req='write', shape=None,
                 dtype='float32', lr_mult=1.0, wd_mult=1.0, init=None,
                 allow_deferred_init=False, differentiable=True,
                 stype='default', grad_stype='default'):
        self._name = name
        self.grad_req = grad_req if differentiable else 'null'
        if isinstance(shape, int):
            shape = (shape,)
        self._shape = tuple(shape) if shape is not None else None
        self.dtype = dtype
        self.lr_mult = lr_mult
        self.wd_mult = wd_mult
        self.init = init
        self.allow_deferred_init = allow_deferred_init
        self.grad_stype = grad_stype  # 'row_sparse' => lazy sparse update
        self._data = None          # OrderedDict[Context, NDArray]
        self._grad = None
        self._deferred_init = None
        self._structure = None     # dotted path, set by Block registration

    # ------------------------------------------------------------------
    @property
    def name(self):
        return self._name

    @property
    def shape(self):
        return self._shape

    @shape.setter
    def shape(self, new_shape):
        if self._shape is None:
            self._shape = tuple(new_shape)
            return
        # allow filling in unknown (0/-1) dims
        assert len(self._shape) == len(new_shape)
        merged = []
        for o, n in zip(self._shape, new_shape):
            if o in (0, -1):
                merged.append(n)
            else:
                assert o == n, f'shape mismatch for {self._name}: {self._shape} vs {new_shape}'
                merged.append(o)
        self._shape = tuple(merged)

    def _shape_known(self):
        return self._shape is not None and all(s > 0 for s in self._shape)

    # ------------------------------------------------------------------
    def initialize(self, init=None, ctx=None, default_init=None, force_reinit=False):
        if self._data is not None and not force_reinit:
            return
        if ctx is None:
            ctx = [current_context()]
        if isinstance(ctx, Context):
            ctx = [ctx]
        default_init = default_init or _init_mod.Uniform()
        init = init or self.init or default_init
        if not self._shape_known():
            if self.allow_deferred_init:
                self._deferred_init = (init, list(ctx))
                return
            raise DeferredInitializationError(
                f'Parameter {self._name} has unknown shape {self._shape}')
        self._finish_init(init, list(ctx))

    def _finish_init(self, init, ctx_list):
        from ..base import native_mode
        init = _init_mod.create(init)
        td = torch_dtype(self.dtype)
        base = torch.empty(self._shape, dtype=torch.float32)
        init(self._name, NDArray(base))
        self._data = OrderedDict()
        self._grad = OrderedDict()
        if native_mode():
            # native runtime: values computed host-side, stored in the own
            # pooled allocator; grads attach to the own tape
            from .. import ndarray as _nd
            host = base.numpy()
            for c in ctx_list:
                nd = _nd.array(host, ctx=c)
                if str(self.dtype) != 'float32':
                    nd = nd.astype(self.dtype)
                self._data[c] = nd
                if self.grad_req != 'null':
                    nd.attach_grad(self.grad_req)
                    self._grad[c] = nd.grad
            self._deferred_init = None
            return
        for c in ctx_list:
            t = base.to(device=c.torch_device, dtype=td)
            nd = NDArray(t)
            self._data[c] = nd
            if self.grad_req != 'null':
                nd.attach_grad(self.grad_req)
                self._grad[c] = NDArray(t.grad)
        self._deferred_init = None

    def _maybe_deferred(self):
        if self._data is None and self._deferred_init is not None and self._shape_known():
            init, ctx_list = self._deferred_init
            self._finish_init(init, ctx_list)

    def finish_deferred_init(self, shape=None):
        if shape is not None:
            self.shape = shape
        self._maybe_deferred()

    # ------------------------------------------------------------------
    def _check_init(self):
        if self._data is None:
            if self._deferred_init is not None:
                raise DeferredInitializationError(
                    f'Parameter {self._name} deferred; run a forward pass first')
            raise RuntimeError(
                f"Parameter '{self._name}' has not been initialized. "
                "Call .initialize() first")

    def data(self, ctx=None):
        self._check_init()
        if ctx is None:
            return next(iter(self._data.values()))
        if isinstance(ctx, Context) and ctx in self._data:
            return self._data[ctx]
        raise RuntimeError(f'Parameter {self._name} not initialized on {ctx}')

    def list_data(self):
        self._check_init()
        return list(self._data.values())

    def grad(self, ctx=None):
        self._check_init()
        if self.grad_req == 'null':
            raise RuntimeError(f"Parameter {self._name} has grad_req='null'")
        if ctx is None:
            return next(iter(self._grad.values()))
        return self._grad[ctx]

    def list_grad(self):
        self._check_init()
        return list(self._grad.values())

    def list_ctx(self):
        self._check_init()
        return list(self._data.keys())

    def zero_grad(self):
        if self._grad is None:
            return
        from .. import _core
        for g in self._grad.values():
            if g.is_native:
                _core.invoke_into('_full', [], [g._h], {'value': '0'})
            else:
                with torch.no_grad():
                    g._t.zero_()

    def set_data(self, data):
        self._check_init()
        for nd in self._data.values():
            if nd.is_native:
                src_nd = data if data.is_native else None
                if src_nd is None:
                    from .. import ndarray as _nd
                    src_nd = _nd.array(data.asnumpy(), ctx=nd.context)
                if str(src_nd.dtype) != str(nd.dtype):
                    src_nd = src_nd.astype(nd.dtype)
                src_nd = src_nd.as_in_context(nd.context)
                src_nd.copyto(nd)
            else:
                with torch.no_grad():
                    nd._t.copy_(data._t.to(nd._t.device, nd._t.dtype))

    def row_sparse_data(self, row_id):
        raise NotImplementedError('row_sparse storage: deferred feature')

    def reset_ctx(self, ctx):
        self._check_init()
        if isinstance(ctx, Context):
            ctx = [ctx]
        cur = next(iter(self._data.values()))
        if cur.is_native:
            self._data = OrderedDict()
            self._grad = OrderedDict()
            for c in ctx:
                nd = cur.as_in_context(c)
                if nd is cur:
                    nd = cur.copy()
                self._data[c] = nd
                if self.grad_req != 'null':
                    nd.attach_grad(self.grad_req)
                    self._grad[c] = nd.grad
            return
        base = cur._t.detach()
        self._data = OrderedDict()
        self._grad = OrderedDict()
        for c in ctx:
            t = base.to(c.torch_device).clone()
            nd = NDArray(t)
            self._data[c] = nd
            if self.grad_req != 'null':
                nd.attach_grad(self.grad_req)
                self._grad[c] = NDArray(t.grad)

    def cast(self, dtype):
        self.dtype = dtype
        if self._data is None:
            return
        td = torch_dtype(dtype)
        for c, nd in list(self._data.items()):
            if nd.is_native:
                from .. import _core
                _core.drop_variable(nd._h)
                new = nd.astype(dtype)
                if self.grad_req != 'null':
                    new.attach_grad(self.grad_req)
                    self._grad[c] = new.grad
                self._data[c] = new
                continue
            t = nd._t.detach().to(td)
            new = NDArray(t)
            if self.grad_req != 'null':
                new.attach_grad(self.grad_req)
                self._grad[c] = NDArray(t.grad)
            self._data[c] = new

    def var(self):
        from ..symbol import var
        return var(self._name, shape=self._shape, dtype=self.dtype)

    def __repr__(self):
        return f'Parameter {self._name} (shape={self._shape}, dtype={self.dtype})'


class Constant(Parameter):
    """Non-differentiable constant parameter (reference gluon.Constant)."""

    def __init__(self, value, name='const'):
        if isinstance(value, NDArray):
            nd_val = value
        else:
            import numpy as np
            nd_val = NDArray(torch.as_tensor(np.asarray(value, dtype='float32')))
        super().__init__(name=name, grad_req='null', shape=nd_val.shape,
                         init=_init_mod.Constant(nd_val))
        self.value = nd_val


class ParameterDict(OrderedDict):
    """Flat name->Parameter mapping (reference ParameterDict API subset)."""

    def initialize(self, init=None, ctx=None, verbose=False, force_reinit=False):
        for p in self.values():
            p.initialize(init=init, ctx=ctx, force_reinit=force_reinit)

    def zero_grad(self):
        for p in self.values():
            p.zero_grad()

    def reset_ctx(self, ctx):
        for p in self.values():
            p.reset_ctx(ctx)

    def setattr(self, name, value):
        for p in self.values():
            setattr(p, name, value)

    def save(self, filename, strip_prefix=''):
        from ..utils import serialization
        arg_dict = {}
        for name, p in self.items():
            weight = p.data(p.list_ctx()[0])
            if name.startswith(strip_prefix):
                name = name[len(strip_prefix):]
            arg_dict[name] = weight.as_in_context(cpu())
        serialization.save_ndarrays(filename, arg_dict)

    def load(self, filename, ctx=None, allow_missing=False,
             ignore_extra=False, restore_prefix=''):
        from ..utils import serialization
        loaded = serialization.load_ndarrays(filename)
        if restore_prefix:
            loaded = {restore_prefix + k: v for k, v in loaded.items()}
        if not allow_missing:
            for name in self:
                assert name in loaded, f'Parameter {name} missing in {filename}'
        for name, data in loaded.items():
            if name not in self:
                if not ignore_extra:
                    raise ValueError(f'Parameter {name} in file but not in dict')
                continue
            p = self[name]
            if p._data is None:
                p.shape = data.shape
                p.initialize(ctx=ctx or [cpu()],
                             default_init=_init_mod.Constant(data))
            p.set_data(data)
