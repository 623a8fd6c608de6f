"""Custom python operators (reference python/mxnet/operator.py:
CustomOp/CustomOpProp + register; C++ bridge src/operator/custom/custom.cc).

MI355X design: no dedicated C++ worker pool — the custom op body runs
inside a torch.autograd.Function so it composes with the native autograd
graph and the HIP stream semantics (python code runs on the current
stream; any torch ops inside dispatch asynchronously as usual).
"""
import torch

from .ndarray.ndarray import NDArray

__all__ = ['CustomOp', 'CustomOpProp', 'register', 'get_all_registered']

_REGISTRY = {}


class CustomOp:
    """Base class for user ops.  Subclasses override forward/backward and
    use self.assign to honor the req mode (reference operator.py:46)."""

    def forward(self, is_train, req, in_data, out_data, aux):
        raise NotImplementedError

    def backward(self, req, out_grad, in_data, out_data, in_grad, aux):
        raise NotImplementedError

    @staticmethod
    def assign(dst, req, src):
        if req in ('null', 0):
            return
        s = src.handle if isinstance(src, NDArray) else src
        if req in ('add', 3):
            dst.handle.add_(s.to(dst.handle.dtype))
        else:  # write / inplace
            dst.handle.copy_(s)


class CustomOpProp:
    """Shape/type inference + op factory (reference operator.py:516)."""

    def __init__(self, need_top_grad=True):
        self.need_top_grad_ = need_top_grad

    def list_arguments(self):
        return ['data']

    def list_outputs(self):
        return ['output']

    def list_auxiliary_states(self):
        return []

    def infer_shape(self, in_shape):
        return in_shape, [in_shape[0]], []

    def infer_type(self, in_type):
        return in_type, [in_type[0]] * len(self.list_outputs()), []

    def create_operator(self, ctx, shapes, dtypes):
        raise NotImplementedError


def register(reg_name):
    """Decorator: ``@mx.operator.register("my_op")`` on a CustomOpProp
    subclass; invoke with ``mx.nd.Custom(*inputs, op_type="my_op")``."""
    def wrap(prop_cls):
        _REGISTRY[reg_name] = prop_cls
        return prop_cls
    return wrap


def get_all_registered():
    return dict(_REGISTRY)


class _CustomFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, prop, op, n_out, *tensors):
        in_data = [NDArray(t) for t in tensors]
        shapes, out_shapes, _ = prop.infer_shape([list(t.shape)
                                                  for t in tensors])
        outs = [NDArray(torch.empty(tuple(s), dtype=tensors[0].dtype,
                                    device=tensors[0].device))
                for s in out_shapes]
        op.forward(torch.is_grad_enabled(), ['write'] * len(outs),
                   in_data, outs, [])
        ctx.op = op
        ctx.in_tensors = tensors
        ctx.out_tensors = [o.handle for o in outs]
        if n_out == 1:
            return outs[0].handle
        return tuple(o.handle for o in outs)

    @staticmethod
    def backward(ctx, *grads):
        op = ctx.op
        in_data = [NDArray(t) for t in ctx.in_tensors]
        out_data = [NDArray(t) for t in ctx.out_tensors]
        out_grad = [NDArray(g.contiguous()) for g in grads]
        in_grad = [NDArray(torch.zeros_like(t)) for t in ctx.in_tensors]
        op.backward(['write'] * len(in_grad), out_grad, in_data, out_data,
                    in_grad, [])
        return (None, None, None) + tuple(g.handle for g in in_grad)


def invoke(op_type, *inputs, **kwargs):
    """``mx.nd.Custom(...)`` entry (reference ndarray Custom op)."""
    prop_cls = _REGISTRY[op_type]
    import inspect
    sig = inspect.signature(prop_cls.__init__)
    accepted = {k: v for k, v in kwargs.items()
                if k in sig.parameters}
    prop = prop_cls(**accepted)
    tensors = [x.handle if isinstance(x, NDArray) else x for x in inputs]
    ctx_dev = tensors[0].device if tensors else 'cpu'
    op = prop.create_operator(ctx_dev, [list(t.shape) for t in tensors],
                              [t.dtype for t in tensors])
    n_out = len(prop.list_outputs())
    out = _CustomFn.apply(prop, op, n_out, *tensors)
    if isinstance(out, tuple):
        return [NDArray(o) for o in out]
    return NDArray(out)
