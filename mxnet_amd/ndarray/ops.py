"""mx.nd operator namespace — MXNet-named ops over NDArray.

Reference parity: the generated wrappers of python/mxnet/ndarray/op.py for
the registered C++ ops (NNVM registry).  Here each op is a
plain function: unwrap NDArray -> torch tensor, run mxnet_amd.ops (HIP
kernels on GPU / torch fp32 oracle on CPU), wrap the result.  The same
names exist in mxnet_amd.symbol producing graph nodes, so classic
``hybrid_forward(F, x)`` code runs unchanged in both modes.
"""
import builtins

import numpy as _np
import torch

from .ndarray import NDArray, zeros, ones, array, concat as _concat_fn
from ..ops import nn as _nn
from ..base import torch_dtype


def _t(x):
    return x._t if isinstance(x, NDArray) else x


def _pair(v):
    if v is None:
        return None
    if isinstance(v, (tuple, list)):
        return tuple(int(x) for x in v)
    return (int(v), int(v))


# --- native-runtime dispatch helpers ---------------------------------------

def _tup_attr(v):
    if v is None:
        return None
    if isinstance(v, int):
        v = (v, v)
    return '(' + ','.join(str(int(x)) for x in v) + ')'


def _ninv(name, inputs, attrs=None, nout=1):
    from .. import _core
    hs = [a._h for a in inputs if a is not None]
    outs = _core.invoke(name, hs, {k: v for k, v in (attrs or {}).items()
                                   if v is not None})
    if nout == 1:
        return NDArray(outs[0])
    return [NDArray(o) for o in outs]


# --- NN ops (legacy capitalized names, reference src/operator/nn) -----------

def FullyConnected(data, weight, bias=None, num_hidden=None, no_bias=False,
                   flatten=True, **kwargs):
    if data.is_native:
        ins = [data, weight] + ([] if (no_bias or bias is None) else [bias])
        return _ninv('FullyConnected', ins,
                     {'flatten': '1' if flatten else '0'})
    return NDArray(_nn.fully_connected(_t(data), _t(weight),
                                       None if no_bias else _t(bias), flatten))


def Convolution(data, weight, bias=None, kernel=None, stride=(1, 1),
                dilate=(1, 1), pad=(0, 0), num_filter=None, num_group=1,
                no_bias=False, layout='NCHW', **kwargs):
    if data.is_native:
        assert layout == 'NHWC', \
            'native runtime convolution is NHWC (MI355X-first layout)'
        from .. import _core
        ins = [data, weight] + ([] if (no_bias or bias is None) else [bias])
        want_stats = (no_bias or bias is None) and _core.is_recording()
        y, stats = _ninv('Convolution', ins,
                         {'kernel': _tup_attr(kernel),
                          'stride': _tup_attr(stride),
                          'pad': _tup_attr(pad),
                          'dilate': _tup_attr(dilate),
                          'num_filter': str(num_filter),
                          'num_group': str(num_group),
                          'want_stats': '1' if want_stats else '0'},
                         nout=2)
        if stats.size > 1:
            # fused per-channel {sum,ssq} from the conv epilogue — the
            # following BatchNorm consumes it (skips its reduce pass)
            y._bn_presums = stats
        return y
    return NDArray(_nn.conv2d(_t(data), _t(weight),
                              None if no_bias else _t(bias),
                              stride=_pair(stride), pad=_pair(pad),
                              dilation=_pair(dilate), groups=num_group,
                              layout=layout or 'NCHW'))


def Activation(data, act_type='relu', **kwargs):
    if data.is_native:
        return _ninv('Activation', [data], {'act_type': act_type})
    return NDArray(_nn.activation(_t(data), act_type))


def LeakyReLU(data, act_type='leaky', slope=0.25, **kwargs):
    if data.is_native:
        if act_type == 'leaky':
            return _ninv('leaky_relu', [data], {'alpha': str(slope)})
        if act_type == 'gelu':
            return _ninv('gelu', [data], {})
        if act_type == 'elu':
            # elu(x) = x>0 ? x : slope*(exp(x)-1), composed
            pos = _ninv('relu', [data], {})
            neg = (((data * -1.0)._invoke('relu', [(data * -1.0)])
                    * -1.0).exp() - 1.0) * slope
            mask = (data < data * 0.0).astype(str(data.dtype))                 if str(data.dtype) != 'float32' else (data < data * 0.0)
            return pos + neg * mask
        if act_type == 'selu':
            a, l = 1.6732632423543772, 1.0507009873554805
            e = LeakyReLU(data, act_type='elu', slope=a)
            return e * l
        raise ValueError(act_type)
    x = _t(data)
    if act_type == 'leaky':
        return NDArray(torch.nn.functional.leaky_relu(x, slope))
    if act_type == 'elu':
        return NDArray(torch.nn.functional.elu(x, slope))
    if act_type == 'selu':
        return NDArray(torch.nn.functional.selu(x))
    if act_type == 'gelu':
        return NDArray(_nn.activation(x, 'gelu'))
    raise ValueError(act_type)


def Pooling(data, kernel=(2, 2), pool_type='max', stride=None, pad=(0, 0),
            global_pool=False, layout='NCHW', count_include_pad=True, **kwargs):
    if data.is_native:
        assert layout == 'NHWC', \
            'native runtime pooling is NHWC (MI355X-first layout)'
        outs = _ninv('Pooling', [data],
                     {'kernel': _tup_attr(kernel),
                      'stride': _tup_attr(stride or kernel),
                      'pad': _tup_attr(pad), 'pool_type': pool_type,
                      'global_pool': '1' if global_pool else '0',
                      'count_include_pad':
                          '1' if count_include_pad else '0'}, nout=2)
        return outs[0]
    return NDArray(_nn.pooling(_t(data), pool_type, _pair(kernel),
                               _pair(stride), _pair(pad), layout,
                               global_pool, count_include_pad))


def BatchNorm(data, gamma, beta, moving_mean, moving_var, eps=1e-5,
              momentum=0.9, fix_gamma=False, use_global_stats=False,
              axis=1, layout=None, fuse_relu=False, residual=None, **kwargs):
    from .. import autograd as _ag
    training = _ag.is_training() and not use_global_stats
    if data.is_native:
        assert axis in (-1, data.ndim - 1), \
            'native runtime BatchNorm is NHWC (channels-last axis)'
        ins = [data, gamma, beta, moving_mean, moving_var]
        if residual is not None:
            ins.append(residual)
        presums = getattr(data, '_bn_presums', None)
        if presums is not None and training:
            ins.append(presums)
            data._bn_presums = None  # consume once
        outs = _ninv('BatchNorm', ins,
                     {'momentum': str(momentum), 'eps': str(eps),
                      'training': '1' if training else '0',
                      'fuse_relu': '1' if fuse_relu else '0',
                      'has_res': '1' if residual is not None else '0',
                      'has_presums':
                          '1' if (presums is not None and training)
                          else '0'}, nout=4)
        return outs[0]
    if layout is None:
        layout = 'NHWC' if axis in (-1, _t(data).dim() - 1) else 'NCHW'
    return NDArray(_nn.batch_norm(_t(data), _t(gamma), _t(beta),
                                  _t(moving_mean), _t(moving_var),
                                  momentum=momentum, eps=eps,
                                  training=training, layout=layout,
                                  fuse_relu=fuse_relu,
                                  residual=_t(residual) if residual is not None else None))


def LayerNorm(data, gamma, beta, axis=-1, eps=1e-5, **kwargs):
    if data.is_native:
        assert axis in (-1, data.ndim - 1)
        return _ninv('LayerNorm', [data, gamma, beta],
                     {'eps': str(eps)}, nout=3)[0]
    return NDArray(_nn.layer_norm(_t(data), _t(gamma), _t(beta), axis, eps))


def Embedding(data, weight, input_dim=None, output_dim=None, dtype=None,
              sparse_grad=False, **kwargs):
    if data.is_native:
        idx = data if data.dtype == _np.dtype('int64') \
            else data.astype('int64')
        return _ninv('Embedding', [idx, weight], {})
    return NDArray(_nn.embedding(_t(data), _t(weight), sparse_grad))


def Dropout(data, p=0.5, mode='training', **kwargs):
    from .. import autograd as _ag
    if data.is_native:
        if not _ag.is_training() or p == 0:
            return data
        import random as _random
        return _ninv('Dropout', [data],
                     {'p': str(p),
                      'seed': str(_random.getrandbits(48))}, nout=2)[0]
    return NDArray(_nn.dropout(_t(data), p, _ag.is_training()))


def softmax(data, axis=-1, temperature=None, **kwargs):
    if data.is_native:
        assert axis in (-1, data.ndim - 1)
        return _ninv('softmax', [data],
                     {'temperature': str(temperature or 1.0)})
    return NDArray(_nn.softmax(_t(data), axis, temperature or 1.0))


def log_softmax(data, axis=-1, temperature=None, **kwargs):
    if data.is_native:
        assert axis in (-1, data.ndim - 1)
        return _ninv('log_softmax', [data],
                     {'temperature': str(temperature or 1.0)})
    return NDArray(_nn.log_softmax(_t(data), axis, temperature or 1.0))


def softmin(data, axis=-1, **kwargs):
    if data.is_native:
        return softmax(data * -1.0, axis=axis)
    return NDArray(_nn.softmax(-_t(data), axis, 1.0))


def SoftmaxOutput(data, label, **kwargs):
    return softmax(data, axis=-1)


def Flatten(data, **kwargs):
    if data.is_native:
        return data.reshape(data.shape[0], -1)
    return NDArray(_t(data).reshape(_t(data).shape[0], -1))


flatten = Flatten


def Concat(*data, dim=1, **kwargs):
    return _concat_fn(list(data), dim=dim)


concat = Concat


def RNN(data, parameters, state, state_cell=None, mode='lstm',
        state_size=None, num_layers=1, bidirectional=False, p=0.0, **kwargs):
    from ..ops import rnn as _rnn
    return _rnn.rnn_ndarray(data, parameters, state, state_cell, mode,
                            state_size, num_layers, bidirectional, p)


# --- elementwise / math ------------------------------------------------------

def _unary(fn, native_op=None, native_fn=None, host_fn=None):
    """Unary factory.  Native routing in priority order: a registry
    kernel (`native_op`), an nd-composed expression (`native_fn`), or a
    documented host numpy fallback (`host_fn`, for the exotic tail the
    reference also ran on generic kernels)."""
    def op(data, **kwargs):
        if getattr(data, 'is_native', False):
            if native_op is not None:
                return _ninv(native_op, [data], {})
            if native_fn is not None:
                return native_fn(data)
            if host_fn is not None:
                from .ndarray import array as _mk
                return _mk(host_fn(data.asnumpy()), ctx=data.context)
            raise RuntimeError(
                'op has no native-runtime path (torch frontend only)')
        return NDArray(fn(_t(data)))
    return op


import numpy as _onp

exp = _unary(torch.exp, native_op='exp')
log = _unary(torch.log, native_op='log')
log2 = _unary(torch.log2,
              native_fn=lambda x: x.log() * (1.0 / _onp.log(2.0)))
log10 = _unary(torch.log10,
               native_fn=lambda x: x.log() * (1.0 / _onp.log(10.0)))
log1p = _unary(torch.log1p, native_fn=lambda x: (x + 1.0).log())
expm1 = _unary(torch.expm1, native_fn=lambda x: x.exp() - 1.0)
sqrt = _unary(torch.sqrt, native_op='sqrt')
rsqrt = _unary(torch.rsqrt, native_fn=lambda x: 1.0 / x.sqrt())
cbrt = _unary(lambda x: torch.sign(x) * torch.abs(x) ** (1.0 / 3),
              host_fn=_onp.cbrt)
square = _unary(torch.square, native_op='square')
abs = _unary(torch.abs, native_op='abs')
sign = _unary(torch.sign,
              native_fn=lambda x: (x > x * 0.0) - (x < x * 0.0))
floor = _unary(torch.floor, host_fn=_onp.floor)
ceil = _unary(torch.ceil, host_fn=_onp.ceil)
round = _unary(torch.round, host_fn=_onp.round)
trunc = _unary(torch.trunc, host_fn=_onp.trunc)
rint = _unary(torch.round, host_fn=_onp.rint)
fix = _unary(torch.trunc, host_fn=_onp.fix)
sin = _unary(torch.sin, host_fn=_onp.sin)
cos = _unary(torch.cos, host_fn=_onp.cos)
tan = _unary(torch.tan, host_fn=_onp.tan)
arcsin = _unary(torch.asin, host_fn=_onp.arcsin)
arccos = _unary(torch.acos, host_fn=_onp.arccos)
arctan = _unary(torch.atan, host_fn=_onp.arctan)
sinh = _unary(torch.sinh, host_fn=_onp.sinh)
cosh = _unary(torch.cosh, host_fn=_onp.cosh)
tanh = _unary(torch.tanh, native_op='tanh')
arcsinh = _unary(torch.asinh, host_fn=_onp.arcsinh)
arccosh = _unary(torch.acosh, host_fn=_onp.arccosh)
arctanh = _unary(torch.atanh, host_fn=_onp.arctanh)
sigmoid = _unary(torch.sigmoid, native_op='sigmoid')
erf = _unary(torch.erf,
             host_fn=lambda a: __import__('scipy.special',
                                          fromlist=['erf']).erf(a))
erfinv = _unary(torch.erfinv,
                host_fn=lambda a: __import__('scipy.special',
                                             fromlist=['erfinv']).erfinv(a))
gamma = _unary(lambda x: torch.exp(torch.lgamma(x)),
               host_fn=lambda a: __import__('scipy.special',
                                            fromlist=['gamma']).gamma(a))
gammaln = _unary(torch.lgamma,
                 host_fn=lambda a: __import__('scipy.special',
                                              fromlist=['gammaln'])
                 .gammaln(a))
relu = _unary(torch.relu, native_op='relu')
negative = _unary(torch.neg, native_op='negative')
reciprocal = _unary(torch.reciprocal, native_fn=lambda x: 1.0 / x)
logical_not = _unary(lambda x: (~x.bool()).to(x.dtype),
                     native_fn=lambda x: x == (x * 0.0))


def _binary(fn, native_op=None):
    def op(lhs, rhs, **kwargs):
        if getattr(lhs, 'is_native', False):
            if native_op is None:
                raise RuntimeError(
                    'op has no native-runtime path (torch frontend only)')
            r = rhs if isinstance(rhs, NDArray) else lhs * 0.0 + float(rhs)
            return _ninv(native_op, [lhs, r], {})
        return NDArray(fn(_t(lhs), _t(rhs)))
    return op


elemwise_add = _binary(torch.add, 'elemwise_add')
elemwise_sub = _binary(torch.sub, 'elemwise_sub')
elemwise_mul = _binary(torch.mul, 'elemwise_mul')
elemwise_div = _binary(torch.div, 'elemwise_div')
broadcast_add = _binary(torch.add, 'elemwise_add')
broadcast_sub = _binary(torch.sub, 'elemwise_sub')
broadcast_mul = _binary(torch.mul, 'elemwise_mul')
broadcast_div = _binary(torch.div, 'elemwise_div')
broadcast_power = _binary(torch.pow, 'power')
broadcast_maximum = _binary(torch.maximum, 'maximum')
broadcast_minimum = _binary(torch.minimum, 'minimum')
broadcast_mod = _binary(torch.remainder)
power = _binary(torch.pow, 'power')
def hypot(lhs, rhs, **kwargs):
    if getattr(lhs, 'is_native', False):
        return (lhs.square() + rhs.square()).sqrt()
    return NDArray(torch.hypot(_t(lhs), _t(rhs)))


def _logical_and_native(a, b):
    return (a != a * 0.0) * (b != b * 0.0)


def broadcast_logical_and(lhs, rhs, **kwargs):
    if getattr(lhs, 'is_native', False):
        return _logical_and_native(lhs, rhs)
    return NDArray((_t(lhs).bool() & _t(rhs).bool()).to(_t(lhs).dtype))


def broadcast_logical_or(lhs, rhs, **kwargs):
    if getattr(lhs, 'is_native', False):
        za = lhs != lhs * 0.0
        zb = rhs != rhs * 0.0
        return ((za + zb) > za * 0.0)
    return NDArray((_t(lhs).bool() | _t(rhs).bool()).to(_t(lhs).dtype))


broadcast_equal = _binary(lambda a, b: (a == b).to(a.dtype), 'equal')
broadcast_not_equal = _binary(lambda a, b: (a != b).to(a.dtype), 'not_equal')
broadcast_greater = _binary(lambda a, b: (a > b).to(a.dtype), 'greater')
broadcast_greater_equal = _binary(lambda a, b: (a >= b).to(a.dtype),
                                  'greater_equal')
broadcast_lesser = _binary(lambda a, b: (a < b).to(a.dtype), 'less')
broadcast_lesser_equal = _binary(lambda a, b: (a <= b).to(a.dtype),
                                 'less_equal')
broadcast_hypot = hypot


def add_n(*args, **kwargs):
    """ElementwiseSum — the KVStore reduce primitive (ndarray_function.cu)."""
    if len(args) == 1 and isinstance(args[0], (list, tuple)):
        args = args[0]
    if getattr(args[0], 'is_native', False):
        out = args[0]
        for a in args[1:]:
            out = out + a
        return out
    out = _t(args[0]).clone()
    for a in args[1:]:
        out += _t(a)
    return NDArray(out)


ElementWiseSum = add_n


def where(condition, x, y, **kwargs):
    if condition.is_native:
        # mask-select composition: m*x + (1-m)*y with a float mask
        m = _ninv('not_equal', [condition, condition * 0.0], {})
        if str(m.dtype) != str(x.dtype):
            m = m.astype(x.dtype)
        return m * x + (m * -1.0 + 1.0) * y
    return NDArray(torch.where(_t(condition).bool(), _t(x), _t(y)))


def maximum(lhs, rhs, **kwargs):
    if isinstance(lhs, NDArray) and lhs.is_native:
        if isinstance(rhs, (int, float)):
            return lhs.clip(float(rhs), 3.4e38)  # max(x,c) = clip-below
        return _ninv('maximum', [lhs, rhs], {})
    return NDArray(torch.maximum(_t(lhs), torch.as_tensor(
        _t(rhs) if isinstance(rhs, NDArray) else rhs)))


def minimum(lhs, rhs, **kwargs):
    if isinstance(lhs, NDArray) and lhs.is_native:
        if isinstance(rhs, (int, float)):
            return lhs.clip(-3.4e38, float(rhs))  # min(x,c) = clip-above
        return _ninv('minimum', [lhs, rhs], {})
    return NDArray(torch.minimum(_t(lhs), torch.as_tensor(
        _t(rhs) if isinstance(rhs, NDArray) else rhs)))


def clip(data, a_min, a_max, **kwargs):
    if data.is_native:
        return data.clip(a_min, a_max)
    return NDArray(torch.clamp(_t(data), a_min, a_max))


# --- reductions --------------------------------------------------------------

def _reduce(fn, method=None, host_fn=None):
    def op(data, axis=None, keepdims=False, **kwargs):
        if getattr(data, 'is_native', False):
            if method is not None:
                return getattr(data, method)(axis=axis, keepdims=keepdims)
            from .ndarray import array as _mk
            r = host_fn(data.asnumpy(), axis=axis, keepdims=keepdims)
            return _mk(_np.asarray(r, dtype='float32'), ctx=data.context)
        x = _t(data)
        if axis is None:
            r = fn(x, None, False)
        else:
            r = fn(x, axis, keepdims)
        return NDArray(r)
    return op


sum = _reduce(lambda x, a, k: x.sum() if a is None else x.sum(dim=a, keepdim=k),
              method='sum')
mean = _reduce(lambda x, a, k: x.mean() if a is None else x.mean(dim=a, keepdim=k),
               method='mean')
prod = _reduce(lambda x, a, k: x.prod() if a is None else x.prod(dim=a, keepdim=k),
               host_fn=_np.prod)
max = _reduce(lambda x, a, k: x.max() if a is None else x.amax(dim=a, keepdim=k),
              method='max')
min = _reduce(lambda x, a, k: x.min() if a is None else x.amin(dim=a, keepdim=k),
              method='min')
nansum = _reduce(lambda x, a, k: x.nansum() if a is None else x.nansum(dim=a, keepdim=k),
                 host_fn=_np.nansum)


def norm(data, ord=2, axis=None, keepdims=False, **kwargs):
    if data.is_native:
        assert ord == 2
        sq = data.square().sum(axis=axis, keepdims=keepdims) \
            if axis is not None else data.square().sum()
        return sq.sqrt()
    x = _t(data)
    if axis is None:
        return NDArray(torch.linalg.vector_norm(x.float(), ord).to(x.dtype))
    return NDArray(torch.linalg.vector_norm(x.float(), ord, dim=axis,
                                            keepdim=keepdims).to(x.dtype))


def argmax(data, axis=None, keepdims=False, **kwargs):
    if data.is_native:
        assert not keepdims
        if axis is None:
            flat = data.reshape(int(data.size))
            return _ninv('argmax', [flat], {'axis': '0'})
        return _ninv('argmax', [data], {'axis': str(axis)})
    x = _t(data)
    out = x.argmax() if axis is None else x.argmax(dim=axis, keepdim=keepdims)
    return NDArray(out.to(torch.float32))


def argmin(data, axis=None, keepdims=False, **kwargs):
    if data.is_native:
        assert not keepdims
        if axis is None:
            flat = data.reshape(int(data.size))
            return _ninv('argmin', [flat], {'axis': '0'})
        return _ninv('argmin', [data], {'axis': str(axis)})
    x = _t(data)
    out = x.argmin() if axis is None else x.argmin(dim=axis, keepdim=keepdims)
    return NDArray(out.to(torch.float32))


def topk(data, axis=-1, k=1, ret_typ='indices', is_ascend=False, **kwargs):
    if data.is_native:
        # ordering ops on the native runtime run host-side (reference
        # used cub device radix sort; a CDNA4 sort kernel is future
        # work) — correct, with a sync cost; rare in training loops
        import numpy as np
        arr = data.asnumpy()
        order = np.argsort(arr, axis=axis)
        if not is_ascend:
            order = np.flip(order, axis=axis)
        sl = [builtins.slice(None)] * arr.ndim
        sl[axis] = builtins.slice(0, k)
        idx = np.ascontiguousarray(order[tuple(sl)]).astype('float32')
        vals = np.take_along_axis(arr, idx.astype('int64'), axis=axis)
        from .ndarray import array as _mk
        if ret_typ == 'value':
            return _mk(vals, ctx=data.context)
        if ret_typ == 'both':
            return _mk(vals, ctx=data.context), _mk(idx, ctx=data.context)
        return _mk(idx, ctx=data.context)
    vals, idx = torch.topk(_t(data), k, dim=axis, largest=not is_ascend)
    if ret_typ == 'value':
        return NDArray(vals)
    if ret_typ == 'both':
        return NDArray(vals), NDArray(idx.to(torch.float32))
    return NDArray(idx.to(torch.float32))


def sort(data, axis=-1, is_ascend=True, **kwargs):
    if data.is_native:
        import numpy as np
        arr = np.sort(data.asnumpy(), axis=axis)
        if not is_ascend:
            arr = np.flip(arr, axis=axis).copy()
        from .ndarray import array as _mk
        return _mk(arr, ctx=data.context)
    return NDArray(torch.sort(_t(data), dim=axis, descending=not is_ascend).values)


def argsort(data, axis=-1, is_ascend=True, dtype='float32', **kwargs):
    if data.is_native:
        arr = _np.argsort(data.asnumpy(), axis=axis)
        if not is_ascend:
            arr = _np.flip(arr, axis=axis).copy()
        from .ndarray import array as _mk
        return _mk(arr.astype('float32'), ctx=data.context)
    return NDArray(torch.argsort(_t(data), dim=axis,
                                 descending=not is_ascend).to(torch_dtype(dtype)))


# --- shape / data movement ---------------------------------------------------

def reshape(data, shape, **kwargs):
    return data.reshape(shape)


def transpose(data, axes=None, **kwargs):
    return data.transpose(axes)


def expand_dims(data, axis, **kwargs):
    if data.is_native:
        return data.expand_dims(axis)
    return NDArray(_t(data).unsqueeze(axis))


def squeeze(data, axis=None, **kwargs):
    if data.is_native:
        return data.squeeze(axis)
    return NDArray(_t(data).squeeze() if axis is None else _t(data).squeeze(axis))


def stack(*data, axis=0, **kwargs):
    if len(data) == 1 and isinstance(data[0], (list, tuple)):
        data = data[0]
    if data and getattr(data[0], 'is_native', False):
        from .ndarray import stack as _stk
        return _stk(list(data), axis=axis)
    return NDArray(torch.stack([_t(d) for d in data], dim=axis))


def split(data, num_outputs, axis=1, squeeze_axis=False, **kwargs):
    if data.is_native:
        outs = data.split(num_outputs, axis=axis)
        if squeeze_axis:
            outs = [o.squeeze(axis=axis) for o in outs]
        return outs if len(outs) > 1 else outs[0]
    outs = torch.chunk(_t(data), num_outputs, dim=axis)
    if squeeze_axis:
        outs = [o.squeeze(axis) for o in outs]
    res = [NDArray(o) for o in outs]
    return res if len(res) > 1 else res[0]


def slice(data, begin, end, step=None, **kwargs):
    if data.is_native:
        sl = []
        for i in range(len(begin)):
            sl.append(builtins.slice(
                begin[i], end[i],
                step[i] if step and step[i] is not None else None))
        return data[tuple(sl)]
    x = _t(data)
    sl = []
    for i in range(len(begin)):
        b = begin[i] if begin[i] is not None else None
        e = end[i] if end[i] is not None else None
        s = step[i] if step and step[i] is not None else None
        sl.append(builtins.slice(b, e, s))
    return NDArray(x[tuple(sl)])


def slice_axis(data, axis, begin, end, **kwargs):
    return data.slice_axis(axis, begin, end)


def slice_like(data, shape_like, axes=None, **kwargs):
    if data.is_native:
        axes2 = axes or range(len(shape_like.shape))
        sl = [builtins.slice(None)] * len(data.shape)
        for ax in axes2:
            sl[ax] = builtins.slice(0, shape_like.shape[ax])
        return data[tuple(sl)]
    x, ref = _t(data), _t(shape_like)
    axes = axes or range(ref.dim())
    sl = [builtins.slice(None)] * x.dim()
    for ax in axes:
        sl[ax] = builtins.slice(0, ref.shape[ax])
    return NDArray(x[tuple(sl)])


def take(a, indices, axis=0, **kwargs):
    if a.is_native:
        assert axis == 0, 'native take: axis 0 (embedding gather)'
        idx = indices if str(indices.dtype) == 'int64' \
            else indices.astype('int64')
        flat = idx.reshape(int(idx.size))
        out = _ninv('Embedding', [flat, a], {})
        oshape = tuple(indices.shape) + tuple(a.shape[1:])
        return out.reshape(oshape)
    return NDArray(torch.index_select(_t(a), axis, _t(indices).long().reshape(-1)))


def pick(data, index, axis=-1, keepdims=False, **kwargs):
    if data.is_native:
        assert axis in (-1, data.ndim - 1) and not keepdims
        idx = index if index.dtype == _np.dtype('int64') \
            else index.astype('int64')
        return _ninv('pick', [data, idx], {})
    x, idx = _t(data), _t(index).long()
    out = torch.gather(x, axis, idx.unsqueeze(axis))
    if not keepdims:
        out = out.squeeze(axis)
    return NDArray(out)


def gather_nd(data, indices, **kwargs):
    if data.is_native:
        from .ndarray import array as _mk
        x = data.asnumpy()
        idx = indices.asnumpy().astype('int64')
        return _mk(x[tuple(idx[i] for i in range(idx.shape[0]))],
                   ctx=data.context)
    x, idx = _t(data), _t(indices).long()
    return NDArray(x[tuple(idx[i] for i in range(idx.shape[0]))])


def one_hot(indices, depth, on_value=1.0, off_value=0.0, dtype='float32', **kwargs):
    if indices.is_native:
        from ..base import core_flag
        idx = indices if indices.dtype == _np.dtype('int64') \
            else indices.astype('int64')
        return _ninv('one_hot', [idx],
                     {'depth': str(depth), 'on_value': str(on_value),
                      'off_value': str(off_value),
                      'dtype': str(core_flag(dtype))})
    oh = torch.nn.functional.one_hot(_t(indices).long(), depth)
    oh = oh.to(torch_dtype(dtype)) * (on_value - off_value) + off_value
    return NDArray(oh)


def tile(data, reps, **kwargs):
    if data.is_native:
        out = data
        from .ndarray import concat as _cat
        for ax, r in enumerate(reps):
            if r > 1:
                out = _cat([out] * int(r), dim=ax)
        return out
    return NDArray(_t(data).repeat(*reps))


def repeat(data, repeats, axis=None, **kwargs):
    if data.is_native:
        if axis is None:
            data = data.reshape(int(data.size))
            axis = 0
        # interleaved repeat: expand a size-1 axis then fold it in
        shp = list(data.shape)
        x = data.reshape(shp[:axis + 1] + [1] + shp[axis + 1:])
        bshape = shp[:axis + 1] + [int(repeats)] + shp[axis + 1:]
        x = x.broadcast_to(tuple(bshape))
        shp[axis] *= int(repeats)
        return x.reshape(shp)
    return NDArray(torch.repeat_interleave(_t(data), repeats, dim=axis))


def pad(data, mode='constant', pad_width=None, constant_value=0, **kwargs):
    # mxnet pad_width is (before,after) per axis starting from axis 0
    pw = []
    for i in range(len(pad_width) // 2 - 1, -1, -1):
        pw += [pad_width[2 * i], pad_width[2 * i + 1]]
    return NDArray(torch.nn.functional.pad(_t(data), pw, mode=mode if mode != 'constant' else 'constant',
                                           value=constant_value))


def broadcast_to(data, shape, **kwargs):
    return data.broadcast_to(shape)


def broadcast_like(data, like, **kwargs):
    if data.is_native:
        return data.broadcast_to(tuple(like.shape))
    return NDArray(_t(data).broadcast_to(_t(like).shape).contiguous())


def broadcast_axis(data, axis, size, **kwargs):
    x = _t(data)
    shape = list(x.shape)
    axes = axis if isinstance(axis, (list, tuple)) else [axis]
    sizes = size if isinstance(size, (list, tuple)) else [size]
    for a, s in zip(axes, sizes):
        shape[a] = s
    return NDArray(x.broadcast_to(shape).contiguous())


def zeros_like(data, **kwargs):
    if data.is_native:
        return _ninv('zeros_like', [data], {})
    return NDArray(torch.zeros_like(_t(data)))


def ones_like(data, **kwargs):
    if data.is_native:
        return _ninv('ones_like', [data], {})
    return NDArray(torch.ones_like(_t(data)))


def cast(data, dtype, **kwargs):
    return data.astype(dtype)


Cast = cast


def dot(lhs, rhs, transpose_a=False, transpose_b=False, **kwargs):
    if lhs.is_native:
        a = lhs.transpose() if transpose_a else lhs
        if transpose_b:
            return _ninv('dot_nt', [a, rhs], {})
        return _ninv('dot_nn', [a, rhs], {})
    a, b = _t(lhs), _t(rhs)
    if transpose_a:
        a = a.t()
    if transpose_b:
        b = b.t()
    return NDArray(_nn.dot(a.contiguous(), b.contiguous()))


def batch_dot(lhs, rhs, transpose_a=False, transpose_b=False, **kwargs):
    if lhs.is_native:
        a = lhs.transpose((0, 2, 1)) if transpose_a else lhs
        b = rhs.transpose((0, 2, 1)) if transpose_b else rhs
        return _ninv('batch_dot', [a, b], {})
    return NDArray(_nn.batch_dot(_t(lhs), _t(rhs), transpose_a, transpose_b))


def linalg_gemm2(A, B, transpose_a=False, transpose_b=False, alpha=1.0, **kwargs):
    if A.is_native:
        if len(A.shape) == 2:
            r = dot(A, B, transpose_a=transpose_a, transpose_b=transpose_b)
        else:
            r = batch_dot(A, B, transpose_a=transpose_a,
                          transpose_b=transpose_b)
        return r if alpha == 1.0 else r * float(alpha)
    a, b = _t(A), _t(B)
    if transpose_a:
        a = a.transpose(-1, -2)
    if transpose_b:
        b = b.transpose(-1, -2)
    return NDArray(alpha * torch.matmul(a, b))


def SequenceMask(data, sequence_length=None, use_sequence_length=False,
                 value=0.0, axis=0, **kwargs):
    x = _t(data)
    if not use_sequence_length or sequence_length is None:
        return NDArray(x)
    seqlen = _t(sequence_length).long()
    T = x.shape[axis]
    ar = torch.arange(T, device=x.device)
    if axis == 0:
        mask = ar.view(-1, 1) < seqlen.view(1, -1)
        mask = mask.view(T, -1, *([1] * (x.dim() - 2)))
    else:
        mask = ar.view(1, -1) < seqlen.view(-1, 1)
        mask = mask.view(-1, T, *([1] * (x.dim() - 2)))
    return NDArray(torch.where(mask, x, torch.full_like(x, value)))


def sequence_mask(*args, **kwargs):
    return SequenceMask(*args, **kwargs)


# --- random ------------------------------------------------------------------

def _native_rand(op, shape, dtype, ctx, attrs):
    import random as _random
    from .. import _core
    from ..base import core_flag
    from .ndarray import _core_ctx
    if isinstance(shape, int):
        shape = (shape,)
    dt, di = _core_ctx(ctx)
    attrs = dict(attrs)
    attrs.update({'shape': '(' + ','.join(str(s) for s in shape) + ',)',
                  'dtype': str(core_flag(dtype)),
                  'seed': str(_random.getrandbits(48)),
                  '__ctx_gpu__': str(di) if dt == 2 else '-1'})
    return NDArray(_core.invoke(op, [], attrs)[0])


def random_uniform(low=0.0, high=1.0, shape=(1,), dtype='float32', ctx=None, **kwargs):
    from ..base import native_mode
    if native_mode():
        return _native_rand('_random_uniform', shape, dtype, ctx,
                            {'low': str(low), 'high': str(high)})
    from ..context import current_context
    dev = (ctx or current_context()).torch_device
    return NDArray(torch.empty(shape, dtype=torch_dtype(dtype), device=dev).uniform_(low, high))


def random_normal(loc=0.0, scale=1.0, shape=(1,), dtype='float32', ctx=None, **kwargs):
    from ..base import native_mode
    if native_mode():
        return _native_rand('_random_normal', shape, dtype, ctx,
                            {'loc': str(loc), 'scale': str(scale)})
    from ..context import current_context
    dev = (ctx or current_context()).torch_device
    return NDArray(torch.empty(shape, dtype=torch_dtype(dtype), device=dev).normal_(loc, scale))


uniform = random_uniform
normal = random_normal


def random_randint(low, high, shape=(1,), dtype='int32', ctx=None, **kwargs):
    from ..context import current_context
    dev = (ctx or current_context()).torch_device
    return NDArray(torch.randint(low, high, shape, dtype=torch_dtype(dtype), device=dev))


def shuffle(data, **kwargs):
    if data.is_native:
        perm = _np.random.permutation(data.shape[0]).astype('float32')
        from .ndarray import array as _mk
        return take(data, _mk(perm, ctx=data.context))
    x = _t(data)
    return NDArray(x[torch.randperm(x.shape[0], device=x.device)])


def sample_multinomial(data, shape=1, get_prob=False, **kwargs):
    if data.is_native:
        from .ndarray import array as _mk
        p = data.asnumpy()
        n = shape if isinstance(shape, int) else shape[0]
        if p.ndim == 1:
            out = _np.random.choice(p.shape[-1], size=n, p=p / p.sum())
        else:
            out = _np.stack([_np.random.choice(p.shape[-1], size=n,
                                               p=row / row.sum())
                             for row in p])
        return _mk(out.astype('int32'), ctx=data.context, dtype='int32')
    x = _t(data)
    n = shape if isinstance(shape, int) else shape[0]
    return NDArray(torch.multinomial(x, n, replacement=True).to(torch.int32))


def LRN(data, alpha=1e-4, beta=0.75, knorm=2, nsize=5, **kwargs):
    """Local response norm (reference lrn.cc; AlexNet-era)."""
    import torch.nn.functional as F
    t = _t(data)
    y = F.local_response_norm(t.float(), size=nsize, alpha=alpha, beta=beta,
                              k=knorm).to(t.dtype)
    return NDArray(y)


def UpSampling(data, scale=2, sample_type='nearest', **kwargs):
    """(reference upsampling.cc): nearest or bilinear spatial upsampling."""
    import torch.nn.functional as F
    t = _t(data)
    mode = 'nearest' if sample_type == 'nearest' else 'bilinear'
    y = F.interpolate(t.float(), scale_factor=scale, mode=mode,
                      align_corners=False if mode == 'bilinear' else None)
    return NDArray(y.to(t.dtype))


def BilinearResize2D(data, height=None, width=None, scale_height=None,
                     scale_width=None, **kwargs):
    """(reference contrib bilinear_resize.cc)."""
    import torch.nn.functional as F
    t = _t(data)
    if height is None:
        height = int(t.shape[2] * scale_height)
        width = int(t.shape[3] * scale_width)
    y = F.interpolate(t.float(), size=(height, width), mode='bilinear',
                      align_corners=True)
    return NDArray(y.to(t.dtype))


def SequenceLast(data, sequence_length=None, use_sequence_length=False,
                 axis=0):
    """(reference sequence_last.cc): last valid step of [T, N, ...]."""
    t = _t(data)
    if not use_sequence_length or sequence_length is None:
        return NDArray(t.select(axis, t.shape[axis] - 1))
    import torch
    sl = _t(sequence_length).long() - 1
    tt = t.movedim(axis, 0)
    idx = sl.view(-1, *([1] * (tt.dim() - 2))).expand(1, *tt.shape[1:])
    return NDArray(tt.gather(0, idx).squeeze(0))


def SequenceReverse(data, sequence_length=None, use_sequence_length=False,
                    axis=0):
    """(reference sequence_reverse.cc)."""
    import torch
    t = _t(data)
    if not use_sequence_length or sequence_length is None:
        return NDArray(torch.flip(t, dims=[axis]))
    tt = t.movedim(axis, 0).clone()
    sl = _t(sequence_length).long()
    for n in range(tt.shape[1]):
        L = int(sl[n])
        tt[:L, n] = torch.flip(tt[:L, n], dims=[0])
    return NDArray(tt.movedim(0, axis))


def smooth_l1(data, scalar=1.0, **kwargs):
    """(reference smooth_l1 op)."""
    if data.is_native:
        s2 = scalar * scalar
        a = data.abs()
        inside = (a < (1.0 / s2)).astype(str(data.dtype)) \
            if str(data.dtype) != 'float32' else (a < (1.0 / s2))
        quad = data.square() * (0.5 * s2)
        lin = a - (0.5 / s2)
        return inside * quad + (inside * -1.0 + 1.0) * lin
    import torch
    t = _t(data)
    s2 = scalar * scalar
    absd = t.abs()
    y = torch.where(absd < 1.0 / s2, 0.5 * s2 * t * t, absd - 0.5 / s2)
    return NDArray(y)


# ---------------------------------------------------------------------------
# tensor-op tail (reference tensor/matrix_op, spatial ops, special math) —
# library-backed, same dispatch as the rest of the namespace
# ---------------------------------------------------------------------------

def moments(data, axes=None, keepdims=False, **kwargs):
    """(mean, var) pair (reference nn/moments.cc)."""
    if data.is_native:
        ax = tuple(axes) if axes is not None \
            else tuple(range(len(data.shape)))
        m = data.mean(axis=ax, keepdims=True)
        v = ((data - m) ** 2).mean(axis=ax, keepdims=keepdims)
        if not keepdims:
            m = data.mean(axis=ax, keepdims=False)
        return m, v
    t = _t(data)
    dims = list(axes) if axes is not None else list(range(t.dim()))
    mean = t.mean(dim=dims, keepdim=keepdims)
    var = t.var(dim=dims, unbiased=False, keepdim=keepdims)
    return NDArray(mean), NDArray(var)


def SwapAxis(data, dim1=0, dim2=0, **kwargs):
    if data.is_native:
        return data.swapaxes(dim1, dim2)
    return NDArray(_t(data).transpose(dim1, dim2).contiguous())


swapaxes = SwapAxis


def depth_to_space(data, block_size, **kwargs):
    import torch.nn.functional as TF
    return NDArray(TF.pixel_shuffle(_t(data), block_size))


def space_to_depth(data, block_size, **kwargs):
    import torch.nn.functional as TF
    return NDArray(TF.pixel_unshuffle(_t(data), block_size))


def cumsum(a, axis=None, **kwargs):
    if a.is_native:
        from .ndarray import array as _mk
        return _mk(_np.cumsum(a.asnumpy(), axis=axis), ctx=a.context)
    t = _t(a)
    if axis is None:
        return NDArray(t.reshape(-1).cumsum(0))
    return NDArray(t.cumsum(axis))


def cumprod(a, axis=None, **kwargs):
    if a.is_native:
        from .ndarray import array as _mk
        return _mk(_np.cumprod(a.asnumpy(), axis=axis), ctx=a.context)
    t = _t(a)
    if axis is None:
        return NDArray(t.reshape(-1).cumprod(0))
    return NDArray(t.cumprod(axis))


def diag(data, k=0, **kwargs):
    if data.is_native:
        from .ndarray import array as _mk
        arr = data.asnumpy()
        r = _np.diagonal(arr, offset=k) if arr.ndim >= 2 \
            else _np.diag(arr, k)
        return _mk(_np.ascontiguousarray(r), ctx=data.context)
    t = _t(data)
    return NDArray(torch.diagonal(t, offset=k).contiguous() if t.dim() >= 2
                   else torch.diag(t, k))


def trace(data, offset=0, **kwargs):
    if data.is_native:
        from .ndarray import array as _mk
        return _mk(_np.trace(data.asnumpy(), offset=offset)
                   .astype('float32').reshape(-1), ctx=data.context)
    return NDArray(torch.diagonal(_t(data), offset=offset).sum(-1))


def meshgrid(*arrays, indexing='xy', **kwargs):
    if arrays and getattr(arrays[0], 'is_native', False):
        from .ndarray import array as _mk
        outs = _np.meshgrid(*[a.asnumpy() for a in arrays],
                            indexing=indexing)
        return [_mk(_np.ascontiguousarray(o), ctx=arrays[0].context)
                for o in outs]
    outs = torch.meshgrid(*[_t(a) for a in arrays], indexing=indexing)
    return [NDArray(o.contiguous()) for o in outs]


def searchsorted(sorted_sequence, values, right=False, **kwargs):
    if sorted_sequence.is_native:
        from .ndarray import array as _mk
        out = _np.searchsorted(sorted_sequence.asnumpy(),
                               values.asnumpy(),
                               side='right' if right else 'left')
        return _mk(out.astype('float32'),
                   ctx=sorted_sequence.context)
    return NDArray(torch.searchsorted(_t(sorted_sequence), _t(values),
                                      right=right))


def bincount(x, weights=None, minlength=0, **kwargs):
    if x.is_native:
        from .ndarray import array as _mk
        out = _np.bincount(x.asnumpy().astype('int64'),
                           weights.asnumpy() if weights is not None
                           else None, minlength)
        return _mk(out.astype('float32'), ctx=x.context)
    return NDArray(torch.bincount(
        _t(x).long(), _t(weights) if weights is not None else None,
        minlength))


def digamma(data, **kwargs):
    if data.is_native:
        import scipy.special as _sp
        from .ndarray import array as _mk
        return _mk(_sp.digamma(data.asnumpy()).astype('float32'),
                   ctx=data.context)
    return NDArray(torch.digamma(_t(data)))


def ravel_multi_index(data, shape, **kwargs):
    t = _t(data).long()  # [ndim, n]
    strides = []
    acc = 1
    for s in reversed(shape):
        strides.append(acc)
        acc *= s
    strides = torch.tensor(list(reversed(strides)), device=t.device)
    return NDArray((t * strides[:, None]).sum(0))


def unravel_index(data, shape, **kwargs):
    t = _t(data).long()
    out = []
    for s in reversed(shape):
        out.append(t % s)
        t = t // s
    return NDArray(torch.stack(list(reversed(out)), dim=0))


def GridGenerator(data, transform_type='affine', target_shape=None, **kwargs):
    """Affine sampling grid (reference spatial_transformer GridGenerator)."""
    import torch.nn.functional as TF
    t = _t(data)
    H, W = target_shape
    theta = t.reshape(-1, 2, 3).float()
    grid = TF.affine_grid(theta, [theta.shape[0], 1, H, W],
                          align_corners=False)
    # reference layout: [N, 2, H, W] (x, y)
    return NDArray(grid.permute(0, 3, 1, 2).contiguous())


def BilinearSampler(data, grid, **kwargs):
    """Sample data at grid locations (reference bilinear_sampler.cc)."""
    import torch.nn.functional as TF
    t = _t(data)
    g = _t(grid)
    g4 = g.permute(0, 2, 3, 1) if g.shape[1] == 2 else g
    return NDArray(TF.grid_sample(t.float(), g4.float(), mode='bilinear',
                                  align_corners=False).to(t.dtype))


def SpatialTransformer(data, loc, target_shape=None,
                       transform_type='affine',
                       sampler_type='bilinear', **kwargs):
    """Affine spatial transformer = GridGenerator + BilinearSampler
    (reference spatial_transformer.cc)."""
    grid = GridGenerator(loc, transform_type, target_shape)
    return BilinearSampler(data, grid)
