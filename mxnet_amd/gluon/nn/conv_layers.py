"""Gluon convolution / pooling layers
(reference python/mxnet/gluon/nn/conv_layers.py).

``layout='NHWC'`` is the MI355X hot path: the implicit-GEMM MFMA conv
kernels consume channel-contiguous tensors (SURVEY §2.3 conv row)."""
import numpy as _np

from ..block import HybridBlock
from ..parameter import Parameter
from ... import initializer as init


def _pair(v):
    if isinstance(v, (tuple, list)):
        return tuple(int(x) for x in v)
    return (int(v), int(v))


class _Conv(HybridBlock):
    def __init__(self, channels, kernel_size, strides, padding, dilation,
                 groups, layout, in_channels=0, activation=None, use_bias=True,
                 weight_initializer=None, bias_initializer='zeros',
                 dtype='float32', **kwargs):
        super().__init__(**kwargs)
        self._channels = channels
        self._in_channels = in_channels
        self._kernel = _pair(kernel_size)
        self._strides = _pair(strides)
        self._padding = _pair(padding)
        self._dilation = _pair(dilation)
        self._groups = groups
        self._layout = layout
        self._act_type = activation
        kh, kw = self._kernel
        if layout == 'NHWC':
            wshape = (channels, kh, kw, in_channels // groups if in_channels else 0)
        else:
            wshape = (channels, in_channels // groups if in_channels else 0, kh, kw)
        self.weight = Parameter('weight', shape=wshape, dtype=dtype,
                                init=weight_initializer, allow_deferred_init=True)
        if use_bias:
            self.bias = Parameter('bias', shape=(channels,), dtype=dtype,
                                  init=init.create(bias_initializer),
                                  allow_deferred_init=True)
        else:
            self.bias = None

    def infer_shape(self, x):
        c_axis = 3 if self._layout == 'NHWC' else 1
        in_c = x.shape[c_axis]
        kh, kw = self._kernel
        if self._layout == 'NHWC':
            self.weight.shape = (self._channels, kh, kw, in_c // self._groups)
        else:
            self.weight.shape = (self._channels, in_c // self._groups, kh, kw)

    def hybrid_forward(self, F, x, weight, bias=None):
        out = F.Convolution(x, weight, bias, kernel=self._kernel,
                            stride=self._strides, dilate=self._dilation,
                            pad=self._padding, num_filter=self._channels,
                            num_group=self._groups, no_bias=bias is None,
                            layout=self._layout)
        if self._act_type:
            out = F.Activation(out, act_type=self._act_type)
        return out

    def __repr__(self):
        return (f'{type(self).__name__}({self._channels}, kernel={self._kernel}, '
                f'stride={self._strides}, layout={self._layout})')


class Conv2D(_Conv):
    def __init__(self, channels, kernel_size, strides=(1, 1), padding=(0, 0),
                 dilation=(1, 1), groups=1, layout='NCHW', **kwargs):
        super().__init__(channels, kernel_size, strides, padding, dilation,
                         groups, layout, **kwargs)


class Conv1D(_Conv):
    """1-D conv implemented on the 2-D kernels with H=1."""

    def __init__(self, channels, kernel_size, strides=1, padding=0,
                 dilation=1, groups=1, layout='NCW', **kwargs):
        k = kernel_size if isinstance(kernel_size, int) else kernel_size[0]
        s = strides if isinstance(strides, int) else strides[0]
        p = padding if isinstance(padding, int) else padding[0]
        d = dilation if isinstance(dilation, int) else dilation[0]
        super().__init__(channels, (1, k), (1, s), (0, p), (1, d), groups,
                         'NCHW' if layout == 'NCW' else 'NHWC', **kwargs)
        self._orig_layout = layout

    def infer_shape(self, x):
        in_c = x.shape[1] if self._orig_layout == 'NCW' else x.shape[2]
        kh, kw = self._kernel
        if self._layout == 'NHWC':
            self.weight.shape = (self._channels, kh, kw, in_c // self._groups)
        else:
            self.weight.shape = (self._channels, in_c // self._groups, kh, kw)

    def forward(self, x):
        if self._orig_layout == 'NCW':
            x3 = x.expand_dims(2)  # N,C,1,W
            self._finish_deferred(x3)
            ctx = self._param_ctx((x,))
            params = self._param_kwargs(ctx)
            from ...ndarray import ops as F
            y = self.hybrid_forward(F, x3, **params)
            return y.squeeze(2)
        raise NotImplementedError(self._orig_layout)


class _Pool(HybridBlock):
    _kind = 'max'

    def __init__(self, pool_size=(2, 2), strides=None, padding=0,
                 layout='NCHW', ceil_mode=False, count_include_pad=True,
                 global_pool=False, **kwargs):
        super().__init__(**kwargs)
        self._pool_size = _pair(pool_size)
        self._strides = _pair(strides) if strides is not None else self._pool_size
        self._padding = _pair(padding)
        self._layout = layout
        self._global = global_pool
        self._cip = count_include_pad

    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=self._pool_size, pool_type=self._kind,
                         stride=self._strides, pad=self._padding,
                         global_pool=self._global, layout=self._layout,
                         count_include_pad=self._cip)

    def __repr__(self):
        return (f'{type(self).__name__}(size={self._pool_size}, '
                f'stride={self._strides}, layout={self._layout})')


class MaxPool2D(_Pool):
    _kind = 'max'


class AvgPool2D(_Pool):
    _kind = 'avg'


class GlobalMaxPool2D(_Pool):
    _kind = 'max'

    def __init__(self, layout='NCHW', **kwargs):
        super().__init__((1, 1), layout=layout, global_pool=True, **kwargs)


class GlobalAvgPool2D(_Pool):
    _kind = 'avg'

    def __init__(self, layout='NCHW', **kwargs):
        super().__init__((1, 1), layout=layout, global_pool=True, **kwargs)
