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


class Conv2DTranspose(HybridBlock):
    """Transposed conv (reference Deconvolution, deconvolution.cc).

    MI355X design: the forward IS the conv backward-data problem, so it
    runs the implicit-GEMM conv_bwd_data kernel; its backward reuses the
    conv fwd / bwd_weight kernels with the roles swapped."""

    def __init__(self, channels, kernel_size, strides=(1, 1), padding=(0, 0),
                 output_padding=(0, 0), dilation=(1, 1), groups=1,
                 layout='NCHW', in_channels=0, activation=None,
                 use_bias=True, weight_initializer=None,
                 bias_initializer='zeros', **kwargs):
        super().__init__(**kwargs)
        self._channels = channels
        self._kernel = _pair(kernel_size)
        self._strides = _pair(strides)
        self._padding = _pair(padding)
        self._out_pad = _pair(output_padding)
        self._dilation = _pair(dilation)
        self._groups = groups
        self._layout = layout
        self._act_type = activation
        kh, kw = self._kernel
        # weight layout matches the reference Deconvolution: [in, out//g, kh, kw]
        # (NHWC path keeps channels last: [in, kh, kw, out//g])
        if layout == 'NHWC':
            wshape = (in_channels, kh, kw,
                      channels // groups if channels else 0)
        else:
            wshape = (in_channels, channels // groups, kh, kw)
        self.weight = Parameter('weight', shape=wshape, dtype='float32',
                                init=weight_initializer,
                                allow_deferred_init=True)
        self.bias = Parameter('bias', shape=(channels,), dtype='float32',
                              init=init.create(bias_initializer),
                              allow_deferred_init=True) if use_bias else None

    def infer_shape(self, x):
        c_axis = 3 if self._layout == 'NHWC' else 1
        in_c = x.shape[c_axis]
        kh, kw = self._kernel
        if self._layout == 'NHWC':
            self.weight.shape = (in_c, kh, kw, self._channels // self._groups)
        else:
            self.weight.shape = (in_c, self._channels // self._groups, kh, kw)

    def forward(self, x):
        if getattr(x, 'is_native', False):
            raise NotImplementedError(
                '%s runs on the torch frontend (set_native(False)); the '
                'native runtime covers NHWC Conv1D/2D via the MFMA igemm '
                'kernels' % type(self).__name__)
        from ...ndarray.ndarray import NDArray
        from ...ops import nn as _onn
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        w = self.weight.data(ctx).handle
        b = self.bias.data(ctx).handle if self.bias is not None else None
        t = x.handle if hasattr(x, 'handle') else x
        if b is not None:
            b = b.to(t.dtype)
        y = _onn.deconv2d(t, w.to(t.dtype), b, self._strides, self._padding,
                          self._out_pad, self._dilation, self._groups,
                          layout=self._layout)
        if self._act_type:
            y = _onn.activation(y, self._act_type)
        return NDArray(y)


class Conv1DTranspose(Conv2DTranspose):
    def __init__(self, channels, kernel_size, strides=1, padding=0,
                 output_padding=0, **kwargs):
        k = kernel_size if isinstance(kernel_size, int) else kernel_size[0]
        s = strides if isinstance(strides, int) else strides[0]
        p = padding if isinstance(padding, int) else padding[0]
        op = output_padding if isinstance(output_padding, int) else output_padding[0]
        super().__init__(channels, (1, k), (1, s), (0, p), (0, op), **kwargs)


class Conv3D(HybridBlock):
    """3-D conv (NCDHW); volumetric nets are outside the MFMA hot set, so
    this runs the library conv3d path (reference also used cuDNN here)."""

    def __init__(self, channels, kernel_size, strides=(1, 1, 1),
                 padding=(0, 0, 0), dilation=(1, 1, 1), groups=1,
                 layout='NCDHW', in_channels=0, activation=None,
                 use_bias=True, weight_initializer=None,
                 bias_initializer='zeros', **kwargs):
        super().__init__(**kwargs)
        def _triple(v):
            return tuple(v) if isinstance(v, (tuple, list)) else (v,) * 3
        self._channels = channels
        self._kernel = _triple(kernel_size)
        self._strides = _triple(strides)
        self._padding = _triple(padding)
        self._dilation = _triple(dilation)
        self._groups = groups
        self._act_type = activation
        kd, kh, kw = self._kernel
        self.weight = Parameter(
            'weight',
            shape=(channels, in_channels // groups if in_channels else 0,
                   kd, kh, kw),
            init=weight_initializer, allow_deferred_init=True)
        self.bias = Parameter('bias', shape=(channels,),
                              init=init.create(bias_initializer),
                              allow_deferred_init=True) if use_bias else None

    def infer_shape(self, x):
        kd, kh, kw = self._kernel
        self.weight.shape = (self._channels, x.shape[1] // self._groups,
                             kd, kh, kw)

    def forward(self, x):
        if getattr(x, 'is_native', False):
            raise NotImplementedError(
                '%s runs on the torch frontend (set_native(False)); the '
                'native runtime covers NHWC Conv1D/2D via the MFMA igemm '
                'kernels' % type(self).__name__)
        import torch.nn.functional as F
        from ...ndarray.ndarray import NDArray
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        w = self.weight.data(ctx).handle
        b = self.bias.data(ctx).handle if self.bias is not None else None
        t = x.handle if hasattr(x, 'handle') else x
        y = F.conv3d(t, w.to(t.dtype), b.to(t.dtype) if b is not None else None,
                     stride=self._strides, padding=self._padding,
                     dilation=self._dilation, groups=self._groups)
        if self._act_type:
            from ...ops import nn as _onn
            y = _onn.activation(y, self._act_type)
        return NDArray(y)


class MaxPool3D(HybridBlock):
    def __init__(self, pool_size=(2, 2, 2), strides=None, padding=0,
                 layout='NCDHW', **kwargs):
        super().__init__(**kwargs)
        self._k = pool_size if isinstance(pool_size, tuple) else (pool_size,) * 3
        self._s = strides or self._k
        self._p = padding if isinstance(padding, tuple) else (padding,) * 3

    def forward(self, x):
        import torch.nn.functional as F
        from ...ndarray.ndarray import NDArray
        return NDArray(F.max_pool3d(x.handle, self._k, self._s, self._p))


class AvgPool3D(MaxPool3D):
    def forward(self, x):
        import torch.nn.functional as F
        from ...ndarray.ndarray import NDArray
        return NDArray(F.avg_pool3d(x.handle, self._k, self._s, self._p))


class PixelShuffle2D(HybridBlock):
    """(reference contrib PixelShuffle): [N, C*r^2, H, W] -> [N, C, H*r, W*r]."""

    def __init__(self, factor, **kwargs):
        super().__init__(**kwargs)
        self._factor = factor if isinstance(factor, int) else factor[0]

    def forward(self, x):
        import torch.nn.functional as F
        from ...ndarray.ndarray import NDArray
        return NDArray(F.pixel_shuffle(x.handle, self._factor))
