"""Gluon basic layers (reference python/mxnet/gluon/nn/basic_layers.py)."""
import torch

from ..block import HybridBlock, Block
from ..parameter import Parameter
from ... import initializer as init


class Dense(HybridBlock):
    """y = act(x W^T + b) — FC on the MFMA GEMM kernel when on GPU."""

    def __init__(self, units, activation=None, use_bias=True, flatten=True,
                 dtype='float32', weight_initializer=None,
                 bias_initializer='zeros', in_units=0, **kwargs):
        super().__init__(**kwargs)
        self._units = units
        self._in_units = in_units
        self._flatten = flatten
        self._act_type = activation
        self.weight = Parameter('weight', shape=(units, in_units), dtype=dtype,
                                init=weight_initializer, allow_deferred_init=True)
        if use_bias:
            self.bias = Parameter('bias', shape=(units,), dtype=dtype,
                                  init=init.create(bias_initializer),
                                  allow_deferred_init=True)
        else:
            self.bias = None

    def infer_shape(self, x):
        in_units = x.shape[-1] if not self._flatten or x.ndim == 2 else \
            int(torch.tensor(x.shape[1:]).prod().item())
        self.weight.shape = (self._units, in_units)

    def hybrid_forward(self, F, x, weight, bias=None):
        out = F.FullyConnected(x, weight, bias, num_hidden=self._units,
                               no_bias=bias is None, flatten=self._flatten)
        if self._act_type:
            out = F.Activation(out, act_type=self._act_type)
        return out

    def __repr__(self):
        return f'Dense({self._units}, act={self._act_type})'


class Dropout(HybridBlock):
    def __init__(self, rate, axes=(), **kwargs):
        super().__init__(**kwargs)
        self._rate = rate

    def hybrid_forward(self, F, x):
        if self._rate == 0:
            return x
        return F.Dropout(x, p=self._rate)

    def __repr__(self):
        return f'Dropout(p={self._rate})'


class BatchNorm(HybridBlock):
    """BatchNorm; axis=1 → NCHW, axis=-1/3 → NHWC (GPU hot path).

    Optional ``fuse_relu`` / residual-add fusion used by the model zoo's
    ResNet hot path (reference BatchNormReLU; batch_norm.cu kernels)."""

    def __init__(self, axis=1, momentum=0.9, epsilon=1e-5, center=True,
                 scale=True, use_global_stats=False, beta_initializer='zeros',
                 gamma_initializer='ones', running_mean_initializer='zeros',
                 running_variance_initializer='ones', in_channels=0,
                 fuse_relu=False, **kwargs):
        super().__init__(**kwargs)
        self._axis = axis
        self._momentum = momentum
        self._epsilon = epsilon
        self._use_global_stats = use_global_stats
        self._fuse_relu = fuse_relu
        self.gamma = Parameter('gamma', shape=(in_channels,),
                               init=init.create(gamma_initializer),
                               allow_deferred_init=True,
                               differentiable=scale)
        self.beta = Parameter('beta', shape=(in_channels,),
                              init=init.create(beta_initializer),
                              allow_deferred_init=True,
                              differentiable=center)
        self.running_mean = Parameter('running_mean', shape=(in_channels,),
                                      grad_req='null',
                                      init=init.create(running_mean_initializer),
                                      allow_deferred_init=True)
        self.running_var = Parameter('running_var', shape=(in_channels,),
                                     grad_req='null',
                                     init=init.create(running_variance_initializer),
                                     allow_deferred_init=True)

    def infer_shape(self, x, *args):
        c = x.shape[self._axis]
        for p in (self.gamma, self.beta, self.running_mean, self.running_var):
            p.shape = (c,)

    def hybrid_forward(self, F, x, residual=None, gamma=None, beta=None,
                       running_mean=None, running_var=None):
        layout = 'NHWC' if self._axis in (-1, 3) and (
            not hasattr(x, 'ndim') or x.ndim == 4) else None
        return F.BatchNorm(x, gamma, beta, running_mean, running_var,
                           eps=self._epsilon, momentum=self._momentum,
                           use_global_stats=self._use_global_stats,
                           axis=self._axis, layout=layout,
                           fuse_relu=self._fuse_relu, residual=residual)

    def forward(self, x, residual=None):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        params = self._param_kwargs(ctx)
        from ...ndarray import ops as F
        return self.hybrid_forward(F, x, residual, **params)

    def cast(self, dtype):
        """BN statistics/affine params stay fp32 under fp16 training
        (reference AMP keeps BatchNorm in the fp32 list, amp/lists);
        also lets the GPU kernels skip a per-call cast."""
        if dtype in ('float16', 'bfloat16'):
            return self
        return super().cast(dtype)

    def __repr__(self):
        return f'BatchNorm(axis={self._axis}, fuse_relu={self._fuse_relu})'


class BatchNormReLU(BatchNorm):
    def __init__(self, **kwargs):
        kwargs['fuse_relu'] = True
        super().__init__(**kwargs)


class SyncBatchNorm(BatchNorm):
    """Cross-rank synchronized BN. On the 1-proc-per-GPU RCCL layout,
    stats are all-reduced over torch.distributed (reference
    sync_batch_norm-inl.h used an intra-process barrier registry)."""

    def __init__(self, in_channels=0, num_devices=None, **kwargs):
        super().__init__(in_channels=in_channels, **kwargs)
        # round-1: per-rank stats (correct single-process); RCCL stat
        # all-reduce lands with the distributed trainer work.


class Embedding(HybridBlock):
    def __init__(self, input_dim, output_dim, dtype='float32',
                 weight_initializer=None, sparse_grad=False, **kwargs):
        super().__init__(**kwargs)
        self._input_dim = input_dim
        self._output_dim = output_dim
        self.weight = Parameter('weight', shape=(input_dim, output_dim),
                                dtype=dtype, init=weight_initializer)

    def hybrid_forward(self, F, x, weight):
        return F.Embedding(x, weight, input_dim=self._input_dim,
                           output_dim=self._output_dim)

    def __repr__(self):
        return f'Embedding({self._input_dim} -> {self._output_dim})'


class LayerNorm(HybridBlock):
    def __init__(self, axis=-1, epsilon=1e-5, center=True, scale=True,
                 beta_initializer='zeros', gamma_initializer='ones',
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        self._axis = axis
        self._epsilon = epsilon
        self.gamma = Parameter('gamma', shape=(in_channels,),
                               init=init.create(gamma_initializer),
                               allow_deferred_init=True)
        self.beta = Parameter('beta', shape=(in_channels,),
                              init=init.create(beta_initializer),
                              allow_deferred_init=True)

    def infer_shape(self, x):
        c = x.shape[self._axis]
        self.gamma.shape = (c,)
        self.beta.shape = (c,)

    def hybrid_forward(self, F, x, gamma, beta):
        return F.LayerNorm(x, gamma, beta, axis=self._axis, eps=self._epsilon)


class GroupNorm(HybridBlock):
    def __init__(self, num_groups=1, epsilon=1e-5, center=True, scale=True,
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        self._num_groups = num_groups
        self._epsilon = epsilon
        self.gamma = Parameter('gamma', shape=(in_channels,), init=init.One(),
                               allow_deferred_init=True)
        self.beta = Parameter('beta', shape=(in_channels,), init=init.Zero(),
                              allow_deferred_init=True)

    def infer_shape(self, x):
        self.gamma.shape = (x.shape[1],)
        self.beta.shape = (x.shape[1],)

    def forward(self, x):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        g = self.gamma.data(ctx)._t
        b = self.beta.data(ctx)._t
        from ...ndarray.ndarray import NDArray
        y = torch.nn.functional.group_norm(x._t, self._num_groups, g, b,
                                           self._epsilon)
        return NDArray(y)


class InstanceNorm(HybridBlock):
    def __init__(self, axis=1, epsilon=1e-5, center=True, scale=False,
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        self._axis = axis
        self._epsilon = epsilon
        self.gamma = Parameter('gamma', shape=(in_channels,), init=init.One(),
                               allow_deferred_init=True)
        self.beta = Parameter('beta', shape=(in_channels,), init=init.Zero(),
                              allow_deferred_init=True)

    def infer_shape(self, x):
        self.gamma.shape = (x.shape[self._axis],)
        self.beta.shape = (x.shape[self._axis],)

    def forward(self, x):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        from ...ndarray.ndarray import NDArray
        y = torch.nn.functional.instance_norm(
            x._t, weight=self.gamma.data(ctx)._t, bias=self.beta.data(ctx)._t,
            eps=self._epsilon)
        return NDArray(y)


class Flatten(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.Flatten(x)

    def __repr__(self):
        return 'Flatten'


class Activation(HybridBlock):
    def __init__(self, activation, **kwargs):
        super().__init__(**kwargs)
        self._act_type = activation

    def hybrid_forward(self, F, x):
        return F.Activation(x, act_type=self._act_type)

    def __repr__(self):
        return f'Activation({self._act_type})'


class Lambda(Block):
    def __init__(self, function, **kwargs):
        super().__init__(**kwargs)
        self._func = function

    def forward(self, *args):
        from ...ndarray import ops as F
        if isinstance(self._func, str):
            return getattr(F, self._func)(*args)
        return self._func(*args)


class HybridLambda(HybridBlock):
    def __init__(self, function, **kwargs):
        super().__init__(**kwargs)
        self._func = function

    def hybrid_forward(self, F, *args):
        if isinstance(self._func, str):
            return getattr(F, self._func)(*args)
        return self._func(F, *args)


class Identity(HybridBlock):
    def hybrid_forward(self, F, x):
        return x
