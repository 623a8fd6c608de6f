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


class _SyncBNFn(torch.autograd.Function):
    """Cross-rank BN: batch statistics (and their gradients) are
    all-reduced over torch.distributed (RCCL on GPU / gloo on CPU) —
    the reference used an intra-process barrier registry
    (sync_batch_norm-inl.h:79-195); one collective per direction here."""

    @staticmethod
    def forward(ctx, x, gamma, beta, rmean, rvar, momentum, eps, axis):
        import torch.distributed as dist
        dims = [d for d in range(x.dim()) if d != axis % x.dim()]
        n_local = x.numel() // x.shape[axis]
        xs = x.float()
        stats = torch.cat([xs.sum(dims), (xs * xs).sum(dims),
                           torch.tensor([float(n_local)], device=x.device)])
        world = dist.get_world_size() if dist.is_initialized() else 1
        if world > 1:
            dist.all_reduce(stats)
        C = x.shape[axis]
        n_total = stats[-1]
        mean = stats[:C] / n_total
        var = (stats[C:2 * C] / n_total - mean * mean).clamp_min(0)
        istd = (var + eps).rsqrt()
        with torch.no_grad():
            unbias = var * n_total / (n_total - 1).clamp_min(1)
            rmean.mul_(momentum).add_((1 - momentum) * mean)
            rvar.mul_(momentum).add_((1 - momentum) * unbias)
        shape = [1] * x.dim()
        shape[axis % x.dim()] = C
        xhat = (xs - mean.view(shape)) * istd.view(shape)
        y = (xhat * gamma.view(shape) + beta.view(shape)).to(x.dtype)
        ctx.save_for_backward(x, gamma, mean, istd)
        ctx.axis, ctx.n_total, ctx.world = axis, float(n_total), world
        return y

    @staticmethod
    def backward(ctx, dy):
        import torch.distributed as dist
        x, gamma, mean, istd = ctx.saved_tensors
        axis = ctx.axis % x.dim()
        dims = [d for d in range(x.dim()) if d != axis]
        C = x.shape[axis]
        shape = [1] * x.dim()
        shape[axis] = C
        g = dy.float()
        xhat = (x.float() - mean.view(shape)) * istd.view(shape)
        s1 = g.sum(dims)
        s2 = (g * xhat).sum(dims)
        if ctx.world > 1:
            pair = torch.cat([s1, s2])
            dist.all_reduce(pair)
            s1, s2 = pair[:C], pair[C:]
        invN = 1.0 / ctx.n_total
        dx = (gamma.view(shape) * istd.view(shape) *
              (g - invN * s1.view(shape) - xhat * invN * s2.view(shape)))
        # s1/s2 are GLOBAL sums after the all-reduce (identical on every
        # rank); the trainer AVERAGES grads across ranks, which preserves
        # them — so return the global sums directly.
        return dx.to(x.dtype), s2, s1, None, None, None, None, None


class SyncBatchNorm(BatchNorm):
    """Cross-rank synchronized BN (see _SyncBNFn)."""

    def __init__(self, in_channels=0, num_devices=None, **kwargs):
        super().__init__(in_channels=in_channels, **kwargs)

    def forward(self, x, residual=None):
        import torch.distributed as dist
        from ... import autograd as _ag
        self._finish_deferred(x)
        if not (_ag.is_training() and dist.is_initialized()
                and dist.get_world_size() > 1):
            return super().forward(x, residual)
        ctx = self._param_ctx((x,))
        from ...ndarray.ndarray import NDArray
        t = x.handle if hasattr(x, 'handle') else x
        y = _SyncBNFn.apply(
            t, self.gamma.data(ctx).handle.float(),
            self.beta.data(ctx).handle.float(),
            self.running_mean.data(ctx).handle,
            self.running_var.data(ctx).handle,
            self._momentum, self._epsilon, self._axis)
        if residual is not None:
            y = y + (residual.handle if hasattr(residual, 'handle')
                     else residual)
        if self._fuse_relu:
            y = torch.relu(y)
        return NDArray(y)


class Embedding(HybridBlock):
    def __init__(self, input_dim, output_dim, dtype='float32',
                 weight_initializer=None, sparse_grad=False, **kwargs):
        super().__init__(**kwargs)
        self._input_dim = input_dim
        self._output_dim = output_dim
        self._sparse_grad = sparse_grad
        self.weight = Parameter('weight', shape=(input_dim, output_dim),
                                dtype=dtype, init=weight_initializer,
                                grad_stype='row_sparse' if sparse_grad
                                else 'default')

    def hybrid_forward(self, F, x, weight):
        return F.Embedding(x, weight, input_dim=self._input_dim,
                           output_dim=self._output_dim,
                           sparse_grad=self._sparse_grad)

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

    def cast(self, dtype):
        """LN affine params stay fp32 under half-precision training (same
        fp32-list treatment as BatchNorm; the fused LN kernel computes
        stats in fp32 and expects fp32 gamma/beta)."""
        if dtype in ('float16', 'bfloat16'):
            return self
        return super().cast(dtype)


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
        if getattr(x, 'is_native', False):
            # composed group-norm over dual-backend nd ops (NCHW-style
            # axis=1 channels, reference group_norm.cc semantics)
            N, C = x.shape[0], x.shape[1]
            G = self._num_groups
            inner = 1
            for d in x.shape[2:]:
                inner *= d
            xg = x.reshape(N, G, (C // G) * inner)
            m = xg.mean(axis=2, keepdims=True)
            d = xg - m
            v = d.square().mean(axis=2, keepdims=True)
            y = (d / (v + self._epsilon).sqrt()).reshape(*x.shape)
            gshape = (1, C) + (1,) * (len(x.shape) - 2)
            g = self.gamma.data(ctx).reshape(gshape)
            b = self.beta.data(ctx).reshape(gshape)
            return y * g + b
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
        if getattr(x, 'is_native', False):
            # per-sample per-channel normalization over spatial axes
            axes = tuple(d for d in range(len(x.shape))
                         if d not in (0, self._axis))
            m = x.mean(axis=axes, keepdims=True)
            d = x - m
            v = d.square().mean(axis=axes, keepdims=True)
            y = d / (v + self._epsilon).sqrt()
            gshape = [1] * len(x.shape)
            gshape[self._axis] = x.shape[self._axis]
            g = self.gamma.data(ctx).reshape(gshape)
            b = self.beta.data(ctx).reshape(gshape)
            return y * g + b
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
