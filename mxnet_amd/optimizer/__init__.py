"""Optimizers (reference python/mxnet/optimizer/*.py + fused update ops
src/operator/optimizer_op.cc:49-1044).

Update math runs per-parameter on the parameter's device: on GPU the
Trainer routes whole parameter groups through the native multi-tensor
fused update kernels (one launch for many tensors, reference
multi_sgd/preloaded_multi_sgd); the per-parameter torch expressions here
are the CPU oracle and the fallback.  Multi-precision (fp16/bf16 weights
with fp32 master copy, reference mp_* ops) is supported by every
optimizer via ``multi_precision``.
"""
import math

import torch

from ..ndarray.ndarray import NDArray

_OPT_REGISTRY = {}


def register(cls):
    _OPT_REGISTRY[cls.__name__.lower()] = cls
    return cls


def create(name, **kwargs):
    if isinstance(name, Optimizer):
        return name
    return _OPT_REGISTRY[name.lower()](**kwargs)


class Optimizer:
    def __init__(self, learning_rate=0.01, wd=0.0, rescale_grad=1.0,
                 clip_gradient=None, lr_scheduler=None, multi_precision=False,
                 param_dict=None, **kwargs):
        self.lr = learning_rate
        self.wd = wd
        self.rescale_grad = rescale_grad
        self.clip_gradient = clip_gradient
        self.lr_scheduler = lr_scheduler
        self.multi_precision = multi_precision
        self.num_update = 0
        self.param_dict = param_dict or {}
        self._index_update_count = {}

    # -- learning-rate plumbing -----------------------------------------
    def _get_lr(self, index):
        lr = self.lr_scheduler(self.num_update) if self.lr_scheduler else self.lr
        p = self.param_dict.get(index)
        if p is not None:
            lr *= getattr(p, 'lr_mult', 1.0)
        return lr

    def _get_wd(self, index):
        wd = self.wd
        p = self.param_dict.get(index)
        if p is not None:
            wd *= getattr(p, 'wd_mult', 1.0)
        return wd

    def _update_count(self, index):
        self._index_update_count[index] = self._index_update_count.get(index, 0) + 1
        self.num_update = max(self.num_update, self._index_update_count[index])

    def set_learning_rate(self, lr):
        self.lr = lr

    @property
    def learning_rate(self):
        return self.lr_scheduler(self.num_update) if self.lr_scheduler else self.lr

    # -- state ----------------------------------------------------------
    def create_state(self, index, weight):
        return None

    def create_state_multi_precision(self, index, weight):
        """fp32 master weight for low-precision params (mp_* reference ops)."""
        if isinstance(weight, NDArray) and weight.is_native:
            if self.multi_precision and str(weight.dtype) in (
                    'float16', 'bfloat16'):
                master = weight.astype('float32')
                return (master, self.create_state(index, master))
            return self.create_state(index, weight)
        w = weight._t if isinstance(weight, NDArray) else weight
        if self.multi_precision and w.dtype in (torch.float16, torch.bfloat16):
            master = w.detach().float().clone()
            return (master, self.create_state(index, NDArray(master)))
        return self.create_state(index, weight)

    def _preprocess_grad(self, grad):
        g = grad.float() * self.rescale_grad
        if self.clip_gradient is not None:
            g = g.clamp(-self.clip_gradient, self.clip_gradient)
        return g

    def update(self, index, weight, grad, state):
        raise NotImplementedError

    def update_multi_precision(self, index, weight, grad, state):
        w = weight._t if isinstance(weight, NDArray) else weight
        if self.multi_precision and isinstance(state, tuple) and \
                isinstance(state[0], torch.Tensor) and state[0].dtype is torch.float32 \
                and w.dtype in (torch.float16, torch.bfloat16):
            master, inner = state
            self.update(index, NDArray(master), grad, inner)
            with torch.no_grad():
                w.copy_(master.to(w.dtype))
            return
        self.update(index, weight, grad, state)


@register
class SGD(Optimizer):
    """SGD with momentum (reference sgd_mom_update, optimizer_op-inl.h)."""

    def __init__(self, momentum=0.0, lazy_update=True, **kwargs):
        super().__init__(**kwargs)
        self.momentum = momentum
        self.lazy_update = lazy_update

    def create_state(self, index, weight):
        if self.momentum == 0:
            return None
        if isinstance(weight, NDArray) and weight.is_native:
            f32 = weight if str(weight.dtype) == 'float32' \
                else weight.astype('float32')
            from ..ndarray.ndarray import zeros_like as _zl
            return _zl(f32)
        w = weight._t if isinstance(weight, NDArray) else weight
        return torch.zeros_like(w, dtype=torch.float32)

    def update(self, index, weight, grad, state):
        from ..ndarray.sparse import RowSparseNDArray
        if isinstance(grad, RowSparseNDArray):
            return self._update_row_sparse(index, weight, grad, state)
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g)
            g = g + wd * w.float()
            if state is not None:
                # reference sgd_mom_update folds lr INTO the momentum
                # buffer (mom = mu*mom - lr*g; w += mom) so trajectories
                # track the reference bit-for-bit under lr schedules
                state.mul_(self.momentum).sub_(lr * g)
                w.add_(state.to(w.dtype))
            else:
                w.sub_((lr * g).to(w.dtype))

    def _update_row_sparse(self, index, weight, grad, state):
        """Lazy row-sparse SGD (reference sgd_mom_update on
        kRowSparseStorage, optimizer_op.cc SGDMomLazyUpdate): only rows
        present in the gradient touch weights / momentum — O(nnz·D)
        instead of O(V·D) for embedding tables."""
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        rows, vals = grad.indices, grad.data
        with torch.no_grad():
            g = self._preprocess_grad(vals)
            wr = w.index_select(0, rows).float()
            g = g + wd * wr
            if state is not None and self.momentum != 0:
                # lr folded into the momentum buffer (reference rule)
                if self.lazy_update:
                    mr = state.index_select(0, rows)
                    mr.mul_(self.momentum).sub_(lr * g)
                    state.index_copy_(0, rows, mr)
                    upd = mr
                else:
                    state.mul_(self.momentum)
                    state.index_add_(0, rows, -lr * g)
                    upd = state.index_select(0, rows)
                w.index_copy_(0, rows, (wr + upd).to(w.dtype))
            else:
                w.index_copy_(0, rows, (wr - lr * g).to(w.dtype))

    def update_multi_precision(self, index, weight, grad, state):
        from ..ndarray.sparse import RowSparseNDArray
        if isinstance(grad, RowSparseNDArray):
            return self._update_row_sparse(index, weight, grad, state)
        return self._ump_dense(index, weight, grad, state)

    def _ump_dense(self, index, weight, grad, state):
        """Single fused HIP kernel on GPU (reference mp_sgd_mom_update):
        rescale+clip+wd+momentum+master-weight update+fp16 cast in one pass."""
        if isinstance(weight, NDArray) and weight.is_native:
            from .. import _core
            self._update_count(index)
            lr, wd = self._get_lr(index), self._get_wd(index)
            master = mom = None
            if isinstance(state, tuple):
                master, mom = state
            else:
                mom = state
            outs = [weight._h]
            if master is not None:
                outs.append(master._h)
            if mom is not None:
                outs.append(mom._h)
            _core.invoke_into(
                'sgd_update', [grad._h], outs,
                {'lr': str(lr), 'momentum': str(self.momentum),
                 'wd': str(wd), 'rescale_grad': str(self.rescale_grad),
                 'clip_gradient': str(self.clip_gradient or 0.0),
                 'has_master': '1' if master is not None else '0'})
            return
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        if type(self) is SGD and w.is_cuda and g.dtype == w.dtype:
            from ..ops.dispatch import hipops, use_hip
            ext = hipops() if use_hip(w) else None
            if ext is not None:
                self._update_count(index)
                lr, wd = self._get_lr(index), self._get_wd(index)
                if self.multi_precision and isinstance(state, tuple) and \
                        isinstance(state[0], torch.Tensor) and \
                        state[0].dtype is torch.float32 and \
                        w.dtype in (torch.float16, torch.bfloat16):
                    master, mom = state
                else:
                    master, mom = None, state
                ext.sgd_update(w, master, g.contiguous(), mom, lr,
                               self.momentum, wd, self.rescale_grad,
                               self.clip_gradient or 0.0)
                return
        super().update_multi_precision(index, weight, grad, state)


@register
class NAG(SGD):
    """Nesterov accelerated SGD (reference nag_mom_update)."""

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            if state is not None:
                state.mul_(self.momentum).add_(g)
                upd = g + self.momentum * state
            else:
                upd = g
            w.sub_((lr * upd).to(w.dtype))


@register
class Adam(Optimizer):
    """Adam (reference adam_update)."""

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon

    def create_state(self, index, weight):
        if isinstance(weight, NDArray) and weight.is_native:
            from ..ndarray.ndarray import zeros_like as _zl
            f32 = weight if str(weight.dtype) == 'float32' \
                else weight.astype('float32')
            return (_zl(f32), _zl(f32))
        w = weight._t if isinstance(weight, NDArray) else weight
        return (torch.zeros_like(w, dtype=torch.float32),
                torch.zeros_like(w, dtype=torch.float32))

    _adamw = False

    def _native_update(self, index, weight, grad, state):
        """Fused adam_update through the native registry (out layout:
        [w, m, v, master?]; kernel src/ops/nn_reg.hip adam_update)."""
        from .. import _core
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        lr_t = lr * math.sqrt(1 - self.beta2 ** t) / (1 - self.beta1 ** t)
        if isinstance(state, tuple) and len(state) == 2 and \
                isinstance(state[1], tuple):
            master, (m, v) = state
        else:
            master, (m, v) = None, state
        outs = [weight._h, m._h, v._h]
        if master is not None:
            outs.append(master._h)
        _core.invoke_into(
            'adam_update', [grad._h], outs,
            {'lr_t': str(lr_t), 'beta1': str(self.beta1),
             'beta2': str(self.beta2), 'eps': str(self.epsilon),
             'wd': str(wd), 'rescale_grad': str(self.rescale_grad),
             'clip_gradient': str(self.clip_gradient or 0.0),
             'adamw': '1' if self._adamw else '0'})

    def update(self, index, weight, grad, state):
        if isinstance(weight, NDArray) and weight.is_native:
            return self._native_update(index, weight, grad, state)
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        lr_t = lr * math.sqrt(1 - self.beta2 ** t) / (1 - self.beta1 ** t)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        m, v = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            w.sub_((lr_t * m / (v.sqrt() + self.epsilon)).to(w.dtype))

    def update_multi_precision(self, index, weight, grad, state):
        """Fused single-kernel Adam on GPU (reference adam_update /
        mp_adam_update): rescale+clip+wd+moment updates+bias-corrected
        step+master-weight cast in ONE launch — the eager torch
        composition was ~10 elementwise launches per parameter and
        DOMINATED the BERT step (profiles/r01_summary.md)."""
        if isinstance(weight, NDArray) and weight.is_native:
            return self._native_update(index, weight, grad, state)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        if w.is_cuda and g.dtype == w.dtype:
            from ..ops.dispatch import hipops, use_hip
            ext = hipops() if use_hip(w) else None
            if ext is not None and g.is_contiguous() and w.is_contiguous():
                self._update_count(index)
                lr, wd = self._get_lr(index), self._get_wd(index)
                t = self._index_update_count[index]
                lr_t = lr * math.sqrt(1 - self.beta2 ** t) / \
                    (1 - self.beta1 ** t)
                if self.multi_precision and isinstance(state, tuple) and \
                        isinstance(state[0], torch.Tensor) and \
                        state[0].dtype is torch.float32 and \
                        w.dtype in (torch.float16, torch.bfloat16):
                    master, (m, v) = state[0], state[1]
                else:
                    master, (m, v) = None, state
                ext.adam_update(w, master, g, m, v, lr_t, self.beta1,
                                self.beta2, self.epsilon, wd,
                                self.rescale_grad,
                                self.clip_gradient or 0.0, self._adamw)
                return
        return super().update_multi_precision(index, weight, grad, state)


@register
class AdamW(Adam):
    _adamw = True
    """Decoupled weight decay (reference contrib/adamw.cc)."""

    def update(self, index, weight, grad, state):
        if isinstance(weight, NDArray) and weight.is_native:
            return self._native_update(index, weight, grad, state)
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        lr_t = lr * math.sqrt(1 - self.beta2 ** t) / (1 - self.beta1 ** t)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        m, v = state
        with torch.no_grad():
            g = self._preprocess_grad(g)
            m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            w.sub_((lr_t * (m / (v.sqrt() + self.epsilon)) + lr * wd * w.float()).to(w.dtype))


@register
class RMSProp(Optimizer):
    def __init__(self, learning_rate=0.001, rho=0.9, momentum=0.9,
                 epsilon=1e-8, centered=False, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.rho, self.momentum, self.epsilon = rho, momentum, epsilon
        self.centered = centered

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        n = torch.zeros_like(w, dtype=torch.float32)
        if self.centered:
            return (n, torch.zeros_like(n), torch.zeros_like(n))
        return (n,)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            n = state[0]
            n.mul_(self.rho).addcmul_(g, g, value=1 - self.rho)
            if self.centered:
                _, mg, mom = state
                mg.mul_(self.rho).add_(g, alpha=1 - self.rho)
                mom.mul_(self.momentum).add_(lr * g / ((n - mg * mg + self.epsilon).sqrt()))
                w.sub_(mom.to(w.dtype))
            else:
                # reference rmsprop_update: epsilon INSIDE the sqrt
                w.sub_((lr * g / (n + self.epsilon).sqrt()).to(w.dtype))


@register
class AdaGrad(Optimizer):
    def __init__(self, learning_rate=0.01, epsilon=1e-7, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.epsilon = epsilon

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return torch.zeros_like(w, dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            state.addcmul_(g, g, value=1.0)
            w.sub_((lr * g / (state.sqrt() + self.epsilon)).to(w.dtype))


@register
class AdaDelta(Optimizer):
    def __init__(self, rho=0.90, epsilon=1e-5, **kwargs):
        super().__init__(**kwargs)
        self.rho, self.epsilon = rho, epsilon

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return (torch.zeros_like(w, dtype=torch.float32),
                torch.zeros_like(w, dtype=torch.float32))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        wd = self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        acc_g, acc_d = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            acc_g.mul_(self.rho).addcmul_(g, g, value=1 - self.rho)
            d = ((acc_d + self.epsilon).sqrt() / (acc_g + self.epsilon).sqrt()) * g
            acc_d.mul_(self.rho).addcmul_(d, d, value=1 - self.rho)
            w.sub_(d.to(w.dtype))


@register
class Signum(Optimizer):
    """signSGD with momentum (reference signum_update)."""

    def __init__(self, learning_rate=0.01, momentum=0.9, wd_lh=0.0, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.momentum = momentum
        self.wd_lh = wd_lh

    def create_state(self, index, weight):
        if self.momentum == 0:
            return None
        if isinstance(weight, NDArray) and weight.is_native:
            f32 = weight if str(weight.dtype) == 'float32' \
                else weight.astype('float32')
            from ..ndarray.ndarray import zeros_like as _zl
            return _zl(f32)
        w = weight._t if isinstance(weight, NDArray) else weight
        return torch.zeros_like(w, dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            if state is not None:
                state.mul_(self.momentum).add_(g, alpha=-(1 - self.momentum))
                w.add_((lr * torch.sign(state)).to(w.dtype))
            else:
                w.sub_((lr * torch.sign(g)).to(w.dtype))


@register
class LAMB(Optimizer):
    """Layer-wise adaptive moments (reference multi_lamb.cu / lamb.py)."""

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-6, lower_bound=None, upper_bound=None,
                 bias_correction=True, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon
        self.lower_bound, self.upper_bound = lower_bound, upper_bound
        self.bias_correction = bias_correction

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return (torch.zeros_like(w, dtype=torch.float32),
                torch.zeros_like(w, dtype=torch.float32))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        m, v = state
        with torch.no_grad():
            g = self._preprocess_grad(g)
            m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            mh, vh = m, v
            if self.bias_correction:
                mh = m / (1 - self.beta1 ** t)
                vh = v / (1 - self.beta2 ** t)
            upd = mh / (vh.sqrt() + self.epsilon) + wd * w.float()
            wnorm = w.float().norm()
            unorm = upd.norm()
            ratio = torch.where(
                (wnorm > 0) & (unorm > 0),
                wnorm / unorm, torch.ones_like(wnorm))
            if self.lower_bound:
                ratio = ratio.clamp(min=self.lower_bound)
            if self.upper_bound:
                ratio = ratio.clamp(max=self.upper_bound)
            w.sub_((lr * ratio * upd).to(w.dtype))


@register
class FTRL(Optimizer):
    def __init__(self, lamda1=0.01, learning_rate=0.1, beta=1.0, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.lamda1, self.beta = lamda1, beta

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return (torch.zeros_like(w, dtype=torch.float32),
                torch.zeros_like(w, dtype=torch.float32))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        z, n = state
        with torch.no_grad():
            g = self._preprocess_grad(g)
            sigma = ((n + g * g).sqrt() - n.sqrt()) / lr
            z.add_(g - sigma * w.float())
            n.add_(g * g)
            wnew = torch.where(
                z.abs() > self.lamda1,
                -(z - torch.sign(z) * self.lamda1) /
                ((self.beta + n.sqrt()) / lr + wd),
                torch.zeros_like(z))
            w.copy_(wnew.to(w.dtype))



@register
class FTML(Optimizer):
    """FTML (reference ftml_update, optimizer_op.cc)."""

    def __init__(self, beta1=0.6, beta2=0.999, epsilon=1e-8, **kwargs):
        super().__init__(**kwargs)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return tuple(torch.zeros_like(w, dtype=torch.float32)
                     for _ in range(3))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        d, v, z = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            v.mul_(self.beta2).add_(g * g, alpha=1 - self.beta2)
            coef = (1 - self.beta1 ** t)
            d_t = coef * ((v / (1 - self.beta2 ** t)).sqrt() + self.epsilon) / lr
            sigma = d_t - self.beta1 * d
            z.mul_(self.beta1).add_((1 - self.beta1) * g - sigma * w.float())
            d.copy_(d_t)
            w.copy_((-z / d_t).to(w.dtype))


@register
class Nadam(Adam):
    """Nesterov Adam (reference nadam in optimizer.py)."""

    def __init__(self, learning_rate=0.001, schedule_decay=0.004, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.schedule_decay = schedule_decay
        self.m_schedule = 1.0

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        m, v = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            mom_t = self.beta1 * (1 - 0.5 * 0.96 ** (t * self.schedule_decay))
            mom_t1 = self.beta1 * (1 - 0.5 * 0.96 ** ((t + 1) * self.schedule_decay))
            self.m_schedule *= mom_t
            ms1 = self.m_schedule * mom_t1
            gp = g / (1 - self.m_schedule)
            m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            mp = m / (1 - ms1)
            vp = v / (1 - self.beta2 ** t)
            mbar = (1 - mom_t) * gp + mom_t1 * mp
            w.sub_((lr * mbar / (vp.sqrt() + self.epsilon)).to(w.dtype))


@register
class AdaBelief(Adam):
    """AdaBelief (reference contrib adabelief)."""

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        t = self._index_update_count[index]
        lr_t = lr * math.sqrt(1 - self.beta2 ** t) / (1 - self.beta1 ** t)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        m, v = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            diff = g - m
            v.mul_(self.beta2).addcmul_(diff, diff, value=1 - self.beta2)
            w.sub_((lr_t * m / (v.sqrt() + self.epsilon)).to(w.dtype))


@register
class LARS(SGD):
    """Layer-wise adaptive rate scaling (reference multi_lars.cc)."""

    def __init__(self, eta=0.001, epsilon=1e-8, **kwargs):
        super().__init__(**kwargs)
        self.eta, self.epsilon = eta, epsilon

    def update(self, index, weight, grad, state):
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            wnorm = w.float().norm()
            gnorm = (g.float() * self.rescale_grad).norm()
            if wnorm > 0 and gnorm > 0:
                trust = self.eta * wnorm / (gnorm + self._get_wd(index) * wnorm
                                            + self.epsilon)
            else:
                trust = 1.0
        saved_lr = self.lr
        try:
            self.lr = float(trust) * self._get_lr(index)
            super().update(index, weight, grad, state)
        finally:
            self.lr = saved_lr


@register
class SGLD(Optimizer):
    """Stochastic gradient Langevin dynamics (reference sgld)."""

    def create_state(self, index, weight):
        return None

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            noise = torch.randn_like(w, dtype=torch.float32) * math.sqrt(lr)
            w.sub_((0.5 * lr * g + noise).to(w.dtype))


@register
class DCASGD(Optimizer):
    """Delay-compensated async SGD (reference dcasgd)."""

    def __init__(self, momentum=0.0, lamda=0.04, **kwargs):
        super().__init__(**kwargs)
        self.momentum, self.lamda = momentum, lamda

    def create_state(self, index, weight):
        w = weight._t if isinstance(weight, NDArray) else weight
        return (torch.zeros_like(w, dtype=torch.float32),
                w.detach().float().clone())

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        w = weight._t if isinstance(weight, NDArray) else weight
        g = grad._t if isinstance(grad, NDArray) else grad
        mom, prev = state
        with torch.no_grad():
            g = self._preprocess_grad(g) + wd * w.float()
            comp = g + self.lamda * g * g * (w.float() - prev)
            mom.mul_(self.momentum).sub_(lr * comp)
            prev.copy_(w.float())
            w.add_(mom.to(w.dtype))


# lowercase aliases matching mx.optimizer.create names
sgd = SGD
adam = Adam
adamw = AdamW
nag = NAG
rmsprop = RMSProp
adagrad = AdaGrad
adadelta = AdaDelta
lamb = LAMB
ftrl = FTRL
signum = Signum
ftml = FTML
nadam = Nadam
adabelief = AdaBelief
lars = LARS
sgld = SGLD
dcasgd = DCASGD


class Updater:
    """kvstore-side updater (reference optimizer/updater.py)."""

    def __init__(self, optimizer):
        self.optimizer = optimizer
        self.states = {}

    def __call__(self, index, grad, weight):
        if index not in self.states:
            self.states[index] = self.optimizer.create_state_multi_precision(index, weight)
        self.optimizer.update_multi_precision(index, weight, grad, self.states[index])

    def get_states(self):
        return self.states


def get_updater(optimizer):
    return Updater(optimizer)
