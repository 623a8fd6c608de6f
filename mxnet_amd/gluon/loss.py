"""Gluon losses (reference python/mxnet/gluon/loss.py)."""
import torch

from .block import HybridBlock
from ..ndarray.ndarray import NDArray


def _t(x):
    return x._t if isinstance(x, NDArray) else x


def _apply_weighting(loss, weight=None, sample_weight=None):
    if sample_weight is not None:
        loss = loss * _t(sample_weight)
    if weight is not None:
        loss = loss * weight
    return loss


def _reshape_like(pred, label):
    return label.reshape(pred.shape)


# -- native-runtime composition helpers --------------------------------------
def _n_mean_nonbatch(loss, batch_axis):
    dims = tuple(d for d in range(loss.ndim) if d != batch_axis)
    return loss.mean(axis=dims) if dims else loss


def _n_relu(x):
    from ..ndarray import ops as _F
    return _F.Activation(x, act_type='relu')


def _n_softplus(x):
    # log(1+exp(x)) = relu(x) + log1p(exp(-|x|)), overflow-safe
    return _n_relu(x) + ((x.abs() * -1.0).exp() + 1.0).log()


def _n_weight(loss, weight, sample_weight):
    if sample_weight is not None:
        loss = loss * sample_weight
    if weight is not None:
        loss = loss * float(weight)
    return loss


class Loss(HybridBlock):
    def __init__(self, weight=None, batch_axis=0, **kwargs):
        super().__init__(**kwargs)
        self._weight = weight
        self._batch_axis = batch_axis

    def _mean_nonbatch(self, loss):
        dims = [d for d in range(loss.dim()) if d != self._batch_axis]
        return loss.mean(dim=dims) if dims else loss

    def forward(self, *args, **kwargs):
        if args and isinstance(args[0], NDArray) and args[0].is_native:
            fn = getattr(self, '_forward_native', None)
            if fn is None:
                raise RuntimeError(
                    f'{type(self).__name__} has no native-runtime path yet')
            return fn(*args, **kwargs)
        nds = [a._t if isinstance(a, NDArray) else a for a in args]
        return NDArray(self._forward(*nds, **kwargs))


class L2Loss(Loss):
    def __init__(self, weight=1.0, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)

    def _forward(self, pred, label, sample_weight=None):
        loss = (pred - _reshape_like(pred, label)) ** 2
        loss = _apply_weighting(loss, self._weight / 2, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        loss = (pred - label.reshape(pred.shape)).square()
        loss = _n_weight(loss, self._weight / 2 if self._weight else None,
                         sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


class L1Loss(Loss):
    def __init__(self, weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)

    def _forward_native(self, pred, label, sample_weight=None):
        loss = (pred - label.reshape(pred.shape)).abs()
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)

    def _forward(self, pred, label, sample_weight=None):
        loss = (pred - _reshape_like(pred, label)).abs()
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)


class SoftmaxCrossEntropyLoss(Loss):
    """Softmax+CE (reference loss.py SoftmaxCrossEntropyLoss); the GPU
    log-softmax runs the native fused kernel via ops.nn."""

    def __init__(self, axis=-1, sparse_label=True, from_logits=False,
                 weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._axis = axis
        self._sparse_label = sparse_label
        self._from_logits = from_logits

    def _forward(self, pred, label, sample_weight=None):
        from ..ops import nn as _nn
        if not self._from_logits:
            pred = _nn.log_softmax(pred, self._axis)
        if self._sparse_label:
            loss = -pred.gather(self._axis,
                                label.long().unsqueeze(self._axis)).squeeze(self._axis)
        else:
            loss = -(pred * label).sum(self._axis)
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        from ..ndarray import ops as F
        if not self._from_logits:
            pred = F.log_softmax(pred, self._axis)
        if self._sparse_label:
            loss = -F.pick(pred, label)
        else:
            loss = -(pred * label).sum(axis=self._axis)
        if self._weight is not None and self._weight != 1.0:
            loss = loss * self._weight
        dims = tuple(d for d in range(loss.ndim) if d != self._batch_axis)
        return loss.mean(axis=dims) if dims else loss


SoftmaxCELoss = SoftmaxCrossEntropyLoss


class SigmoidBinaryCrossEntropyLoss(Loss):
    def __init__(self, from_sigmoid=False, weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._from_sigmoid = from_sigmoid

    def _forward(self, pred, label, sample_weight=None, pos_weight=None):
        label = _reshape_like(pred, label)
        if not self._from_sigmoid:
            loss = torch.nn.functional.binary_cross_entropy_with_logits(
                pred, label, reduction='none', pos_weight=pos_weight)
        else:
            eps = 1e-12
            loss = -(label * torch.log(pred + eps)
                     + (1 - label) * torch.log(1 - pred + eps))
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None,
                        pos_weight=None):
        assert pos_weight is None, \
            'pos_weight: use the torch frontend for this option'
        label = label.reshape(pred.shape)
        if not self._from_sigmoid:
            # softplus(pred) - label*pred (standard logits BCE)
            loss = _n_softplus(pred) - label * pred
        else:
            eps = 1e-12
            loss = ((pred + eps).log() * label
                    + ((pred * -1.0) + (1 + eps)).log() * (label * -1.0 + 1.0)) * -1.0
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


SigmoidBCELoss = SigmoidBinaryCrossEntropyLoss


class KLDivLoss(Loss):
    def __init__(self, from_logits=True, axis=-1, weight=None, batch_axis=0,
                 **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._from_logits = from_logits
        self._axis = axis

    def _forward(self, pred, label, sample_weight=None):
        from ..ops import nn as _nn
        if not self._from_logits:
            pred = _nn.log_softmax(pred, self._axis)
        loss = label * (torch.log(label + 1e-12) - pred)
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        from ..ndarray import ops as _F
        if not self._from_logits:
            pred = _F.log_softmax(pred, axis=self._axis)
        loss = label * ((label + 1e-12).log() - pred)
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


class HuberLoss(Loss):
    def __init__(self, rho=1.0, weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._rho = rho

    def _forward(self, pred, label, sample_weight=None):
        label = _reshape_like(pred, label)
        loss = (pred - label).abs()
        loss = torch.where(loss > self._rho,
                           loss - 0.5 * self._rho,
                           (0.5 / self._rho) * loss ** 2)
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        # piecewise huber as clip+relu composition:
        # (0.5/rho)*min(|d|,rho)^2 + relu(|d|-rho)
        d = (pred - label.reshape(pred.shape)).abs()
        rho = float(self._rho)
        loss = d.clip(0.0, rho).square() * (0.5 / rho) + _n_relu(d - rho)
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


class HingeLoss(Loss):
    def __init__(self, margin=1, weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._margin = margin

    def _forward(self, pred, label, sample_weight=None):
        label = _reshape_like(pred, label)
        loss = torch.relu(self._margin - pred * label)
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        z = pred * label.reshape(pred.shape) * -1.0 + float(self._margin)
        loss = _n_relu(z)
        if type(self).__name__ == 'SquaredHingeLoss':
            loss = loss.square()
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


class SquaredHingeLoss(HingeLoss):
    def _forward(self, pred, label, sample_weight=None):
        label = _reshape_like(pred, label)
        loss = torch.relu(self._margin - pred * label) ** 2
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)


class LogisticLoss(Loss):
    def __init__(self, weight=None, batch_axis=0, label_format='signed', **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._label_format = label_format

    def _forward(self, pred, label, sample_weight=None):
        label = _reshape_like(pred, label)
        if self._label_format == 'binary':
            label = 2 * label - 1
        loss = torch.nn.functional.softplus(-pred * label)
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return self._mean_nonbatch(loss)

    def _forward_native(self, pred, label, sample_weight=None):
        label = label.reshape(pred.shape)
        if self._label_format == 'binary':
            label = label * 2.0 - 1.0
        loss = _n_softplus(pred * label * -1.0)
        loss = _n_weight(loss, self._weight, sample_weight)
        return _n_mean_nonbatch(loss, self._batch_axis)


class TripletLoss(Loss):
    def __init__(self, margin=1, weight=None, batch_axis=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._margin = margin

    def _forward(self, pred, positive, negative, sample_weight=None):
        dims = list(range(1, pred.dim()))
        loss = ((pred - positive) ** 2 - (pred - negative) ** 2)
        loss = loss.sum(dim=dims) if dims else loss
        loss = torch.relu(loss + self._margin)
        return _apply_weighting(loss, self._weight, sample_weight)


class PoissonNLLLoss(Loss):
    def __init__(self, weight=None, from_logits=True, batch_axis=0,
                 compute_full=False, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._from_logits = from_logits
        self._compute_full = compute_full

    def _forward(self, pred, target, sample_weight=None, epsilon=1e-08):
        target = _reshape_like(pred, target)
        if self._from_logits:
            loss = torch.exp(pred) - target * pred
        else:
            loss = pred - target * torch.log(pred + epsilon)
        if self._compute_full:
            stirling = (target * torch.log(target + epsilon) - target
                        + 0.5 * torch.log(2 * torch.pi * (target + epsilon)))
            stirling = torch.where(target <= 1, torch.zeros_like(stirling), stirling)
            loss = loss + stirling
        loss = _apply_weighting(loss, self._weight, sample_weight)
        return loss.mean()


class CosineEmbeddingLoss(Loss):
    def __init__(self, weight=None, batch_axis=0, margin=0, **kwargs):
        super().__init__(weight, batch_axis, **kwargs)
        self._margin = margin

    def _forward(self, input1, input2, label, sample_weight=None):
        cos = torch.nn.functional.cosine_similarity(
            input1.reshape(input1.shape[0], -1),
            input2.reshape(input2.shape[0], -1), dim=1)
        label = label.reshape(-1)
        loss = torch.where(label == 1, 1 - cos,
                           torch.relu(cos - self._margin))
        return _apply_weighting(loss, self._weight, sample_weight)


class CTCLoss(Loss):
    """CTC (reference warp-ctc based ctc_loss.cu; torch CPU/GPU ctc here —
    hand-HIP CTC is roadmap)."""

    def __init__(self, layout='NTC', label_layout='NT', weight=None, **kwargs):
        super().__init__(weight, 0, **kwargs)
        self._layout = layout
        self._label_layout = label_layout

    def _forward(self, pred, label, pred_lengths=None, label_lengths=None,
                 sample_weight=None):
        if self._layout == 'NTC':
            pred = pred.transpose(0, 1)  # -> TNC
        T, N, C = pred.shape
        logp = torch.nn.functional.log_softmax(pred.float(), dim=-1)
        if pred_lengths is None:
            pred_lengths = torch.full((N,), T, dtype=torch.long, device=pred.device)
        else:
            pred_lengths = pred_lengths.long()
        if label_lengths is None:
            label_lengths = (label >= 0).sum(dim=-1).long()
        loss = torch.nn.functional.ctc_loss(
            logp, label.long(), pred_lengths, label_lengths,
            blank=0, reduction='none', zero_infinity=True)
        return _apply_weighting(loss, self._weight, sample_weight)
