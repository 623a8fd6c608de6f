"""Distribution transformations (reference gluon/probability/transformation):
bijectors composing with TransformedDistribution.
"""
import torch.distributions as _td
import torch.distributions.transforms as _tt

from .distributions import Distribution, _t, _wrap

__all__ = ['Transformation', 'TransformedDistribution', 'AffineTransform',
           'ExpTransform', 'PowerTransform', 'SigmoidTransform',
           'SoftmaxTransform', 'AbsTransform', 'ComposeTransform']


class Transformation:
    _torch_t = None

    def __init__(self, *args, **kwargs):
        args = tuple(_t(a) for a in args)
        kwargs = {k: _t(v) for k, v in kwargs.items()}
        self._tf = self._torch_t(*args, **kwargs)

    def __call__(self, x):
        return _wrap(self._tf(_t(x)))

    def inv(self, y):
        return _wrap(self._tf.inv(_t(y)))

    def log_det_jacobian(self, x, y):
        return _wrap(self._tf.log_abs_det_jacobian(_t(x), _t(y)))


def _mk(name, tcls):
    return type(name, (Transformation,), {'_torch_t': tcls})


AffineTransform = _mk('AffineTransform', _tt.AffineTransform)
ExpTransform = _mk('ExpTransform', _tt.ExpTransform)
PowerTransform = _mk('PowerTransform', _tt.PowerTransform)
SigmoidTransform = _mk('SigmoidTransform', _tt.SigmoidTransform)
SoftmaxTransform = _mk('SoftmaxTransform', _tt.SoftmaxTransform)
AbsTransform = _mk('AbsTransform', _tt.AbsTransform)


class ComposeTransform(Transformation):
    def __init__(self, parts):
        self._tf = _tt.ComposeTransform([p._tf for p in parts])


class TransformedDistribution(Distribution):
    """base distribution + chain of transformations."""

    def __init__(self, base, transforms):
        if not isinstance(transforms, (list, tuple)):
            transforms = [transforms]
        self._args = {}
        self._dist = _td.TransformedDistribution(
            base._dist, [t._tf for t in transforms])
