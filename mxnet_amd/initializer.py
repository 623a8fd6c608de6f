"""Weight initializers (reference python/mxnet/initializer.py)."""
import math
import re

import numpy as _np
import torch

_REGISTRY = {}


def register(cls):
    _REGISTRY[cls.__name__.lower()] = cls
    return cls


def create(init):
    if init is None:
        return Uniform()
    if isinstance(init, Initializer):
        return init
    if isinstance(init, str):
        name = init.lower()
        aliases = {'zeros': 'zero', 'ones': 'one', 'gaussian': 'normal',
                   'xavier': 'xavier', 'msraprelu': 'msraprelu'}
        name = aliases.get(name, name)
        if name in _REGISTRY:
            return _REGISTRY[name]()
        raise ValueError(f'unknown initializer {init}')
    raise TypeError(type(init))


class Initializer:
    """Base: dispatch by parameter name like the reference (initializer.py:97)."""

    def __call__(self, name, arr):
        if name.endswith('gamma'):
            self._init_gamma(name, arr)
        elif name.endswith('beta'):
            self._init_beta(name, arr)
        elif name.endswith('running_mean') or name.endswith('moving_mean'):
            self._init_zero(name, arr)
        elif name.endswith('running_var') or name.endswith('moving_var'):
            self._init_one(name, arr)
        elif name.endswith('bias'):
            self._init_bias(name, arr)
        else:
            self._init_weight(name, arr)

    def init_weight(self, name, arr):
        self._init_weight(name, arr)

    def _init_gamma(self, name, arr):
        self._init_one(name, arr)

    def _init_beta(self, name, arr):
        self._init_zero(name, arr)

    def _init_bias(self, name, arr):
        self._init_zero(name, arr)

    def _init_zero(self, name, arr):
        with torch.no_grad():
            arr._t.zero_()

    def _init_one(self, name, arr):
        with torch.no_grad():
            arr._t.fill_(1.0)

    def _init_weight(self, name, arr):
        raise NotImplementedError

    def __repr__(self):
        return self.__class__.__name__


@register
class Zero(Initializer):
    def _init_weight(self, name, arr):
        self._init_zero(name, arr)


zeros = Zero


@register
class One(Initializer):
    def _init_weight(self, name, arr):
        self._init_one(name, arr)


ones = One


@register
class Constant(Initializer):
    def __init__(self, value=0.0):
        self.value = value

    def _init_weight(self, name, arr):
        with torch.no_grad():
            if hasattr(self.value, 'is_native') and self.value.is_native:
                # native source (e.g. loaded checkpoint arrays in native
                # mode): bridge through numpy into the init buffer
                src = torch.from_numpy(self.value.asnumpy())
                arr._t.copy_(src.to(arr._t.device, arr._t.dtype))
            elif hasattr(self.value, '_t'):
                arr._t.copy_(self.value._t.to(arr._t.device, arr._t.dtype))
            else:
                arr._t.fill_(float(self.value))


@register
class Uniform(Initializer):
    def __init__(self, scale=0.07):
        self.scale = scale

    def _init_weight(self, name, arr):
        with torch.no_grad():
            arr._t.uniform_(-self.scale, self.scale)


@register
class Normal(Initializer):
    def __init__(self, sigma=0.01):
        self.sigma = sigma

    def _init_weight(self, name, arr):
        with torch.no_grad():
            arr._t.normal_(0, self.sigma)


@register
class Orthogonal(Initializer):
    def __init__(self, scale=1.414, rand_type='uniform'):
        self.scale = scale

    def _init_weight(self, name, arr):
        with torch.no_grad():
            torch.nn.init.orthogonal_(arr._t, gain=self.scale)


@register
class Xavier(Initializer):
    """Xavier/Glorot (reference initializer.py Xavier)."""

    def __init__(self, rnd_type='uniform', factor_type='avg', magnitude=3):
        self.rnd_type = rnd_type
        self.factor_type = factor_type
        self.magnitude = float(magnitude)

    def _init_weight(self, name, arr):
        shape = arr.shape
        hw_scale = 1.0
        if len(shape) < 2:
            raise ValueError(f'Xavier requires >=2D weight, got {shape} for {name}')
        if len(shape) > 2:
            hw_scale = _np.prod(shape[2:])
        fan_in, fan_out = shape[1] * hw_scale, shape[0] * hw_scale
        factor = {'avg': (fan_in + fan_out) / 2.0,
                  'in': fan_in, 'out': fan_out}[self.factor_type]
        scale = math.sqrt(self.magnitude / factor)
        with torch.no_grad():
            if self.rnd_type == 'uniform':
                arr._t.uniform_(-scale, scale)
            else:
                arr._t.normal_(0, scale)


@register
class MSRAPrelu(Xavier):
    def __init__(self, factor_type='avg', slope=0.25):
        magnitude = 2.0 / (1 + slope ** 2)
        super().__init__('gaussian', factor_type, magnitude)


@register
class LSTMBias(Initializer):
    """Forget-gate bias = 1 (reference initializer.py LSTMBias)."""

    def __init__(self, forget_bias=1.0):
        self.forget_bias = forget_bias

    def _init_weight(self, name, arr):
        with torch.no_grad():
            arr._t.zero_()
            n = arr.shape[0] // 4
            arr._t[n:2 * n] = self.forget_bias

    def _init_bias(self, name, arr):
        self._init_weight(name, arr)
