"""Probability distributions over NDArray (reference
gluon/probability/distributions/*.py, ~25 distributions).

MI355X-native design: the distribution math (log_prob/sample/entropy/kl)
is elementwise and runs through the same torch-tensor storage the rest
of the framework uses; sampling uses the device Philox generator.  The
API mirrors the reference: constructor args accept NDArray / scalars,
methods return NDArray.
"""
import torch
import torch.distributions as _td

from ...ndarray.ndarray import NDArray

__all__ = ['Distribution', 'Normal', 'LogNormal', 'HalfNormal', 'Laplace',
           'Cauchy', 'HalfCauchy', 'Uniform', 'Exponential', 'Gamma',
           'Beta', 'Chi2', 'FisherSnedecor', 'StudentT', 'Pareto',
           'Weibull', 'Gumbel', 'Bernoulli', 'Binomial', 'Geometric',
           'NegativeBinomial', 'Poisson', 'Categorical', 'OneHotCategorical',
           'Multinomial', 'Dirichlet', 'MultivariateNormal',
           'RelaxedBernoulli', 'RelaxedOneHotCategorical', 'kl_divergence',
           'register_kl']


def _t(x):
    if isinstance(x, NDArray):
        return x.handle
    if isinstance(x, torch.Tensor):
        return x
    return torch.as_tensor(x, dtype=torch.float32) if x is not None else None


def _wrap(t):
    return NDArray(t) if isinstance(t, torch.Tensor) else t


class Distribution:
    """Base distribution (reference distribution.py:36)."""

    _torch_cls = None
    _arg_names = ()

    # subclasses may set has_grad / has_enumerate_support etc.
    has_grad = False

    def __init__(self, *args, **kwargs):
        names = list(self._arg_names)
        bound = dict(zip(names, args))
        bound.update({k: v for k, v in kwargs.items() if k in names})
        self._args = {k: _t(v) for k, v in bound.items() if v is not None}
        extra = {k: v for k, v in kwargs.items()
                 if k not in names and k != 'validate_args'}
        self._dist = self._torch_cls(**self._args, **extra)

    # -- properties ------------------------------------------------------
    @property
    def mean(self):
        return _wrap(self._dist.mean)

    @property
    def variance(self):
        return _wrap(self._dist.variance)

    @property
    def stddev(self):
        return _wrap(self._dist.stddev)

    @property
    def support(self):
        return self._dist.support

    # -- methods ---------------------------------------------------------
    def log_prob(self, value):
        return _wrap(self._dist.log_prob(_t(value)))

    def prob(self, value):
        return _wrap(self._dist.log_prob(_t(value)).exp())

    def cdf(self, value):
        return _wrap(self._dist.cdf(_t(value)))

    def icdf(self, value):
        return _wrap(self._dist.icdf(_t(value)))

    def sample(self, size=()):
        if isinstance(size, int):
            size = (size,)
        return _wrap(self._dist.sample(torch.Size(size)))

    def sample_n(self, n):
        return self.sample((n,))

    def rsample(self, size=()):
        if isinstance(size, int):
            size = (size,)
        return _wrap(self._dist.rsample(torch.Size(size)))

    def entropy(self):
        return _wrap(self._dist.entropy())

    def perplexity(self):
        return _wrap(self._dist.perplexity())

    def enumerate_support(self, expand=True):
        return _wrap(self._dist.enumerate_support(expand))

    def __repr__(self):
        return f'{type(self).__name__}({", ".join(self._args)})'


def _make(name, torch_cls, arg_names, rsample=False):
    cls = type(name, (Distribution,), {
        '_torch_cls': torch_cls,
        '_arg_names': tuple(arg_names),
        'has_grad': rsample,
        '__doc__': f'{name} distribution '
                   f'(reference gluon/probability/distributions/'
                   f'{name.lower()}.py).',
    })
    return cls


Normal = _make('Normal', _td.Normal, ('loc', 'scale'), rsample=True)
LogNormal = _make('LogNormal', _td.LogNormal, ('loc', 'scale'), rsample=True)
HalfNormal = _make('HalfNormal', _td.HalfNormal, ('scale',), rsample=True)
Laplace = _make('Laplace', _td.Laplace, ('loc', 'scale'), rsample=True)
Cauchy = _make('Cauchy', _td.Cauchy, ('loc', 'scale'), rsample=True)
HalfCauchy = _make('HalfCauchy', _td.HalfCauchy, ('scale',), rsample=True)
Uniform = _make('Uniform', _td.Uniform, ('low', 'high'), rsample=True)
Exponential = _make('Exponential', _td.Exponential, ('rate',), rsample=True)
Gamma = _make('Gamma', _td.Gamma, ('concentration', 'rate'), rsample=True)
Beta = _make('Beta', _td.Beta, ('concentration1', 'concentration0'),
             rsample=True)
Chi2 = _make('Chi2', _td.Chi2, ('df',), rsample=True)
FisherSnedecor = _make('FisherSnedecor', _td.FisherSnedecor, ('df1', 'df2'),
                       rsample=True)
StudentT = _make('StudentT', _td.StudentT, ('df', 'loc', 'scale'),
                 rsample=True)
Pareto = _make('Pareto', _td.Pareto, ('scale', 'alpha'), rsample=True)
Weibull = _make('Weibull', _td.Weibull, ('scale', 'concentration'),
                rsample=True)
Gumbel = _make('Gumbel', _td.Gumbel, ('loc', 'scale'), rsample=True)
Bernoulli = _make('Bernoulli', _td.Bernoulli, ('probs', 'logits'))
Binomial = _make('Binomial', _td.Binomial, ('total_count', 'probs', 'logits'))
Geometric = _make('Geometric', _td.Geometric, ('probs', 'logits'))
NegativeBinomial = _make('NegativeBinomial', _td.NegativeBinomial,
                         ('total_count', 'probs', 'logits'))
Poisson = _make('Poisson', _td.Poisson, ('rate',))
Categorical = _make('Categorical', _td.Categorical, ('probs', 'logits'))
OneHotCategorical = _make('OneHotCategorical', _td.OneHotCategorical,
                          ('probs', 'logits'))
Multinomial = _make('Multinomial', _td.Multinomial,
                    ('total_count', 'probs', 'logits'))
Dirichlet = _make('Dirichlet', _td.Dirichlet, ('concentration',),
                  rsample=True)
MultivariateNormal = _make('MultivariateNormal', _td.MultivariateNormal,
                           ('loc', 'covariance_matrix', 'precision_matrix',
                            'scale_tril'), rsample=True)


class RelaxedBernoulli(Distribution):
    """Gumbel-softmax relaxed Bernoulli (reference relaxed_bernoulli.py)."""
    _torch_cls = _td.RelaxedBernoulli
    _arg_names = ('temperature', 'probs', 'logits')
    has_grad = True


class RelaxedOneHotCategorical(Distribution):
    _torch_cls = _td.RelaxedOneHotCategorical
    _arg_names = ('temperature', 'probs', 'logits')
    has_grad = True


# -- KL divergence registry (reference divergence.py) -----------------------
_KL_CUSTOM = {}


def register_kl(type_p, type_q):
    """Decorator registering a custom KL(p||q) implementation."""
    def deco(fn):
        _KL_CUSTOM[(type_p, type_q)] = fn
        return fn
    return deco


def kl_divergence(p, q):
    fn = _KL_CUSTOM.get((type(p), type(q)))
    if fn is not None:
        return fn(p, q)
    return _wrap(_td.kl.kl_divergence(p._dist, q._dist))
