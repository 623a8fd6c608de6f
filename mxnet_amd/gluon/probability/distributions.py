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


HalfNormal = _make('HalfNormal', _td.HalfNormal, ('scale',), rsample=True)
Cauchy = _make('Cauchy', _td.Cauchy, ('loc', 'scale'), rsample=True)
HalfCauchy = _make('HalfCauchy', _td.HalfCauchy, ('scale',), rsample=True)
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
Binomial = _make('Binomial', _td.Binomial, ('total_count', 'probs', 'logits'))
NegativeBinomial = _make('NegativeBinomial', _td.NegativeBinomial,
                         ('total_count', 'probs', 'logits'))
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


# ---------------------------------------------------------------------------
# own-math distributions: the formulas below are written out over
# elementwise tensor ops (the reference writes the same formulas over
# mx.nd ops — gluon/probability/distributions/*.py); only the tail of
# exotic distributions still delegates to torch.distributions.
# ---------------------------------------------------------------------------
import math as _math

_LOG_SQRT_2PI = 0.5 * _math.log(2.0 * _math.pi)


class _OwnDistribution(Distribution):
    """Base for distributions with hand-written math; keeps the public
    Distribution API (log_prob/sample/mean/variance/entropy/cdf...)."""

    _torch_cls = None  # set per class: interop shim for transforms/kl tail

    @property
    def _dist(self):
        if self._torch_cls is None:
            raise NotImplementedError(
                f'{type(self).__name__}: method not implemented')
        # lazily build the torch counterpart (TransformedDistribution /
        # unregistered-KL interop only; our own math handles the rest)
        return self._torch_cls(**self._args)

    def __init__(self, *args, **kwargs):  # no torch.distributions object
        names = list(self._arg_names)
        bound = dict(zip(names, args))
        bound.update({k: v for k, v in kwargs.items() if k in names})
        self._args = {k: _t(v) for k, v in bound.items() if v is not None}
        for k in names:
            setattr(self, k, self._args.get(k))

    def _shape(self, size):
        if isinstance(size, int):
            size = (size,)
        b = None
        for v in self._args.values():
            if isinstance(v, torch.Tensor):
                b = v.shape if b is None else torch.broadcast_shapes(b, v.shape)
        b = b or torch.Size(())
        return torch.Size(tuple(size)) + b

    def prob(self, value):
        return _wrap(_t(self.log_prob(value)).exp())

    def sample_n(self, n):
        return self.sample((n,))

    def __repr__(self):
        return f'{type(self).__name__}({", ".join(self._args)})'


class Normal(_OwnDistribution):
    _torch_cls = _td.Normal
    """Gaussian (reference distributions/normal.py): every method is the
    closed-form expression over elementwise ops."""
    _arg_names = ('loc', 'scale')
    has_grad = True

    @property
    def mean(self):
        return _wrap(self.loc + torch.zeros_like(self.scale))

    @property
    def stddev(self):
        return _wrap(self.scale + torch.zeros_like(self.loc))

    @property
    def variance(self):
        return _wrap(self.scale ** 2 + torch.zeros_like(self.loc))

    def log_prob(self, value):
        v = _t(value)
        z = (v - self.loc) / self.scale
        return _wrap(-0.5 * z * z - torch.log(self.scale) - _LOG_SQRT_2PI)

    def cdf(self, value):
        v = _t(value)
        return _wrap(0.5 * (1 + torch.erf(
            (v - self.loc) / (self.scale * _math.sqrt(2.0)))))

    def icdf(self, value):
        v = _t(value)
        return _wrap(self.loc + self.scale * _math.sqrt(2.0)
                     * torch.erfinv(2 * v - 1))

    def entropy(self):
        return _wrap(0.5 + _LOG_SQRT_2PI + torch.log(
            self.scale + torch.zeros_like(self.loc)))

    def rsample(self, size=()):
        eps = torch.randn(self._shape(size), dtype=self.scale.dtype
                          if isinstance(self.scale, torch.Tensor)
                          else torch.float32)
        return _wrap(self.loc + eps * self.scale)

    sample = rsample


class LogNormal(Normal):
    """exp of a Normal (reference lognormal.py)."""
    _torch_cls = _td.LogNormal

    @property
    def mean(self):
        return _wrap(torch.exp(self.loc + 0.5 * self.scale ** 2))

    @property
    def variance(self):
        s2 = self.scale ** 2
        return _wrap((torch.exp(s2) - 1) * torch.exp(2 * self.loc + s2))

    @property
    def stddev(self):
        return _wrap(torch.sqrt(_t(self.variance)))

    def log_prob(self, value):
        v = _t(value)
        lv = torch.log(v)
        z = (lv - self.loc) / self.scale
        return _wrap(-0.5 * z * z - torch.log(self.scale)
                     - _LOG_SQRT_2PI - lv)

    def cdf(self, value):
        return super().cdf(torch.log(_t(value)))

    def entropy(self):
        return _wrap(0.5 + _LOG_SQRT_2PI + torch.log(
            self.scale + torch.zeros_like(self.loc)) + self.loc
            + torch.zeros_like(self.scale))

    def rsample(self, size=()):
        return _wrap(torch.exp(_t(super().rsample(size))))

    sample = rsample


class Laplace(_OwnDistribution):
    _torch_cls = _td.Laplace
    _arg_names = ('loc', 'scale')
    has_grad = True

    @property
    def mean(self):
        return _wrap(self.loc + torch.zeros_like(self.scale))

    @property
    def variance(self):
        return _wrap(2 * self.scale ** 2 + torch.zeros_like(self.loc))

    @property
    def stddev(self):
        return _wrap(torch.sqrt(_t(self.variance)))

    def log_prob(self, value):
        v = _t(value)
        return _wrap(-torch.abs(v - self.loc) / self.scale
                     - torch.log(2 * self.scale))

    def cdf(self, value):
        v = _t(value)
        z = (v - self.loc) / self.scale
        return _wrap(0.5 - 0.5 * torch.sign(z) * torch.expm1(-z.abs()))

    def entropy(self):
        return _wrap(1 + torch.log(2 * self.scale)
                     + torch.zeros_like(self.loc))

    def rsample(self, size=()):
        u = torch.rand(self._shape(size)) - 0.5
        return _wrap(self.loc - self.scale * torch.sign(u)
                     * torch.log1p(-2 * u.abs()))

    sample = rsample


class Uniform(_OwnDistribution):
    _torch_cls = _td.Uniform
    _arg_names = ('low', 'high')
    has_grad = True

    @property
    def mean(self):
        return _wrap((self.low + self.high) / 2)

    @property
    def variance(self):
        return _wrap((self.high - self.low) ** 2 / 12)

    def log_prob(self, value):
        v = _t(value)
        inside = (v >= self.low) & (v < self.high)
        lp = -torch.log(self.high - self.low)
        return _wrap(torch.where(inside, lp + torch.zeros_like(v),
                                 torch.full_like(v, float('-inf'))))

    def cdf(self, value):
        v = _t(value)
        return _wrap(((v - self.low) / (self.high - self.low)).clamp(0, 1))

    def entropy(self):
        return _wrap(torch.log(self.high - self.low))

    def rsample(self, size=()):
        u = torch.rand(self._shape(size))
        return _wrap(self.low + u * (self.high - self.low))

    sample = rsample


class Exponential(_OwnDistribution):
    _torch_cls = _td.Exponential
    _arg_names = ('rate',)
    has_grad = True

    @property
    def mean(self):
        return _wrap(1.0 / self.rate)

    @property
    def variance(self):
        return _wrap(self.rate ** -2)

    def log_prob(self, value):
        v = _t(value)
        return _wrap(torch.log(self.rate) - self.rate * v)

    def cdf(self, value):
        return _wrap(-torch.expm1(-self.rate * _t(value)))

    def icdf(self, value):
        return _wrap(-torch.log1p(-_t(value)) / self.rate)

    def entropy(self):
        return _wrap(1 - torch.log(self.rate))

    def rsample(self, size=()):
        u = torch.rand(self._shape(size))
        return _wrap(-torch.log1p(-u) / self.rate)

    sample = rsample


class Gumbel(_OwnDistribution):
    _torch_cls = _td.Gumbel
    _arg_names = ('loc', 'scale')
    has_grad = True
    _EULER = 0.57721566490153286555

    @property
    def mean(self):
        return _wrap(self.loc + self.scale * self._EULER)

    @property
    def variance(self):
        return _wrap((_math.pi ** 2 / 6) * self.scale ** 2
                     + torch.zeros_like(self.loc))

    def log_prob(self, value):
        z = (_t(value) - self.loc) / self.scale
        return _wrap(-(z + torch.exp(-z)) - torch.log(self.scale))

    def cdf(self, value):
        z = (_t(value) - self.loc) / self.scale
        return _wrap(torch.exp(-torch.exp(-z)))

    def entropy(self):
        return _wrap(torch.log(self.scale) + 1 + self._EULER
                     + torch.zeros_like(self.loc))

    def rsample(self, size=()):
        u = torch.rand(self._shape(size)).clamp_min(1e-20)
        return _wrap(self.loc - self.scale * torch.log(-torch.log(u)))

    sample = rsample


def _probs_logits(self):
    if getattr(self, 'logits', None) is not None:
        lg = self.logits
        return torch.sigmoid(lg), lg
    pr = self.probs
    pr = pr.clamp(1e-7, 1 - 1e-7)
    return pr, torch.log(pr) - torch.log1p(-pr)


class Bernoulli(_OwnDistribution):
    _torch_cls = _td.Bernoulli
    _arg_names = ('probs', 'logits')

    @property
    def mean(self):
        return _wrap(_probs_logits(self)[0])

    @property
    def variance(self):
        p = _probs_logits(self)[0]
        return _wrap(p * (1 - p))

    def log_prob(self, value):
        v = _t(value)
        _, lg = _probs_logits(self)
        # -BCEWithLogits: v*log(p) + (1-v)*log(1-p), numerically via logits
        return _wrap(v * lg - torch.nn.functional.softplus(lg))

    def entropy(self):
        p, lg = _probs_logits(self)
        return _wrap(torch.nn.functional.softplus(lg) - p * lg)

    def sample(self, size=()):
        p = _probs_logits(self)[0]
        shape = self._shape(size)
        return _wrap((torch.rand(shape) < p).to(torch.float32))


class Geometric(_OwnDistribution):
    _torch_cls = _td.Geometric
    """P(X=k) = (1-p)^k p, k = 0,1,2,... (reference geometric.py)."""
    _arg_names = ('probs', 'logits')

    @property
    def mean(self):
        p = _probs_logits(self)[0]
        return _wrap((1 - p) / p)

    @property
    def variance(self):
        p = _probs_logits(self)[0]
        return _wrap((1 - p) / p ** 2)

    def log_prob(self, value):
        v = _t(value)
        p, _ = _probs_logits(self)
        return _wrap(v * torch.log1p(-p) + torch.log(p))

    def entropy(self):
        p, _ = _probs_logits(self)
        return _wrap(-((1 - p) * torch.log1p(-p) + p * torch.log(p)) / p)

    def sample(self, size=()):
        p = _probs_logits(self)[0]
        u = torch.rand(self._shape(size)).clamp_min(1e-20)
        return _wrap(torch.floor(torch.log(u) / torch.log1p(-p)))


class Poisson(_OwnDistribution):
    _torch_cls = _td.Poisson
    _arg_names = ('rate',)

    @property
    def mean(self):
        return _wrap(self.rate)

    @property
    def variance(self):
        return _wrap(self.rate)

    def log_prob(self, value):
        v = _t(value)
        return _wrap(v * torch.log(self.rate) - self.rate
                     - torch.lgamma(v + 1))

    def sample(self, size=()):
        # Knuth multiplication method per element (rates are small in
        # practice for the API's users; vectorized over the batch)
        lam = self.rate + torch.zeros(self._shape(size))
        L = torch.exp(-lam)
        k = torch.zeros_like(lam)
        p = torch.ones_like(lam)
        active = torch.ones_like(lam, dtype=torch.bool)
        for _ in range(10000):
            p = torch.where(active, p * torch.rand_like(p), p)
            newly_done = active & (p <= L)
            active = active & ~newly_done
            if not bool(active.any()):
                break
            k = k + active.to(k.dtype)
        return _wrap(k)


def _cat_logits(self):
    if getattr(self, 'logits', None) is not None:
        lg = self.logits
    else:
        lg = torch.log(self.probs.clamp_min(1e-30))
    return lg - torch.logsumexp(lg, dim=-1, keepdim=True)


class Categorical(_OwnDistribution):
    _torch_cls = _td.Categorical
    _arg_names = ('probs', 'logits')

    @property
    def mean(self):
        lg = _cat_logits(self)
        k = torch.arange(lg.shape[-1], dtype=torch.float32)
        return _wrap((lg.exp() * k).sum(-1))

    def log_prob(self, value):
        lg = _cat_logits(self)
        idx = _t(value).long()
        return _wrap(torch.gather(
            lg, -1, idx.unsqueeze(-1)).squeeze(-1))

    def entropy(self):
        lg = _cat_logits(self)
        return _wrap(-(lg.exp() * lg).sum(-1))

    def perplexity(self):
        return _wrap(torch.exp(_t(self.entropy())))

    def sample(self, size=()):
        lg = _cat_logits(self)
        shape = tuple(size) if not isinstance(size, int) else (size,)
        # Gumbel-max trick: argmax(logits + G) ~ Categorical(logits)
        g_shape = torch.Size(shape) + lg.shape
        u = torch.rand(g_shape).clamp_min(1e-20)
        g = -torch.log(-torch.log(u))
        return _wrap((lg + g).argmax(-1).to(torch.float32))

    def enumerate_support(self, expand=True):
        lg = _cat_logits(self)
        n = lg.shape[-1]
        vals = torch.arange(n, dtype=torch.float32)
        if expand and lg.dim() > 1:
            vals = vals.view((n,) + (1,) * (lg.dim() - 1)).expand(
                (n,) + lg.shape[:-1])
        return _wrap(vals)


class OneHotCategorical(Categorical):
    _torch_cls = _td.OneHotCategorical

    def log_prob(self, value):
        lg = _cat_logits(self)
        return _wrap((lg * _t(value)).sum(-1))

    def sample(self, size=()):
        idx = _t(super().sample(size)).long()
        lg = _cat_logits(self)
        return _wrap(torch.nn.functional.one_hot(
            idx, lg.shape[-1]).to(torch.float32))


# own KL formulas for the common pairs (reference divergence.py)
@register_kl(Normal, Normal)
def _kl_normal_normal(p, q):
    var_ratio = (p.scale / q.scale) ** 2
    t1 = ((p.loc - q.loc) / q.scale) ** 2
    return _wrap(0.5 * (var_ratio + t1 - 1 - torch.log(var_ratio)))


@register_kl(Bernoulli, Bernoulli)
def _kl_bern_bern(p, q):
    pp, _ = _probs_logits(p)
    qp, _ = _probs_logits(q)
    return _wrap(pp * (torch.log(pp) - torch.log(qp))
                 + (1 - pp) * (torch.log1p(-pp) - torch.log1p(-qp)))


@register_kl(Categorical, Categorical)
def _kl_cat_cat(p, q):
    lp = _cat_logits(p)
    lq = _cat_logits(q)
    return _wrap((lp.exp() * (lp - lq)).sum(-1))


@register_kl(Exponential, Exponential)
def _kl_exp_exp(p, q):
    r = p.rate / q.rate
    return _wrap(torch.log(r) + 1.0 / r - 1
                 + torch.zeros_like(p.rate + q.rate))


@register_kl(Uniform, Uniform)
def _kl_unif_unif(p, q):
    res = torch.log((q.high - q.low) / (p.high - p.low))
    oob = (q.low > p.low) | (q.high < p.high)
    return _wrap(torch.where(oob, torch.full_like(res, float('inf')), res))
