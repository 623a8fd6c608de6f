"""gluon.probability own-math distributions validated against
torch.distributions closed forms (the reference writes the same
formulas over mx.nd ops — gluon/probability/distributions/*.py)."""
import numpy as np
import pytest
import torch
import torch.distributions as td

import mxnet_amd as mx
from mxnet_amd.gluon.probability import distributions as D


def _nd(a):
    return mx.nd.array(np.asarray(a, dtype='float32'))


CASES = [
    (D.Normal, td.Normal, {'loc': [0.0, 1.5], 'scale': [1.0, 0.5]},
     [[-0.3, 2.0]]),
    (D.LogNormal, td.LogNormal, {'loc': [0.0, 0.2], 'scale': [1.0, 0.5]},
     [[0.7, 2.0]]),
    (D.Laplace, td.Laplace, {'loc': [0.0, -1.0], 'scale': [1.0, 2.0]},
     [[0.4, -3.0]]),
    (D.Uniform, td.Uniform, {'low': [0.0, -1.0], 'high': [1.0, 3.0]},
     [[0.5, 0.0]]),
    (D.Exponential, td.Exponential, {'rate': [1.0, 3.0]}, [[0.2, 1.0]]),
    (D.Gumbel, td.Gumbel, {'loc': [0.0, 2.0], 'scale': [1.0, 0.7]},
     [[0.1, 1.0]]),
    (D.Bernoulli, td.Bernoulli, {'probs': [0.3, 0.8]}, [[1.0, 0.0]]),
    (D.Geometric, td.Geometric, {'probs': [0.3, 0.6]}, [[2.0, 0.0]]),
    (D.Poisson, td.Poisson, {'rate': [1.5, 4.0]}, [[2.0, 3.0]]),
]


@pytest.mark.parametrize('own_cls,torch_cls,args,values',
                         CASES, ids=[c[0].__name__ for c in CASES])
def test_log_prob_mean_var_match_torch(own_cls, torch_cls, args, values):
    own = own_cls(**{k: _nd(v) for k, v in args.items()})
    ref = torch_cls(**{k: torch.tensor(v) for k, v in args.items()})
    for v in values:
        np.testing.assert_allclose(own.log_prob(_nd(v)).asnumpy(),
                                   ref.log_prob(torch.tensor(v)).numpy(),
                                   rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(own.mean.asnumpy(), ref.mean.numpy(),
                               rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(own.variance.asnumpy(),
                               ref.variance.numpy(), rtol=1e-5, atol=1e-6)
    try:
        ref_ent = ref.entropy().numpy()
    except NotImplementedError:
        ref_ent = None
    if ref_ent is not None:
        np.testing.assert_allclose(own.entropy().asnumpy(), ref_ent,
                                   rtol=1e-5, atol=1e-6)


def test_categorical_matches_torch():
    logits = [[0.5, -1.0, 2.0], [0.0, 0.0, 0.0]]
    own = D.Categorical(logits=_nd(logits))
    ref = td.Categorical(logits=torch.tensor(logits))
    val = [2.0, 0.0]
    np.testing.assert_allclose(
        own.log_prob(_nd(val)).asnumpy(),
        ref.log_prob(torch.tensor(val).long()).numpy(), rtol=1e-5)
    np.testing.assert_allclose(own.entropy().asnumpy(),
                               ref.entropy().numpy(), rtol=1e-5)
    s = own.sample((5000,)).asnumpy()
    assert s.shape == (5000, 2)
    # empirical frequencies track softmax(logits) row 0
    freq = np.bincount(s[:, 0].astype(int), minlength=3) / 5000
    np.testing.assert_allclose(freq, ref.probs[0].numpy(), atol=0.03)


def test_onehot_categorical():
    own = D.OneHotCategorical(probs=_nd([0.2, 0.3, 0.5]))
    ref = td.OneHotCategorical(probs=torch.tensor([0.2, 0.3, 0.5]))
    v = [0.0, 1.0, 0.0]
    np.testing.assert_allclose(own.log_prob(_nd(v)).asnumpy(),
                               ref.log_prob(torch.tensor(v)).numpy(),
                               rtol=1e-5)
    s = own.sample((10,)).asnumpy()
    assert s.shape == (10, 3) and (s.sum(-1) == 1).all()


def test_sampling_moments():
    rs = torch.manual_seed(0)
    n = 40000
    d = D.Normal(loc=_nd([2.0]), scale=_nd([3.0]))
    s = d.sample((n,)).asnumpy()
    assert abs(s.mean() - 2.0) < 0.08 and abs(s.std() - 3.0) < 0.08
    e = D.Exponential(rate=_nd([2.0])).sample((n,)).asnumpy()
    assert abs(e.mean() - 0.5) < 0.02
    p = D.Poisson(rate=_nd([3.0])).sample((n,)).asnumpy()
    assert abs(p.mean() - 3.0) < 0.1 and abs(p.var() - 3.0) < 0.25
    b = D.Bernoulli(probs=_nd([0.7])).sample((n,)).asnumpy()
    assert abs(b.mean() - 0.7) < 0.02


def test_kl_formulas_match_torch():
    pairs = [
        (D.Normal(loc=_nd([0.0]), scale=_nd([1.0])),
         D.Normal(loc=_nd([1.0]), scale=_nd([2.0])),
         td.Normal(torch.tensor([0.0]), torch.tensor([1.0])),
         td.Normal(torch.tensor([1.0]), torch.tensor([2.0]))),
        (D.Bernoulli(probs=_nd([0.3])), D.Bernoulli(probs=_nd([0.6])),
         td.Bernoulli(torch.tensor([0.3])),
         td.Bernoulli(torch.tensor([0.6]))),
        (D.Categorical(probs=_nd([0.2, 0.8])),
         D.Categorical(probs=_nd([0.5, 0.5])),
         td.Categorical(torch.tensor([0.2, 0.8])),
         td.Categorical(torch.tensor([0.5, 0.5]))),
        (D.Exponential(rate=_nd([2.0])), D.Exponential(rate=_nd([0.5])),
         td.Exponential(torch.tensor([2.0])),
         td.Exponential(torch.tensor([0.5]))),
    ]
    for p, q, tp, tq in pairs:
        np.testing.assert_allclose(
            D.kl_divergence(p, q).asnumpy(),
            td.kl.kl_divergence(tp, tq).numpy(), rtol=1e-5, atol=1e-6)


def test_rsample_grad_flows():
    loc = torch.tensor([0.5], requires_grad=True)
    d = D.Normal(loc=mx.nd.from_torch(loc), scale=_nd([1.0]))
    y = d.rsample((8,))
    y.handle.sum().backward()
    assert loc.grad is not None and float(loc.grad) == 8.0
