"""mx.np / mx.npx numpy-compatible API surface (reference python/mxnet/numpy)."""
import numpy as onp
import pytest

import mxnet_amd as mx
from mxnet_amd import numpy as np
from mxnet_amd.numpy import linalg as LA


def test_creation_and_math():
    a = np.arange(12).reshape(3, 4) if hasattr(np.arange(12), 'reshape') \
        else np.array(onp.arange(12).reshape(3, 4))
    x = np.array(onp.arange(12.).reshape(3, 4))
    assert x.shape == (3, 4)
    y = np.ones((3, 4))
    z = np.add(x, y)
    onp.testing.assert_allclose(z.asnumpy(), onp.arange(12.).reshape(3, 4) + 1)
    assert float(np.sum(z).asnumpy()) == float((onp.arange(12.) + 1).sum())
    onp.testing.assert_allclose(np.exp(np.zeros(3)).asnumpy(), onp.ones(3))
    m = np.matmul(x, np.ones((4, 2)))
    assert m.shape == (3, 2)


def test_reductions_and_sorting():
    x = np.array(onp.array([[3., 1., 2.], [0., 5., 4.]]))
    assert float(np.max(x).asnumpy()) == 5.0
    onp.testing.assert_allclose(np.mean(x, axis=0).asnumpy(),
                                [1.5, 3.0, 3.0])
    s = np.sort(x, axis=1) if hasattr(np, 'sort') else None
    if s is not None:
        onp.testing.assert_allclose(s.asnumpy()[0], [1., 2., 3.])
    am = np.argmax(x, axis=1)
    assert list(am.asnumpy()) == [0, 1]


def test_linspace_eye_full():
    onp.testing.assert_allclose(np.linspace(0, 1, 5).asnumpy(),
                                onp.linspace(0, 1, 5), rtol=1e-6)
    onp.testing.assert_allclose(np.eye(3).asnumpy(), onp.eye(3))
    onp.testing.assert_allclose(np.full((2, 2), 7.0).asnumpy(),
                                onp.full((2, 2), 7.0))


def test_linalg_suite():
    a = onp.random.RandomState(0).randn(4, 4).astype(onp.float32)
    spd = a @ a.T + 4 * onp.eye(4, dtype=onp.float32)
    x = np.array(spd)
    L = LA.cholesky(x)
    onp.testing.assert_allclose((L.handle @ L.handle.T).numpy(), spd,
                                rtol=1e-4, atol=1e-4)
    inv = LA.inv(x)
    onp.testing.assert_allclose((x.handle @ inv.handle).numpy(), onp.eye(4),
                                atol=1e-4)
    u, s, vt = LA.svd(np.array(a))
    assert s.shape == (4,)
    w, v = LA.eigh(x)
    assert float(w.handle.min()) > 0  # SPD
    n = LA.norm(np.array(a))
    onp.testing.assert_allclose(float(n.asnumpy()),
                                onp.linalg.norm(a), rtol=1e-5)


def test_einsum_and_random():
    x = np.array(onp.random.rand(3, 4).astype(onp.float32))
    y = np.array(onp.random.rand(4, 5).astype(onp.float32))
    z = np.einsum('ij,jk->ik', x, y)
    onp.testing.assert_allclose(z.asnumpy(), x.asnumpy() @ y.asnumpy(),
                                rtol=1e-5)
    from mxnet_amd.numpy import random as npr
    r = npr.uniform(0, 1, (100,)) if hasattr(npr, 'uniform') else None
    if r is not None:
        assert 0 <= float(r.handle.min()) and float(r.handle.max()) <= 1


def test_numpy_api_tail():
    """np tail coverage (reference src/operator/numpy np_* families)."""
    x = np.array(onp.arange(12.).reshape(3, 4))
    onp.testing.assert_allclose(np.trace(x).asnumpy(), 15.0)
    onp.testing.assert_allclose(np.flip(x, 1).asnumpy(),
                                onp.flip(onp.arange(12.).reshape(3, 4), 1))
    onp.testing.assert_allclose(np.median(x).asnumpy(),
                                onp.median(onp.arange(12.)))
    onp.testing.assert_allclose(
        np.pad(x, 1).asnumpy(), onp.pad(onp.arange(12.).reshape(3, 4), 1))
    onp.testing.assert_allclose(
        np.interp(np.array([0.5, 2.5]), np.array([0., 1., 2., 3.]),
                  np.array([0., 2., 4., 6.])).asnumpy(), [1.0, 5.0])
    onp.testing.assert_allclose(np.cov(x).asnumpy(),
                                onp.cov(onp.arange(12.).reshape(3, 4)),
                                rtol=1e-5)
    onp.testing.assert_allclose(
        np.gradient(x, axis=1).asnumpy(),
        onp.gradient(onp.arange(12.).reshape(3, 4), axis=1))
    h, e = np.histogram(x, bins=4)
    assert float(h.handle.sum()) == 12
    onp.testing.assert_allclose(np.kron(np.array(onp.eye(2)),
                                        np.array(onp.ones((2, 2)))).asnumpy(),
                                onp.kron(onp.eye(2), onp.ones((2, 2))))
    onp.testing.assert_allclose(
        np.polyval(np.array([1., 0., -1.]), np.array([2.0])).asnumpy(),
        [3.0])
    q, r = np.divmod(np.array([7., -7.]), np.array([3., 3.]))
    onp.testing.assert_allclose(q.asnumpy(), [2., -3.])
    onp.testing.assert_allclose(r.asnumpy(), [1., 2.])
