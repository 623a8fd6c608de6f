import numpy as np
import pytest

import mxnet_amd as mx
from mxnet_amd import nd


def test_create_and_convert():
    a = nd.array([[1, 2], [3, 4]])
    assert a.shape == (2, 2)
    assert a.dtype == np.float32
    assert np.allclose(a.asnumpy(), [[1, 2], [3, 4]])
    b = nd.array(np.arange(6, dtype='int32').reshape(2, 3))
    assert b.dtype == np.int32


def test_creation_ops():
    assert nd.zeros((2, 3)).asnumpy().sum() == 0
    assert nd.ones((2, 3)).asnumpy().sum() == 6
    assert nd.full((2, 2), 7).asnumpy().sum() == 28
    assert np.allclose(nd.arange(5).asnumpy(), np.arange(5))


def test_arithmetic():
    a = nd.array([1.0, 2.0, 3.0])
    b = nd.array([4.0, 5.0, 6.0])
    assert np.allclose((a + b).asnumpy(), [5, 7, 9])
    assert np.allclose((a - b).asnumpy(), [-3, -3, -3])
    assert np.allclose((a * b).asnumpy(), [4, 10, 18])
    assert np.allclose((b / a).asnumpy(), [4, 2.5, 2])
    assert np.allclose((a ** 2).asnumpy(), [1, 4, 9])
    assert np.allclose((2 + a).asnumpy(), [3, 4, 5])
    a += 1
    assert np.allclose(a.asnumpy(), [2, 3, 4])


def test_reductions():
    a = nd.array(np.arange(12, dtype='float32').reshape(3, 4))
    assert a.sum().asscalar() == 66
    assert np.allclose(a.sum(axis=0).asnumpy(), [12, 15, 18, 21])
    assert np.allclose(a.mean(axis=1).asnumpy(), [1.5, 5.5, 9.5])
    assert a.max().asscalar() == 11
    assert a.argmax(axis=1).asnumpy().tolist() == [3, 3, 3]


def test_shape_ops():
    a = nd.array(np.arange(24, dtype='float32').reshape(2, 3, 4))
    assert a.reshape((6, 4)).shape == (6, 4)
    assert a.reshape((-1, 4)).shape == (6, 4)
    assert a.reshape((0, -1)).shape == (2, 12)
    assert a.transpose().shape == (4, 3, 2)
    assert a.transpose((1, 0, 2)).shape == (3, 2, 4)
    assert a.expand_dims(0).shape == (1, 2, 3, 4)
    assert a.flatten().shape == (2, 12)
    assert nd.concat(a, a, dim=1).shape == (2, 6, 4)
    assert nd.stack(a, a, axis=0).shape == (2, 2, 3, 4)


def test_indexing():
    a = nd.array(np.arange(20, dtype='float32').reshape(4, 5))
    assert a[1, 2].asscalar() == 7
    assert a[1:3].shape == (2, 5)
    a[0, :] = 0
    assert a[0].asnumpy().sum() == 0
    idx = nd.array([0, 2], dtype='int64')
    assert nd.take(a, idx).shape == (2, 5)


def test_nn_ops_cpu():
    x = nd.random_uniform(shape=(2, 8))
    assert np.allclose(nd.softmax(x).asnumpy().sum(axis=1), 1, atol=1e-5)
    r = nd.relu(nd.array([-1.0, 1.0]))
    assert np.allclose(r.asnumpy(), [0, 1])
    s = nd.sigmoid(nd.zeros((3,)))
    assert np.allclose(s.asnumpy(), 0.5)


def test_dtype_cast():
    a = nd.ones((2, 2))
    h = a.astype('float16')
    assert h.dtype == np.float16
    back = h.astype('float32')
    assert np.allclose(back.asnumpy(), 1)


def test_context():
    a = nd.ones((2,), ctx=mx.cpu())
    assert a.context == mx.cpu()
    b = a.as_in_context(mx.cpu())
    assert b.context.device_type == 'cpu'


def test_broadcast_binary():
    a = nd.ones((2, 1, 3))
    b = nd.ones((1, 4, 3))
    assert (a + b).shape == (2, 4, 3)
    assert np.allclose(nd.broadcast_maximum(a * 3, b).asnumpy(), 3)


def test_where_clip():
    a = nd.array([-2.0, 0.5, 3.0])
    assert np.allclose(a.clip(-1, 1).asnumpy(), [-1, 0.5, 1])
    cond = nd.array([1.0, 0.0, 1.0])
    w = nd.where(cond, a, -a)
    assert np.allclose(w.asnumpy(), [-2, -0.5, 3])


def test_topk_sort():
    a = nd.array([3.0, 1.0, 2.0])
    assert nd.topk(a, k=2).asnumpy().tolist() == [0, 2]
    assert nd.sort(a).asnumpy().tolist() == [1, 2, 3]


def test_waitall():
    nd.waitall()  # no-op on CPU, must not raise


def test_sparse_ndarray():
    """row_sparse / CSR storage (reference test_sparse_ndarray.py)."""
    import numpy as np
    import torch
    from mxnet_amd.ndarray import sparse as S
    rs = S.row_sparse_array((np.ones((2, 3)), [1, 4]), shape=(6, 3))
    d = rs.tostype('default').asnumpy()
    assert d[1].sum() == 3 and d[0].sum() == 0
    csr = S.csr_matrix(np.eye(4, dtype=np.float32))
    out = S.sparse_dot(csr, torch.ones(4, 2))
    assert out.shape == (4, 2)
    both = S.add(rs, S.row_sparse_array((np.ones((1, 3)), [4]), shape=(6, 3)))
    assert both.tostype('default').asnumpy()[4].sum() == 6
    assert S.retain(rs, [4]).indices.tolist() == [4]
    assert S.zeros('row_sparse', (5, 2)).tostype('default').asnumpy().sum() == 0


def test_embedding_sparse_grad_end_to_end():
    """Embedding(sparse_grad=True) -> RowSparse grad -> SGD lazy update
    matches the dense path; untouched rows never move under momentum
    (reference: sparse_grad Embedding + sgd lazy_update)."""
    import torch
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, nn as gnn

    results = {}
    for sparse in (False, True):
        torch.manual_seed(5)
        emb = gnn.Embedding(50, 8, sparse_grad=sparse)
        emb.initialize()
        tr = Trainer(emb.collect_params(), 'sgd',
                     {'learning_rate': 0.1, 'momentum': 0.9,
                      'lazy_update': sparse}, kvstore=None)
        idx = mx.nd.from_torch(torch.tensor([[1, 3, 3], [7, 1, 4]]))
        for _ in range(2):
            with autograd.record():
                out = emb(idx)
                L = mx.nd.from_torch((out.handle ** 2).sum())
            L.backward()
            tr.step(1)
        results[sparse] = emb.weight.data().asnumpy().copy()

    touched = [1, 3, 4, 7]
    onp = np
    # touched rows agree between dense and sparse paths
    onp.testing.assert_allclose(results[True][touched],
                                results[False][touched], rtol=1e-5, atol=1e-6)
    # untouched rows identical to init in both
    torch.manual_seed(5)
    emb0 = gnn.Embedding(50, 8)
    emb0.initialize()
    w0 = emb0.weight.data().asnumpy()
    untouched = [r for r in range(50) if r not in touched]
    onp.testing.assert_allclose(results[True][untouched], w0[untouched])
    onp.testing.assert_allclose(results[False][untouched], w0[untouched])


def test_fp64_op_coverage():
    """Core op set runs in float64 end-to-end (reference supports fp64
    across nn/tensor families; robustness audit)."""
    import torch
    x64 = mx.nd.from_torch(torch.randn(4, 6).double())
    w64 = mx.nd.from_torch(torch.randn(3, 6).double())
    assert nd.FullyConnected(x64, w64, None, num_hidden=3,
                             no_bias=True).handle.dtype == torch.float64
    assert nd.softmax(x64).handle.dtype == torch.float64
    x4 = mx.nd.from_torch(torch.randn(1, 3, 8, 8).double())
    wc = mx.nd.from_torch(torch.randn(5, 3, 3, 3).double())
    y = nd.Convolution(x4, wc, None, kernel=(3, 3), num_filter=5,
                       no_bias=True)
    assert y.handle.dtype == torch.float64
    g = mx.nd.from_torch(torch.ones(3).double())
    b = mx.nd.from_torch(torch.zeros(3).double())
    rm = mx.nd.from_torch(torch.zeros(3).double())
    rv = mx.nd.from_torch(torch.ones(3).double())
    assert nd.BatchNorm(x4, g, b, rm, rv).handle.dtype == torch.float64
    from mxnet_amd.gluon import nn as gnn
    net = gnn.Dense(4, in_units=6, dtype='float64')
    net.initialize()
    assert net(x64).handle.dtype == torch.float64


def test_tensor_op_tail():
    """moments/cumsum/diag/trace/meshgrid/spatial-transformer and friends
    (reference tensor/matrix_op + spatial_transformer.cc)."""
    import torch
    x = mx.nd.from_torch(torch.arange(12.).reshape(3, 4))
    m, v = mx.nd.moments(x, axes=[0])
    np.testing.assert_allclose(m.asnumpy(),
                               x.asnumpy().mean(0), rtol=1e-6)
    assert float(mx.nd.cumsum(x, axis=1).handle[0, -1]) == 6
    assert float(mx.nd.trace(x).handle) == 15
    g = mx.nd.meshgrid(mx.nd.from_torch(torch.arange(3.)),
                       mx.nd.from_torch(torch.arange(2.)))
    ref = np.meshgrid(np.arange(3.), np.arange(2.))
    np.testing.assert_allclose(g[0].asnumpy(), ref[0])
    idx = mx.nd.unravel_index(mx.nd.from_torch(torch.tensor([5, 11])),
                              (3, 4))
    assert idx.handle.tolist() == [[1, 2], [1, 3]]
    assert mx.nd.ravel_multi_index(idx, (3, 4)).handle.tolist() == [5, 11]
    assert mx.nd.bincount(
        mx.nd.from_torch(torch.tensor([0, 1, 1, 3]))).handle.tolist() \
        == [1, 2, 0, 1]
    # spatial transformer with identity affine returns the input
    data = mx.nd.from_torch(torch.randn(2, 3, 8, 8))
    theta = mx.nd.from_torch(torch.tensor([[1., 0, 0, 0, 1, 0]] * 2))
    out = mx.nd.SpatialTransformer(data, theta, target_shape=(8, 8))
    np.testing.assert_allclose(out.asnumpy(), data.asnumpy(), atol=1e-4)
    s2d = mx.nd.space_to_depth(
        mx.nd.from_torch(torch.arange(16.).reshape(1, 1, 4, 4)), 2)
    assert s2d.shape == (1, 4, 2, 2)
