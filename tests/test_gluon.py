import os

import numpy as np
import pytest

import mxnet_amd as mx
from mxnet_amd import gluon, nd, autograd
from mxnet_amd.gluon import nn


def _lenet():
    net = nn.HybridSequential()
    net.add(nn.Conv2D(6, 5, activation='relu'),
            nn.MaxPool2D(2, 2),
            nn.Conv2D(16, 3, activation='relu'),
            nn.MaxPool2D(2, 2),
            nn.Flatten(),
            nn.Dense(120, activation='relu'),
            nn.Dense(84, activation='relu'),
            nn.Dense(10))
    return net


def test_dense_forward_backward():
    net = nn.Dense(4, in_units=3)
    net.initialize()
    x = nd.random_uniform(shape=(2, 3))
    with autograd.record():
        y = net(x)
        loss = (y * y).sum()
    loss.backward()
    w = net.weight
    assert w.grad().shape == (4, 3)
    assert float(w.grad().norm().asscalar()) > 0


def test_deferred_init():
    net = nn.Dense(4)
    net.initialize()
    x = nd.random_uniform(shape=(2, 7))
    y = net(x)
    assert y.shape == (2, 4)
    assert net.weight.shape == (4, 7)


def test_lenet_trains_cpu():
    """Milestone A (BASELINE config 1): Gluon LeNet, mx.cpu, KVStore local."""
    np.random.seed(0)
    net = _lenet()
    net.initialize(mx.init.Xavier(), ctx=mx.cpu())
    trainer = gluon.Trainer(net.collect_params(), 'adam',
                            {'learning_rate': 0.002}, kvstore='local')
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
    x = nd.random_uniform(shape=(32, 1, 28, 28))
    y = nd.array(np.random.randint(0, 10, (32,)).astype('float32'))
    first = None
    for _ in range(60):
        with autograd.record():
            loss = loss_fn(net(x), y)
        loss.backward()
        trainer.step(32)
        if first is None:
            first = loss.mean().asscalar()
    final = loss.mean().asscalar()
    assert final < first * 0.5, (first, final)


def test_save_load_parameters(tmp_path):
    net = _lenet()
    net.initialize()
    x = nd.random_uniform(shape=(2, 1, 28, 28))
    y1 = net(x)
    f = str(tmp_path / 'model.params')
    net.save_parameters(f)
    net2 = _lenet()
    net2.load_parameters(f)
    y2 = net2(x)
    assert np.allclose(y1.asnumpy(), y2.asnumpy(), atol=1e-6)


def test_collect_params_names():
    net = _lenet()
    names = list(net._collect_params_with_prefix().keys())
    assert '0.weight' in names
    assert '5.bias' in names


def test_sequential_slicing():
    net = _lenet()
    sub = net[:2]
    assert len(sub) == 2


def test_batchnorm_layer():
    net = nn.BatchNorm()
    net.initialize()
    x = nd.random_uniform(shape=(4, 3, 8, 8))
    with autograd.record():
        y = net(x)
    assert y.shape == x.shape
    # training-mode normalization: per-channel mean ~0
    m = y.asnumpy().mean(axis=(0, 2, 3))
    assert np.allclose(m, 0, atol=1e-3)


def test_batchnorm_fused_relu_residual():
    net = nn.BatchNorm(fuse_relu=True)
    net.initialize()
    x = nd.random_uniform(shape=(2, 4, 4, 3))  # will pick NCHW (axis=1)
    y = net(x)
    assert float(y.min().asscalar()) >= 0


def test_embedding_layer():
    net = nn.Embedding(10, 4)
    net.initialize()
    idx = nd.array([[1, 2], [3, 4]], dtype='int64')
    y = net(idx)
    assert y.shape == (2, 2, 4)


def test_layernorm_layer():
    net = nn.LayerNorm()
    net.initialize()
    x = nd.random_uniform(shape=(4, 16))
    y = net(x)
    assert np.allclose(y.asnumpy().mean(axis=-1), 0, atol=1e-5)


def test_dropout_modes():
    net = nn.Dropout(0.5)
    x = nd.ones((100, 100))
    y_pred = net(x)  # not recording -> identity
    assert np.allclose(y_pred.asnumpy(), 1)
    with autograd.record():
        y_train = net(x)
    frac_zero = (y_train.asnumpy() == 0).mean()
    assert 0.3 < frac_zero < 0.7


def test_losses():
    pred = nd.random_uniform(shape=(4, 5))
    label = nd.array([0, 1, 2, 3], dtype='float32')
    l = gluon.loss.SoftmaxCrossEntropyLoss()(pred, label)
    assert l.shape == (4,)
    l2 = gluon.loss.L2Loss()(pred, nd.zeros((4, 5)))
    assert l2.shape == (4,)
    l1 = gluon.loss.L1Loss()(pred, nd.zeros((4, 5)))
    hb = gluon.loss.HuberLoss()(pred, nd.zeros((4, 5)))
    bce = gluon.loss.SigmoidBCELoss()(pred, nd.ones((4, 5)))
    for x in (l1, hb, bce):
        assert x.shape == (4,)


def test_metrics():
    from mxnet_amd.gluon import metric
    acc = metric.Accuracy()
    pred = nd.array([[0.9, 0.1], [0.2, 0.8]])
    label = nd.array([0, 1])
    acc.update(label, pred)
    assert acc.get()[1] == 1.0
    topk = metric.TopKAccuracy(top_k=2)
    topk.update(label, pred)
    assert topk.get()[1] == 1.0


def test_trainer_allreduce_update_split():
    """allreduce_grads + update as separate calls (AMP pattern)."""
    net = nn.Dense(3, in_units=4)
    net.initialize()
    tr = gluon.Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.1},
                       kvstore='local')
    x = nd.random_uniform(shape=(2, 4))
    with autograd.record():
        loss = (net(x) ** 2).sum()
    loss.backward()
    tr.allreduce_grads()
    tr.update(2)


def test_constant_parameter():
    c = gluon.Constant([1.0, 2.0], name='c')
    c.initialize()
    assert np.allclose(c.data().asnumpy(), [1, 2])


@pytest.mark.parametrize('opt', ['sgd', 'adam', 'adamw', 'nag', 'rmsprop',
                                 'adagrad', 'lamb', 'signum'])
def test_optimizers_reduce_loss(opt):
    np.random.seed(0)
    net = nn.Dense(1, in_units=8)
    net.initialize()
    tr = gluon.Trainer(net.collect_params(), opt,
                       {'learning_rate': 0.05}, kvstore='local')
    x = nd.random_uniform(shape=(16, 8))
    y = nd.random_uniform(shape=(16, 1))
    loss_fn = gluon.loss.L2Loss()
    first = None
    for _ in range(40):
        with autograd.record():
            loss = loss_fn(net(x), y).mean()
        loss.backward()
        tr.step(16)
        if first is None:
            first = loss.asscalar()
    assert loss.asscalar() < first
