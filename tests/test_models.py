"""Model-family smoke tests on CPU (fwd+bwd+step, fp32 oracle path)."""
import pytest
import torch

import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import Trainer
from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
from mxnet_amd.ndarray.ndarray import NDArray


def _train_step(net, x, label, classes):
    loss_fn = SoftmaxCrossEntropyLoss()
    with autograd.record():
        out = net(x)
        L = loss_fn(out, label)
    L.backward()
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.01, 'momentum': 0.9}, kvstore='local')
    tr.step(x.shape[0])
    val = float(L.mean().asnumpy())
    assert val == val and abs(val) < 1e3, f'bad loss {val}'
    return val


@pytest.mark.parametrize('name', ['resnet18_v1', 'resnet50_v1', 'resnet18_v2',
                                  'vgg11', 'alexnet', 'squeezenet1_0',
                                  'densenet121', 'mobilenet0_5', 'resnext50_32x4d',
                                  'inception_v3'])
def test_vision_model(name):
    from mxnet_amd.gluon.model_zoo import vision
    size = 299 if name == 'inception_v3' else 64
    if name in ('alexnet', 'vgg11'):
        size = 224
    net = getattr(vision, name)(classes=10)
    net.initialize()
    x = mx.nd.array(torch.randn(2, 3, size, size))
    _train_step(net, x, mx.nd.array(torch.randint(0, 10, (2,))), 10)


def test_vision_model_nhwc():
    from mxnet_amd.gluon.model_zoo import vision
    net = vision.resnet18_v1(classes=10, layout='NHWC')
    net.initialize()
    x = mx.nd.array(torch.randn(2, 64, 64, 3))
    _train_step(net, x, mx.nd.array(torch.randint(0, 10, (2,))), 10)


def test_bert_model():
    from mxnet_amd.models.bert import BERTModel
    net = BERTModel(vocab_size=500, units=64, hidden_size=128, num_layers=2,
                    num_heads=4, max_length=64)
    net.initialize()
    tokens = mx.nd.from_torch(torch.randint(0, 500, (2, 16)))
    mask = mx.nd.from_torch(torch.ones(2, 16, dtype=torch.bool))
    with autograd.record():
        seq, pooled, mlm, nsp = net(tokens, None, mask)
        L = NDArray(mlm.handle.float().mean() + nsp.handle.float().mean())
    L.backward()
    tr = Trainer(net.collect_params(), 'adam', {'learning_rate': 1e-4},
                 kvstore='local')
    tr.step(2)
    assert seq.shape == (2, 16, 64)


def test_lstm_model():
    from mxnet_amd.gluon import rnn
    net = rnn.LSTM(hidden_size=32, num_layers=2)
    net.initialize()
    x = mx.nd.array(torch.randn(5, 3, 16))  # [T, N, I]
    with autograd.record():
        out = net(x)
        L = NDArray(out.handle.float().mean())
    L.backward()
    assert out.shape == (5, 3, 32)


def test_lenet_mnist_cpu_local_kvstore():
    """BASELINE config 1: Gluon LeNet on MNIST-shaped synthetic tensors,
    ctx=cpu, KVStore='local' — loss decreases over a few steps."""
    import torch
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, nn
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    torch.manual_seed(0)
    net = nn.HybridSequential()
    net.add(nn.Conv2D(6, 5, padding=2, activation='relu', in_channels=1),
            nn.MaxPool2D(2),
            nn.Conv2D(16, 5, activation='relu', in_channels=6),
            nn.MaxPool2D(2),
            nn.Flatten(),
            nn.Dense(120, activation='relu', in_units=16 * 5 * 5),
            nn.Dense(84, activation='relu', in_units=120),
            nn.Dense(10, in_units=84))
    net.initialize(ctx=mx.cpu())
    tr = Trainer(net.collect_params(), 'adam',
                 {'learning_rate': 2e-3}, kvstore='local')
    loss_fn = SoftmaxCrossEntropyLoss()
    X = mx.nd.from_torch(torch.randn(32, 1, 28, 28))
    Y = mx.nd.from_torch(torch.randint(0, 10, (32,)))
    losses = []
    for _ in range(60):
        with autograd.record():
            L = loss_fn(net(X), Y)
        L.backward()
        tr.step(32)
        losses.append(float(L.handle.mean()))
    # memorizing one batch: the loss must fall decisively
    assert losses[-1] < 1.2, (losses[0], losses[-1])
