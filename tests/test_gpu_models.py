"""Every model family takes a real fp16 training step on the MI355X
(catches GPU-dispatch gaps outside the ResNet hot path)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


def _train_step(net, x_shape, classes=10, dtype='float16'):
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    net.initialize(ctx=mx.gpu(0))
    net.cast(dtype)
    tdt = torch.float16 if dtype == 'float16' else torch.float32
    x = mx.nd.from_torch(torch.randn(*x_shape, device=DEV, dtype=tdt))
    label = mx.nd.from_torch(torch.randint(0, classes, (x_shape[0],),
                                           device=DEV))
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.01, 'momentum': 0.9,
                  'multi_precision': True}, kvstore=None)
    loss_fn = SoftmaxCrossEntropyLoss()
    for _ in range(2):
        with autograd.record():
            out = net(x)
            L = loss_fn(out, label)
        L.backward()
        tr.step(x_shape[0])
    val = float(L.mean().asnumpy())
    assert val == val and abs(val) < 1e4, f'bad loss {val}'
    return val


@pytest.mark.parametrize('name,shape', [
    ('resnet50_v1', (4, 64, 64, 3)),        # NHWC hot path
    ('resnet18_v2', (4, 64, 64, 3)),
    ('vgg11', (2, 224, 224, 3)),
    ('alexnet', (4, 3, 224, 224)),          # NCHW path + LRN-free
    ('squeezenet1_0', (4, 3, 96, 96)),
    ('densenet121', (2, 3, 64, 64)),
    ('mobilenet0_5', (4, 3, 64, 64)),       # depthwise kernels
    ('mobilenet_v2_1_0', (4, 3, 64, 64)),
    ('inception_v3', (2, 3, 299, 299)),
    ('resnext50_32x4d', (2, 64, 64, 3)),  # grouped igemm kernels
])
def test_vision_family_gpu(name, shape):
    from mxnet_amd.gluon.model_zoo import vision
    kwargs = {'classes': 10}
    if len(shape) == 4 and shape[-1] == 3:
        kwargs['layout'] = 'NHWC'
    net = getattr(vision, name)(**kwargs)
    # vgg keeps exercising the fp32 kernel paths; the rest run fp16
    dtype = 'float32' if name.startswith('vgg') else 'float16'
    _train_step(net, shape, dtype=dtype)


def test_bert_gpu():
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.models.bert import BERTModel
    from mxnet_amd.ndarray.ndarray import NDArray
    net = BERTModel(vocab_size=1000, units=128, hidden_size=256,
                    num_layers=2, num_heads=4, max_length=64)
    net.initialize(ctx=mx.gpu(0))
    net.cast('float16')
    tokens = mx.nd.from_torch(torch.randint(0, 1000, (4, 32), device=DEV))
    mask = mx.nd.from_torch(torch.ones(4, 32, dtype=torch.bool, device=DEV))
    tr = Trainer(net.collect_params(), 'adam',
                 {'learning_rate': 1e-4, 'multi_precision': True},
                 kvstore=None)
    with autograd.record():
        seq, pooled, mlm, nsp = net(tokens, None, mask)
        L = NDArray((mlm.handle.float().mean() +
                     nsp.handle.float().mean()))
    L.backward()
    tr.step(4)
    assert np.isfinite(L.asnumpy()).all()


def test_lstm_gpu():
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, rnn, nn
    from mxnet_amd.gluon.block import Block
    from mxnet_amd.ndarray.ndarray import NDArray

    class M(Block):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(100, 64)
            self.lstm = rnn.LSTM(hidden_size=64, num_layers=2)
            self.fc = nn.Dense(100, flatten=False)

        def forward(self, x):
            return self.fc(self.lstm(self.emb(x)))

    net = M()
    net.initialize(ctx=mx.gpu(0))
    net.cast('float16')
    x = mx.nd.from_torch(torch.randint(0, 100, (8, 4), device=DEV))
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.1, 'multi_precision': True},
                 kvstore=None)
    with autograd.record():
        out = net(x)
        L = NDArray(out.handle.float().mean())
    L.backward()
    tr.step(4)
    assert np.isfinite(L.asnumpy()).all()


def test_export_symbolblock_serve_gpu(tmp_path):
    """Reference deploy workflow on GPU: export -> SymbolBlock.imports ->
    hipGraph-captured serving matches the exporting net."""
    import os
    import mxnet_amd as mx
    from mxnet_amd.gluon import SymbolBlock
    from mxnet_amd.gluon.model_zoo import vision
    net = vision.resnet18_v1(classes=10)
    net.initialize(ctx=mx.gpu(0))
    net.cast('float16')
    x = mx.nd.from_torch(torch.randn(2, 3, 64, 64, device=DEV).half())
    ref = net(x).asnumpy()
    sym_f, par_f = net.export(str(tmp_path / 'm'))
    assert os.path.exists(sym_f) and os.path.exists(par_f)
    served = SymbolBlock.imports(sym_f, ['data'], par_f, ctx=mx.gpu(0))
    served.cast('float16')
    out = served(x).asnumpy()
    np.testing.assert_allclose(out.astype(np.float32),
                               ref.astype(np.float32), rtol=2e-2, atol=2e-2)


@pytest.mark.gpu
def test_embedding_sparse_grad_gpu():
    """Sparse-grad Embedding + lazy SGD on device tensors."""
    import torch
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, nn as gnn
    torch.manual_seed(2)
    emb = gnn.Embedding(1000, 64, sparse_grad=True)
    emb.initialize(ctx=mx.gpu(0))
    tr = Trainer(emb.collect_params(), 'sgd',
                 {'learning_rate': 0.1, 'momentum': 0.9}, kvstore=None)
    w0 = emb.weight.data().handle.clone()
    idx = mx.nd.from_torch(torch.randint(0, 50, (8, 16), device='cuda'))
    for _ in range(2):
        with autograd.record():
            out = emb(idx)
            L = mx.nd.from_torch((out.handle.float() ** 2).sum())
        L.backward()
        tr.step(1)
    w1 = emb.weight.data().handle
    touched = torch.unique(idx.handle)
    moved = (w1 - w0).abs().sum(dim=1)
    assert (moved[touched] > 0).all()
    untouched = torch.ones(1000, dtype=torch.bool, device='cuda')
    untouched[touched] = False
    assert moved[untouched].abs().max() == 0
