"""Native-runtime frontend: Gluon models training on the own C++ runtime
(pooled HIP storage + threaded engine + registry ops + own autograd tape)
with torch.autograd not involved (VERDICT item 4 done-criterion:
LeNet config-1 end-to-end; numerics cross-checked against the
torch-backed path on identical init/data)."""
import random

import numpy as np
import pytest
import torch

import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.base import set_native
from mxnet_amd.gluon import nn, Trainer
from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss


@pytest.fixture
def native():
    prev = set_native(True)
    yield
    set_native(prev)


def _lenet():
    net = nn.HybridSequential()
    net.add(nn.Conv2D(6, kernel_size=5, padding=2, activation='relu',
                      layout='NHWC'),
            nn.MaxPool2D(pool_size=2, strides=2, layout='NHWC'),
            nn.Conv2D(16, kernel_size=5, activation='relu', layout='NHWC'),
            nn.MaxPool2D(pool_size=2, strides=2, layout='NHWC'),
            nn.Dense(120, activation='relu'),
            nn.Dense(84, activation='relu'),
            nn.Dense(10))
    return net


def test_native_ndarray_basics(native):
    a = mx.nd.ones((3, 4))
    assert a.is_native
    b = mx.nd.array([[1, 2, 3, 4]] * 3)
    c = (a + b * 2).sum(axis=1)
    np.testing.assert_allclose(c.asnumpy(), [24, 24, 24])
    d = a.astype('float16')
    assert str(d.dtype) == 'float16'
    np.testing.assert_allclose(d.astype('float32').asnumpy(), a.asnumpy())


def test_native_lenet_trains(native):
    """Gluon LeNet, ctx=cpu, end-to-end on the native runtime
    (BASELINE config 1; torch.autograd never sees these arrays)."""
    random.seed(0); np.random.seed(0); torch.manual_seed(0)
    net = _lenet()
    net.initialize()
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.5, 'momentum': 0.9}, kvstore=None)
    lf = SoftmaxCrossEntropyLoss()
    rs = np.random.RandomState(0)
    x = mx.nd.array(rs.randn(16, 28, 28, 1))
    y = mx.nd.array(rs.randint(0, 10, (16,)).astype('int64'))
    assert x.is_native
    losses = []
    for _ in range(20):
        with autograd.record():
            L = lf(net(x), y).mean()
        L.backward()
        tr.step(1)
        losses.append(float(L.asnumpy()))
    mx.nd.waitall()
    # monotone-ish decrease (this config converges slowly by design; the
    # trajectory-equality test below is the strong numerics check)
    assert losses[-1] < losses[0] - 0.2, losses


def test_native_matches_torch_backend():
    """Same init/data on both runtimes -> identical loss trajectories."""
    def run(native_flag):
        prev = set_native(native_flag)
        try:
            random.seed(0); np.random.seed(0); torch.manual_seed(0)
            net = nn.HybridSequential()
            net.add(nn.Conv2D(6, kernel_size=5, padding=2,
                              activation='relu', layout='NHWC'),
                    nn.MaxPool2D(pool_size=2, strides=2, layout='NHWC'),
                    nn.Dense(32, activation='relu'),
                    nn.Dense(10))
            net.initialize()
            tr = Trainer(net.collect_params(), 'sgd',
                         {'learning_rate': 0.5, 'momentum': 0.9},
                         kvstore=None)
            lf = SoftmaxCrossEntropyLoss()
            rs = np.random.RandomState(0)
            x = mx.nd.array(rs.randn(16, 12, 12, 1))
            y = mx.nd.array(rs.randint(0, 10, (16,)).astype('int64'))
            out = []
            for _ in range(10):
                with autograd.record():
                    L = lf(net(x), y).mean()
                L.backward()
                tr.step(1)
                out.append(float(L.asnumpy()))
            return out
        finally:
            set_native(prev)

    torch_traj = run(False)
    native_traj = run(True)
    np.testing.assert_allclose(native_traj, torch_traj, atol=2e-3)
    assert native_traj[-1] < native_traj[0] * 0.7


def test_native_bn_dropout_embedding(native):
    random.seed(0); np.random.seed(0); torch.manual_seed(0)
    net = nn.HybridSequential()
    net.add(nn.Conv2D(8, kernel_size=3, padding=1, layout='NHWC'),
            nn.BatchNorm(axis=-1),
            nn.Activation('relu'),
            nn.Dense(16, activation='tanh'),
            nn.Dropout(0.5),
            nn.Dense(4))
    net.initialize()
    tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.1},
                 kvstore=None)
    lf = SoftmaxCrossEntropyLoss()
    rs = np.random.RandomState(1)
    x = mx.nd.array(rs.randn(8, 6, 6, 3))
    y = mx.nd.array(rs.randint(0, 4, (8,)).astype('int64'))
    for _ in range(3):
        with autograd.record():
            L = lf(net(x), y).mean()
        L.backward()
        tr.step(1)
    v = float(L.asnumpy())
    assert v == v  # finite
    # eval mode uses running stats
    out = net(x)
    assert out.shape == (8, 4)


def test_native_gpu_guard(native):
    """Native ops must fail loudly rather than fall back silently."""
    a = mx.nd.ones((2, 2))
    with pytest.raises(Exception):
        mx.nd.ops._ninv('definitely_not_an_op', [a], {})


def test_native_basic_indexing(native):
    """Basic int/slice indexing on native arrays is the recorded
    `_strided_copy` op with a scatter backward (reference
    ndarray.py slicing; here reduce.hip _strided_copy)."""
    ref = np.arange(24, dtype=np.float32).reshape(4, 6)
    x = mx.nd.array(ref)
    assert x.is_native
    for key in [(slice(1, 3), slice(None, None, 2)), (slice(None), 0),
                2, (3, 5), (slice(None, None, -1),), (Ellipsis, 1)]:
        np.testing.assert_allclose(x[key].asnumpy(), np.ascontiguousarray(ref[key]))
    # backward scatters into the leaf
    w = mx.nd.array(ref)
    w.attach_grad()
    with autograd.record():
        z = (w[1:3, ::3] * 2.0).sum()
    z.backward()
    exp = np.zeros_like(ref)
    exp[1:3, ::3] = 2.0
    np.testing.assert_allclose(w.grad.asnumpy(), exp)


def test_native_bert_matches_torch_backend():
    """Tiny BERT: native runtime and torch frontend produce the same
    loss and qkv gradient on identical params/data (pos-embed slice,
    additive mask, composed attention, pick-NLL loss all native)."""
    from mxnet_amd.models.bert import BERTModel
    from mxnet_amd.ndarray import ops as F

    def run(use_native):
        prev = set_native(use_native)
        try:
            torch.manual_seed(0)
            net = BERTModel(vocab_size=50, units=16, hidden_size=32,
                            num_layers=2, num_heads=2, max_length=32,
                            dropout=0.0)
            net.initialize(ctx=mx.cpu())
            rng = np.random.RandomState(42)
            tokens = mx.nd.array(rng.randint(0, 50, (2, 8)), dtype='int64')
            mask = mx.nd.array((rng.rand(2, 8) > 0.2).astype('float32'))
            net(tokens, None, mask)  # finish deferred init
            params = net.collect_params()
            for k, p in params.items():
                rs = np.random.RandomState(
                    sum(ord(c) for c in k) % (2**31))
                p.set_data(mx.nd.array(
                    rs.randn(*p.shape).astype('float32') * 0.05))
            with autograd.record():
                seq, pooled, mlm, nsp = net(tokens, None, mask)
                logp = F.log_softmax(mlm.reshape(-1, 50))
                lab = mx.nd.array(rng.randint(0, 50, (16,)),
                                  dtype='float32')
                loss = (F.pick(logp, lab, axis=-1) * -1.0).mean()
            loss.backward()
            kq = [k for k in params if 'qkv' in k and 'weight' in k][0]
            return loss.asscalar(), params[kq].grad(mx.cpu()).asnumpy()
        finally:
            set_native(prev)

    l_n, g_n = run(True)
    l_t, g_t = run(False)
    np.testing.assert_allclose(l_n, l_t, rtol=2e-4)
    np.testing.assert_allclose(g_n, g_t, rtol=2e-3, atol=2e-5)


def test_native_bert_adam_trains(native):
    """BERT + Trainer('adam') end-to-end on the native runtime: the
    fused adam_update registry op reduces the loss (VERDICT item 4
    breadth: second model family beyond the conv nets)."""
    from mxnet_amd.models.bert import BERTModel
    from mxnet_amd.ndarray import ops as F
    np.random.seed(0)
    net = BERTModel(vocab_size=100, units=32, hidden_size=64,
                    num_layers=2, num_heads=4, max_length=64, dropout=0.1)
    net.initialize(ctx=mx.cpu())
    tokens = mx.nd.array(np.random.randint(0, 100, (2, 8)), dtype='int64')
    mask = mx.nd.array(np.ones((2, 8)), dtype='float32')
    net(tokens, None, mask)
    tr = Trainer(net.collect_params(), 'adam', {'learning_rate': 1e-3})
    losses = []
    for _ in range(4):
        with autograd.record():
            _, _, mlm, _ = net(tokens, None, mask)
            logp = F.log_softmax(mlm.reshape(-1, 100))
            lab = mx.nd.array(np.zeros((16,)), dtype='float32')
            loss = (F.pick(logp, lab, axis=-1) * -1.0).mean()
        loss.backward()
        tr.step(2)
        losses.append(loss.asscalar())
    assert losses[-1] < losses[0]


def test_native_fused_adam_matches_reference(native):
    """multi_adam_update (one kernarg-batched launch for all params)
    must produce the same update as the per-parameter reference rule."""
    from mxnet_amd.gluon.parameter import Parameter
    from mxnet_amd.gluon import Trainer
    rs = np.random.RandomState(7)
    params = []
    refs = []
    for i, shape in enumerate([(5, 3), (7,), (2, 2, 2)]):
        w = rs.randn(*shape).astype('float32')
        g = rs.randn(*shape).astype('float32')
        p = Parameter(f'p{i}', shape=shape)
        p.initialize(ctx=mx.cpu())
        p.set_data(mx.nd.array(w))
        mx.nd.array(g).copyto(p.list_grad()[0])
        params.append(p)
        refs.append((w, g))
    tr = Trainer(params, 'adam', {'learning_rate': 0.01}, kvstore=None)
    tr._optimizer.rescale_grad = 1.0
    tr._update(False)
    mx.nd.waitall()
    b1, b2, eps, lr = 0.9, 0.999, 1e-8, 0.01
    lr_t = lr * np.sqrt(1 - b2) / (1 - b1)
    for p, (w, g) in zip(params, refs):
        m = (1 - b1) * g
        v = (1 - b2) * g * g
        exp = w - lr_t * m / (np.sqrt(v) + eps)
        np.testing.assert_allclose(p.data(mx.cpu()).asnumpy(), exp,
                                   rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize('mode,bidir', [('lstm', False), ('lstm', True),
                                        ('gru', False), ('rnn', True)])
def test_native_rnn_matches_torch_backend(mode, bidir):
    """LSTM/GRU/RNN on the native runtime (composed registry ops,
    recorded param slices, concat/flip via strided kernels) match the
    torch-backed path on identical params/input — fwd and full param
    gradient."""
    from mxnet_amd.gluon import rnn as _rnn

    def run(use_native):
        prev = set_native(use_native)
        try:
            cls = {'lstm': _rnn.LSTM, 'gru': _rnn.GRU, 'rnn': _rnn.RNN}[mode]
            net = cls(8, num_layers=2, bidirectional=bidir)
            net.initialize(ctx=mx.cpu())
            x = mx.nd.array(np.random.RandomState(1).randn(5, 3, 4),
                            dtype='float32')
            net(x)
            w = np.random.RandomState(2).randn(
                *net.parameters.shape).astype('float32') * 0.2
            net.parameters.set_data(mx.nd.array(w))
            with autograd.record():
                y = net(x)
                L = (y * y).sum()
            L.backward()
            return y.asnumpy(), net.parameters.grad(mx.cpu()).asnumpy()
        finally:
            set_native(prev)

    yn, gn = run(True)
    yt, gt = run(False)
    np.testing.assert_allclose(yn, yt, rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(gn, gt, rtol=1e-3, atol=1e-4)


def test_native_concat_stack(native):
    a = mx.nd.array(np.arange(6.).reshape(2, 3))
    b = mx.nd.array(np.arange(6., 12.).reshape(2, 3))
    from mxnet_amd.ndarray.ndarray import concat, stack
    np.testing.assert_array_equal(
        concat([a, b], dim=0).asnumpy(),
        np.arange(12.).reshape(4, 3))
    np.testing.assert_array_equal(
        concat([a, b], dim=1).asnumpy(),
        np.hstack([np.arange(6.).reshape(2, 3),
                   np.arange(6., 12.).reshape(2, 3)]))
    np.testing.assert_array_equal(
        stack([a, b], axis=0).asnumpy(),
        np.arange(12.).reshape(2, 2, 3))


@pytest.mark.parametrize('name,size', [
    ('resnet18_v1', 32), ('resnet18_v2', 32), ('vgg11', 32),
    ('alexnet', 64), ('squeezenet1_0', 64), ('densenet121', 32),
    ('mobilenet0_5', 32), ('mobilenet_v2_1_0', 32), ('inception_v3', 96),
    ('resnext50_32x4d', 32)])
def test_native_model_zoo_family(native, name, size):
    """Every model-zoo family runs forward+backward on the native
    runtime (the concat-based families ride the native concat op)."""
    from mxnet_amd.gluon.model_zoo import vision
    net = getattr(vision, name)(layout='NHWC', classes=10)
    net.initialize(ctx=mx.cpu())
    x = mx.nd.array(np.random.RandomState(0).randn(1, size, size, 3)
                    .astype('float32'))
    with autograd.record():
        y = net(x)
        L = (y * y).sum()
    L.backward()
    assert y.shape == (1, 10)
    v = L.asscalar()
    assert v == v


def test_native_losses_match_torch_backend():
    """Every composable loss produces identical values on the native
    runtime and the torch frontend (closed-form compositions over
    dual-backend nd ops; softplus/huber via relu+clip identities)."""
    from mxnet_amd.gluon import loss as gl
    rs = np.random.RandomState(0)
    p = rs.randn(4, 5).astype('float32')
    l = rs.randn(4, 5).astype('float32')
    lb = (rs.rand(4, 5) > 0.5).astype('float32')
    sign = np.sign(rs.randn(4, 5)).astype('float32')
    probs = np.abs(rs.rand(4, 5)).astype('float32')
    probs /= probs.sum(1, keepdims=True)
    cases = [
        (gl.L1Loss(), (p, l)), (gl.L2Loss(), (p, l)),
        (gl.HuberLoss(rho=0.7), (p, l)),
        (gl.HingeLoss(margin=1.0), (p, sign)),
        (gl.SquaredHingeLoss(margin=1.0), (p, sign)),
        (gl.LogisticLoss(), (p, sign)),
        (gl.SigmoidBinaryCrossEntropyLoss(), (p, lb)),
        (gl.KLDivLoss(from_logits=False), (p, probs)),
    ]
    for loss_fn, args in cases:
        prev = set_native(True)
        try:
            a_n = loss_fn(*[mx.nd.array(a) for a in args]).asnumpy()
        finally:
            set_native(prev)
        prev = set_native(False)
        try:
            a_t = loss_fn(*[mx.nd.array(a) for a in args]).asnumpy()
        finally:
            set_native(prev)
        np.testing.assert_allclose(a_n, a_t, rtol=1e-5, atol=1e-6,
                                   err_msg=type(loss_fn).__name__)


def test_native_norm_layers_and_prelu(native):
    """GroupNorm/InstanceNorm/PReLU compose from dual-backend nd ops on
    the native runtime and match the torch closed forms."""
    import torch
    from mxnet_amd.gluon import nn as gnn
    rs = np.random.RandomState(0)
    x = rs.randn(2, 8, 5, 5).astype('float32')
    gn = gnn.GroupNorm(num_groups=4)
    gn.initialize()
    np.testing.assert_allclose(
        gn(mx.nd.array(x)).asnumpy(),
        torch.nn.functional.group_norm(torch.tensor(x), 4, torch.ones(8),
                                       torch.zeros(8), 1e-5).numpy(),
        rtol=1e-4, atol=1e-5)
    inorm = gnn.InstanceNorm()
    inorm.initialize()
    np.testing.assert_allclose(
        inorm(mx.nd.array(x)).asnumpy(),
        torch.nn.functional.instance_norm(
            torch.tensor(x), weight=torch.ones(8), bias=torch.zeros(8),
            eps=1e-5).numpy(), rtol=1e-4, atol=1e-5)
    pr = gnn.PReLU(in_channels=8)
    pr.initialize()
    np.testing.assert_allclose(
        pr(mx.nd.array(x)).asnumpy(),
        torch.nn.functional.prelu(torch.tensor(x),
                                  torch.full((8,), 0.25)).numpy(),
        rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize('cellf', [
    lambda: __import__('mxnet_amd.gluon.rnn', fromlist=['x']).LSTMCell(6),
    lambda: __import__('mxnet_amd.gluon.rnn', fromlist=['x']).GRUCell(6),
    lambda: __import__('mxnet_amd.gluon.rnn', fromlist=['x']).RNNCell(6),
], ids=['lstm', 'gru', 'rnn'])
def test_rnn_cells_native_match_torch(cellf):
    """RNN cells compose dual-backend nd ops: unrolled outputs and
    gradients match between the native runtime and the torch frontend."""
    def run(use_native):
        prev = set_native(use_native)
        try:
            cell = cellf()
            cell.initialize()
            x = mx.nd.array(np.random.RandomState(1)
                            .randn(5, 3, 4).astype('float32'))
            cell.unroll(3, x, layout='NTC', merge_outputs=True)
            params = cell.collect_params()
            for k, p in params.items():
                rs = np.random.RandomState(sum(ord(c) for c in k) % 1000)
                p.set_data(mx.nd.array(
                    rs.randn(*p.shape).astype('float32') * 0.3))
            with autograd.record():
                outs, _ = cell.unroll(3, x, layout='NTC',
                                      merge_outputs=True)
                L = (outs * outs).sum()
            L.backward()
            k0 = sorted(params)[0]
            return outs.asnumpy(), params[k0].grad(mx.cpu()).asnumpy()
        finally:
            set_native(prev)
    yn, gn = run(True)
    yt, gt = run(False)
    np.testing.assert_allclose(yn, yt, rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(gn, gt, rtol=1e-3, atol=1e-4)


def test_native_estimator_dataloader_fit(native):
    """End-to-end Gluon train loop on the native runtime: DataLoader
    batchify collates onto native arrays, Estimator fit runs, loss
    drops."""
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    from mxnet_amd.gluon.contrib.estimator import Estimator
    from mxnet_amd.gluon.data import ArrayDataset, DataLoader
    rs = np.random.RandomState(0)
    X = rs.randn(64, 10).astype('float32')
    w_true = rs.randn(10, 3).astype('float32')
    y = np.argmax(X @ w_true, axis=1).astype('float32')
    net = nn.HybridSequential()
    net.add(nn.Dense(16, activation='relu'), nn.Dense(3))
    net.initialize()
    loader = DataLoader(ArrayDataset(X, y), batch_size=16)
    batch = next(iter(loader))
    assert batch[0].is_native
    tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.2},
                 kvstore=None)
    est = Estimator(net=net, loss=SoftmaxCrossEntropyLoss(), trainer=tr)
    est.fit(train_data=loader, epochs=12)
    out = net(mx.nd.array(X))
    acc = (out.asnumpy().argmax(1) == y).mean()
    assert acc > 0.6, acc


def test_native_op_batch(native):
    """Frontend batch that previously had no native branches:
    maximum/minimum/where/comparisons/dot/batch_dot/norm/argmax/argmin/
    tile/repeat/take/sort/topk all run on the native runtime and match
    numpy."""
    F = mx.nd.ops
    rs = np.random.RandomState(0)
    xa = rs.randn(3, 4).astype('float32')
    ya = rs.randn(3, 4).astype('float32')
    x, y = mx.nd.array(xa), mx.nd.array(ya)
    np.testing.assert_allclose(F.maximum(x, y).asnumpy(), np.maximum(xa, ya))
    np.testing.assert_allclose(F.minimum(x, 0.1).asnumpy(),
                               np.minimum(xa, 0.1), rtol=1e-6)
    np.testing.assert_allclose((x > y).asnumpy(),
                               (xa > ya).astype('float32'))
    cond = mx.nd.array((xa > 0).astype('float32'))
    np.testing.assert_allclose(F.where(cond, x, y).asnumpy(),
                               np.where(xa > 0, xa, ya))
    np.testing.assert_allclose(F.dot(x, y, transpose_b=True).asnumpy(),
                               xa @ ya.T, rtol=1e-5)
    a3 = rs.randn(2, 3, 4).astype('float32')
    b3 = rs.randn(2, 4, 5).astype('float32')
    np.testing.assert_allclose(
        F.batch_dot(mx.nd.array(a3), mx.nd.array(b3)).asnumpy(),
        a3 @ b3, rtol=1e-5)
    np.testing.assert_allclose(x.norm().asnumpy(), [np.linalg.norm(xa)],
                               rtol=1e-5)
    np.testing.assert_allclose(x.argmax(axis=1).asnumpy(), xa.argmax(1))
    np.testing.assert_allclose(F.tile(x, (2, 3)).asnumpy(),
                               np.tile(xa, (2, 3)))
    np.testing.assert_allclose(F.repeat(x, 2, axis=0).asnumpy(),
                               np.repeat(xa, 2, 0))
    w = rs.randn(6, 5).astype('float32')
    idx = np.array([[0, 2], [5, 1]], 'float32')
    np.testing.assert_allclose(
        F.take(mx.nd.array(w), mx.nd.array(idx)).asnumpy(),
        w[idx.astype(int)])
    np.testing.assert_allclose(F.sort(x).asnumpy(), np.sort(xa, -1))
    v, i = F.topk(x, k=2, ret_typ='both')
    np.testing.assert_allclose(v.asnumpy(), -np.sort(-xa, -1)[:, :2])


def test_native_export_symbolblock_roundtrip(native):
    """export() -> SymbolBlock.imports -> native inference: the symbol
    interpreter executes through the dual-backend nd ops, so the
    checkpoint pair serves the native runtime too."""
    import os
    import tempfile
    from mxnet_amd.gluon import SymbolBlock
    net = nn.HybridSequential()
    net.add(nn.Dense(8, activation='relu'), nn.Dense(4))
    net.initialize()
    x = mx.nd.array(np.random.RandomState(0).randn(2, 6).astype('float32'))
    ref = net(x).asnumpy()
    d = tempfile.mkdtemp()
    path = os.path.join(d, 'm')
    net.export(path)
    sb = SymbolBlock.imports(path + '-symbol.json', ['data'],
                             path + '-0000.params')
    out = sb(x)
    assert out.is_native
    np.testing.assert_allclose(out.asnumpy(), ref, rtol=1e-5, atol=1e-6)


def test_native_trainer_state_roundtrip(native):
    """Trainer.save_states/load_states on the native runtime: momentum
    buffers pickle as tagged numpy, reload onto the right context, and
    training continues (checkpoint/resume parity)."""
    import tempfile
    net = nn.Dense(4)
    net.initialize()
    x = mx.nd.array(np.random.RandomState(0).randn(3, 5).astype('float32'))
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.1, 'momentum': 0.9}, kvstore=None)
    with autograd.record():
        L = (net(x) ** 2).sum()
    L.backward()
    tr.step(1)
    s0 = tr._states[0]
    mom = (s0[1] if isinstance(s0, tuple) else s0).asnumpy()
    f = tempfile.mktemp()
    tr.save_states(f)
    tr.load_states(f)
    s0 = tr._states[0]
    np.testing.assert_allclose(
        (s0[1] if isinstance(s0, tuple) else s0).asnumpy(), mom)
    with autograd.record():
        L = (net(x) ** 2).sum()
    L.backward()
    tr.step(1)  # training continues on restored state


def test_native_pickle_and_mp_dataloader(native):
    """Native NDArrays pickle (numpy round-trip, same backend/context),
    so the multiprocessing DataLoader ships native batches across the
    worker boundary."""
    import pickle
    x = mx.nd.array(np.arange(6.).reshape(2, 3))
    y = pickle.loads(pickle.dumps(x))
    assert y.is_native
    np.testing.assert_array_equal(y.asnumpy(), x.asnumpy())
    from mxnet_amd.gluon.data import ArrayDataset, DataLoader
    rs = np.random.RandomState(0)
    X = rs.randn(32, 6).astype('float32')
    yl = rs.randn(32).astype('float32')
    loader = DataLoader(ArrayDataset(X, yl), batch_size=8, num_workers=2)
    tot = 0
    for bx, by in loader:
        assert bx.is_native
        tot += bx.shape[0]
    assert tot == 32


def test_native_random_creation(native):
    """nd.random uniform/normal run the native philox kernels with
    correct moments and bounds."""
    u = mx.nd.random.uniform(low=2.0, high=5.0, shape=(2000,))
    assert u.is_native
    a = u.asnumpy()
    assert a.min() >= 2.0 and a.max() <= 5.0
    assert abs(a.mean() - 3.5) < 0.2
    n = mx.nd.random.normal(loc=1.0, scale=2.0, shape=(20000,))
    b = n.asnumpy()
    assert abs(b.mean() - 1.0) < 0.1 and abs(b.std() - 2.0) < 0.1


def test_native_unary_binary_families(native):
    """The generated unary/binary op families route natively: registry
    kernels, composed expressions, or documented host fallbacks."""
    F = mx.nd.ops
    x = mx.nd.array(np.array([0.2, 0.5, 0.9], np.float32))
    np.testing.assert_allclose(F.log1p(x).asnumpy(),
                               np.log1p([0.2, 0.5, 0.9]), rtol=1e-6)
    np.testing.assert_allclose(F.rsqrt(x).asnumpy(),
                               1 / np.sqrt([0.2, 0.5, 0.9]), rtol=1e-5)
    np.testing.assert_allclose(F.sin(x).asnumpy(),
                               np.sin([0.2, 0.5, 0.9]), rtol=1e-6)
    np.testing.assert_allclose(
        F.sign(mx.nd.array(np.array([-2., 0., 3.]))).asnumpy(), [-1, 0, 1])
    np.testing.assert_allclose(
        F.broadcast_mul(x, mx.nd.array(np.array([2.0], np.float32)))
        .asnumpy(), [0.4, 1.0, 1.8], rtol=1e-6)
    np.testing.assert_allclose(
        F.logical_not(mx.nd.array(np.array([0., 1., 2.]))).asnumpy(),
        [1, 0, 0])
    np.testing.assert_allclose(F.reciprocal(x).asnumpy(),
                               1 / np.array([0.2, 0.5, 0.9]), rtol=1e-6)
    import scipy.special as sp
    np.testing.assert_allclose(F.erf(x).asnumpy(),
                               sp.erf([0.2, 0.5, 0.9]), rtol=1e-5)


def test_native_reduce_and_tail_ops(native):
    """The fn-form reductions and tail ops route natively (sum/mean/
    max/min via the reduce kernels; host fallbacks numerically exact)."""
    F = mx.nd.ops
    rs = np.random.RandomState(0)
    xa = rs.rand(4, 6).astype('float32') + 0.1
    ya = rs.rand(4, 6).astype('float32') + 0.1
    x, y = mx.nd.array(xa), mx.nd.array(ya)
    np.testing.assert_allclose(F.sum(x, axis=1).asnumpy(), xa.sum(1),
                               rtol=1e-5)
    np.testing.assert_allclose(F.mean(x, axis=0).asnumpy(), xa.mean(0),
                               rtol=1e-5)
    np.testing.assert_allclose(F.max(x, axis=1).asnumpy(), xa.max(1))
    np.testing.assert_allclose(F.cumsum(x, axis=1).asnumpy(),
                               np.cumsum(xa, 1), rtol=1e-5)
    m, v = F.moments(x, axes=(0,))
    np.testing.assert_allclose(v.asnumpy(), xa.var(0), rtol=1e-4)
    np.testing.assert_allclose(F.hypot(x, y).asnumpy(), np.hypot(xa, ya),
                               rtol=1e-5)
    np.testing.assert_allclose(
        F.smooth_l1(x, 1.0).asnumpy(),
        np.where(np.abs(xa) < 1, 0.5 * xa * xa, np.abs(xa) - 0.5),
        rtol=1e-5)
    np.testing.assert_allclose(F.add_n(x, y).asnumpy(), xa + ya,
                               rtol=1e-6)
    np.testing.assert_allclose(F.broadcast_greater(x, y).asnumpy(),
                               (xa > ya).astype('f'))
    np.testing.assert_allclose(F.swapaxes(x, 0, 1).asnumpy(), xa.T)
    np.testing.assert_allclose(
        F.linalg_gemm2(x, y, transpose_b=True).asnumpy(), xa @ ya.T,
        rtol=1e-4)


def test_native_op_surface_fuzz(native):
    """Every public single/two-array op either runs natively or raises
    a clean error — never a crash (incl. empty/int64/zero-dim inputs);
    the torch-only tail must not grow."""
    import inspect
    F = mx.nd.ops
    x = mx.nd.array(np.random.RandomState(0).rand(4, 6)
                    .astype('float32') + 0.1)
    y = mx.nd.array(np.random.RandomState(1).rand(4, 6)
                    .astype('float32') + 0.1)
    # degenerate inputs must never crash the process
    for adv in [mx.nd.array(np.zeros((0,), 'float32')),
                mx.nd.array(np.arange(6).astype('int64')),
                mx.nd.array(np.zeros((3, 0), 'float32'))]:
        for name in sorted(dir(F)):
            if name.startswith('_'):
                continue
            fn = getattr(F, name)
            if not callable(fn):
                continue
            try:
                sig = inspect.signature(fn)
                nr = len([p for p in sig.parameters.values()
                          if p.default is p.empty and p.kind not in
                          (p.VAR_POSITIONAL, p.VAR_KEYWORD)])
            except (ValueError, TypeError):
                continue
            try:
                r = fn(adv) if nr <= 1 else \
                    (fn(adv, adv) if nr == 2 else None)
                if hasattr(r, 'asnumpy'):
                    r.asnumpy()
            except Exception:
                pass
    allowed_torch_only = {
        'BilinearResize2D', 'BilinearSampler', 'GridGenerator', 'LRN',
        'SequenceLast', 'SequenceMask', 'SequenceReverse',
        'SpatialTransformer', 'UpSampling', 'broadcast_mod',
        'depth_to_space', 'space_to_depth', 'ravel_multi_index',
        'unravel_index', 'sequence_mask'}
    missing = []
    for name in sorted(dir(F)):
        if name.startswith('_'):
            continue
        fn = getattr(F, name)
        if not callable(fn):
            continue
        try:
            sig = inspect.signature(fn)
            nreq = len([p for p in sig.parameters.values()
                        if p.default is p.empty and p.kind not in
                        (p.VAR_POSITIONAL, p.VAR_KEYWORD)])
        except (ValueError, TypeError):
            continue
        try:
            r = fn(x) if nreq <= 1 else (fn(x, y) if nreq == 2 else None)
            if hasattr(r, 'asnumpy'):
                r.asnumpy()
        except RuntimeError as e:
            if 'native' in str(e) or 'torch frontend' in str(e):
                missing.append(name)
        except Exception:
            pass  # arg-shape errors etc. are fine
    unexpected = set(missing) - allowed_torch_only
    assert not unexpected, f'ops lost their native path: {unexpected}'
    # drain the engine's global-exception slot: worker-side failures
    # triggered intentionally above must not leak into later tests
    from mxnet_amd import _core
    for _ in range(50):
        try:
            _core.wait_all()
            break
        except RuntimeError:
            continue


def test_native_layer_sweep(native):
    """Every basic Gluon layer runs forward+backward on the native
    runtime (fuzz-style sweep with plausible inputs)."""
    rs = np.random.RandomState(0)
    x4 = mx.nd.array(rs.randn(2, 8, 8, 3).astype('float32'))
    x2 = mx.nd.array(rs.randn(2, 12).astype('float32'))
    xc = mx.nd.array(rs.randn(2, 4, 5, 5).astype('float32'))
    cases = [
        (nn.Dense(5), x2), (nn.Dense(5, flatten=True), x4),
        (nn.Dropout(0.3), x2), (nn.BatchNorm(axis=-1), x4),
        (nn.LayerNorm(), x2), (nn.GroupNorm(num_groups=1), xc),
        (nn.InstanceNorm(), xc), (nn.Flatten(), x4),
        (nn.Activation('relu'), x2), (nn.LeakyReLU(0.1), x2),
        (nn.PReLU(), x2), (nn.ELU(), x2), (nn.SELU(), x2),
        (nn.Swish(), x2), (nn.GELU(), x2),
        (nn.Conv2D(4, 3, padding=1, layout='NHWC'), x4),
        (nn.MaxPool2D(2, layout='NHWC'), x4),
        (nn.AvgPool2D(2, layout='NHWC'), x4),
        (nn.GlobalAvgPool2D(layout='NHWC'), x4),
        (nn.Embedding(10, 6),
         mx.nd.array(rs.randint(0, 10, (2, 3)), dtype='int64')),
    ]
    for layer, inp in cases:
        layer.initialize()
        with autograd.record():
            y = layer(inp)
            L = (y * y).sum()
        L.backward()
        v = L.asscalar()
        assert v == v, type(layer).__name__


def test_native_sparse_grad_embedding_dense_fallback(native):
    """sparse_grad=True embeddings train natively via correct dense
    grads (row-sparse laziness is a torch-frontend optimization)."""
    net = nn.Embedding(20, 4, sparse_grad=True)
    net.initialize()
    tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.5},
                 kvstore=None)
    idx = mx.nd.array(np.array([1, 3, 3]), dtype='int64')
    w0 = net.weight.data(mx.cpu()).asnumpy().copy()
    with autograd.record():
        e = net(idx)
        L = (e * e).sum()
    L.backward()
    tr.step(1)
    w1 = net.weight.data(mx.cpu()).asnumpy()
    assert not np.allclose(w1[1], w0[1])
    np.testing.assert_array_equal(w1[0], w0[0])


def test_native_error_paths(native):
    """Misuse raises clean errors (rank checks, layout asserts, clean
    torch-frontend pointers) — never UB."""
    x2 = mx.nd.array(np.ones((4, 6), 'float32'))
    with pytest.raises(RuntimeError, match='3-D'):
        mx.nd.ops.batch_dot(x2, x2).asnumpy()
    with pytest.raises(AssertionError, match='NHWC'):
        net = nn.Conv2D(4, 3, layout='NCHW')
        net.initialize()
        net(mx.nd.array(np.ones((1, 3, 6, 6), 'float32')))
    with pytest.raises(NotImplementedError, match='torch frontend'):
        net = nn.Conv3D(4, 3)
        net.initialize()
        net(mx.nd.array(np.ones((1, 2, 4, 4, 4), 'float32')))
    with pytest.raises(IndexError):
        x2[99]
    with pytest.raises(RuntimeError, match='incompatible|broadcast'):
        (x2 + mx.nd.array(np.ones((5, 7), 'float32'))).asnumpy()
