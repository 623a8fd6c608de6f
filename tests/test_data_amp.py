"""Data pipeline + AMP coverage (reference tests/python/unittest/
test_gluon_data.py, test_amp.py shapes)."""
import numpy as np
import pytest
import torch

import mxnet_amd as mx
from mxnet_amd.gluon import data as gdata
from mxnet_amd.gluon.data.vision import transforms


def test_array_dataset_and_loader():
    X = np.random.randn(37, 4).astype(np.float32)
    Y = np.arange(37).astype(np.float32)
    ds = gdata.ArrayDataset(X, Y)
    assert len(ds) == 37
    loader = gdata.DataLoader(ds, batch_size=8, shuffle=True,
                              last_batch='keep')
    seen = 0
    for xb, yb in loader:
        assert xb.shape[1] == 4
        seen += xb.shape[0]
    assert seen == 37


def test_dataloader_multiworker():
    ds = gdata.ArrayDataset(np.arange(64, dtype=np.float32).reshape(32, 2))
    loader = gdata.DataLoader(ds, batch_size=4, num_workers=2)
    total = sum(b.shape[0] for b in loader)
    assert total == 32


def test_transforms_pipeline():
    tr = transforms.Compose([transforms.ToTensor(),
                             transforms.Normalize(0.5, 0.5)])
    img = mx.nd.array(np.random.rand(8, 8, 3).astype(np.float32))
    out = tr(img)
    assert out.shape == (3, 8, 8)
    assert abs(float(out.handle.mean())) < 2.0


def test_synthetic_dataset_with_transform():
    from mxnet_amd.gluon.data.vision.datasets import SyntheticImageDataset
    ds = SyntheticImageDataset(length=16, shape=(8, 8, 3), num_classes=4)
    ds2 = ds.transform_first(transforms.ToTensor())
    x, y = ds2[0]
    assert x.shape == (3, 8, 8) and 0 <= int(y) < 4


def test_sampler_batchify():
    from mxnet_amd.gluon.data import sampler as smp
    s = list(smp.SequentialSampler(5))
    assert s == [0, 1, 2, 3, 4]
    r = list(smp.RandomSampler(5))
    assert sorted(r) == [0, 1, 2, 3, 4]
    from mxnet_amd.gluon.data.batchify import Stack
    out = Stack()([np.ones((2, 2)), np.zeros((2, 2))])
    assert out.shape == (2, 2, 2)


def test_amp_loss_scaler_overflow_skip():
    from mxnet_amd import amp
    from mxnet_amd.gluon import nn, Trainer
    from mxnet_amd import autograd
    from mxnet_amd.ndarray.ndarray import NDArray
    net = nn.Dense(2, in_units=3)
    net.initialize()
    tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.1},
                 kvstore=None)
    amp.init()
    amp.init_trainer(tr)
    x = mx.nd.array(torch.randn(4, 3))
    w_before = net.weight.data().asnumpy().copy()
    with autograd.record():
        out = net(x)
        loss = NDArray(out.handle.mean())
    with amp.scale_loss(loss, tr) as scaled:
        scaled.backward()
    # poison a gradient: all_finite must veto the step
    net.weight.grad().handle[0, 0] = float('inf')
    grads = [p.grad() for p in tr._params]
    assert not amp.all_finite(grads)
    scaler = tr._amp_loss_scaler
    before = scaler.loss_scale
    scaler.update_scale(True)   # overflow -> halve scale
    assert scaler.loss_scale < before
    after = scaler.loss_scale
    scaler.update_scale(False)
    assert scaler.loss_scale >= after


def test_recordfile_dataset(tmp_path):
    from mxnet_amd.io.recordio import MXIndexedRecordIO
    path = str(tmp_path / 'x')
    w = MXIndexedRecordIO(path + '.idx', path + '.rec', 'w')
    for i in range(5):
        w.write_idx(i, bytes([i] * 4))
    w.close()
    ds = gdata.RecordFileDataset(path + '.rec')
    assert len(ds) == 5
    assert ds[3] == bytes([3] * 4)


def test_threaded_dataloader_native():
    """C++ ThreadedBatcher-backed loader: ordered epochs, shuffled
    coverage, multi-array batches (reference io/dataloader.cc)."""
    import numpy as np
    from mxnet_amd.gluon.data import ThreadedDataLoader
    X = np.arange(100 * 4, dtype=np.float32).reshape(100, 4)
    Y = np.arange(100, dtype=np.int64)
    dl = ThreadedDataLoader((X, Y), batch_size=16, num_workers=3)
    assert len(dl) == 7
    seen = []
    for xb, yb in dl:
        seen.extend(yb.handle.tolist())
        np.testing.assert_allclose(xb.asnumpy(), X[yb.handle.numpy()])
    assert seen == list(range(100))
    dl2 = ThreadedDataLoader((X, Y), batch_size=32, shuffle=True,
                             num_workers=4, last_batch='discard')
    tot = sorted(sum((yb.handle.tolist() for _, yb in dl2), []))
    assert len(tot) == 96 and len(set(tot)) == 96
    # two epochs over the same loader work (fresh batcher per epoch)
    n2 = sum(1 for _ in dl2)
    assert n2 == 3


def test_native_all_finite():
    """AMP finiteness check on the native runtime (multi_all_finite
    registry op; loss-scaler overflow detection path)."""
    import numpy as np
    import mxnet_amd as mx
    from mxnet_amd.base import set_native
    from mxnet_amd.amp import all_finite
    prev = set_native(True)
    try:
        a = mx.nd.array(np.ones((4, 4)))
        assert all_finite([a]) is True
        assert all_finite([a, mx.nd.array(np.array([1.0, np.inf]))]) is False
        assert all_finite([mx.nd.array(np.array([np.nan]))]) is False
    finally:
        set_native(prev)


def test_native_amp_loop_and_overflow_skip():
    """AMP dynamic loss scaling on the native runtime: scale_loss,
    fused finiteness check, overflow skips the update and shrinks the
    scale (round-1 ADVICE high-severity behavior, native path)."""
    import numpy as np
    import mxnet_amd as mx
    from mxnet_amd import autograd, amp
    from mxnet_amd.base import set_native
    from mxnet_amd.gluon import nn, Trainer
    prev = set_native(True)
    try:
        net = nn.Dense(4)
        net.initialize()
        tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.1},
                     kvstore=None)
        amp.init_trainer(tr)
        x = mx.nd.array(np.random.RandomState(0).randn(3, 5)
                        .astype('float32'))
        for _ in range(2):
            with autograd.record():
                out = net(x)
                L = (out * out).sum()
                with amp.scale_loss(L, tr) as scaled:
                    pass
            scaled.backward()
            tr.step(1)
        w = net.weight.data(mx.cpu()).asnumpy().copy()
        s0 = tr._amp_loss_scaler.loss_scale
        g = net.weight.list_grad()[0]
        mx.nd.array(np.full(g.shape, np.inf, 'float32')).copyto(g)
        tr._update(False)
        np.testing.assert_array_equal(
            net.weight.data(mx.cpu()).asnumpy(), w)
        assert tr._amp_loss_scaler.loss_scale < s0
    finally:
        set_native(prev)
