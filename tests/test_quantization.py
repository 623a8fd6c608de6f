"""int8 quantization (reference contrib/quantization.py +
quantize_graph_pass.cc): calibration modes, layer swapping, accuracy of
the quantized net against the fp32 original."""
import numpy as np
import pytest

import mxnet_amd as mx
from mxnet_amd.contrib.quantization import (
    quantize, dequantize, calib_minmax, calib_entropy, quantize_net)
from mxnet_amd.gluon import nn


def test_quantize_roundtrip():
    x = mx.nd.array(np.random.RandomState(0).randn(64, 32) * 3)
    q, scale = quantize(x)
    back = dequantize(q, scale)
    err = np.abs(back.asnumpy() - x.asnumpy()).max()
    assert err <= scale * 0.5 + 1e-6


def test_calib_entropy_clips_outliers():
    """Entropy calibration should pick a threshold well below a lone
    extreme outlier (minmax would waste the int8 range on it)."""
    rs = np.random.RandomState(1)
    vals = rs.randn(20000).astype('float32')
    vals[0] = 1000.0
    x = mx.nd.array(vals)
    s_mm = calib_minmax(x)
    s_kl = calib_entropy(x)
    assert s_kl < s_mm / 10


def _mlp():
    net = nn.HybridSequential()
    net.add(nn.Dense(32, activation='relu'), nn.Dense(10))
    net.initialize()
    # realistic weight magnitudes (fresh init is near zero, which makes
    # relative-error checks meaningless)
    rs = np.random.RandomState(42)
    x = mx.nd.array(rs.randn(2, 20).astype('float32'))
    net(x)
    for k, p in net.collect_params().items():
        p.set_data(mx.nd.array(
            rs.randn(*p.shape).astype('float32') * 0.3))
    return net


def test_quantize_net_swaps_and_matches():
    np.random.seed(0)
    net = _mlp()
    x = mx.nd.array(np.random.randn(16, 20).astype('float32'))
    ref = net(x).asnumpy()
    swapped = quantize_net(net, calib_data=[x], calib_mode='naive')
    assert len(swapped) == 2
    out = net(x).asnumpy()  # layers were replaced in the tree
    # int8 inference tracks fp32 within quantization noise
    denom = np.abs(ref).max() or 1.0
    assert np.abs(out - ref).max() / denom < 0.06
    # static scales were fixed by calibration
    for _, _, q in swapped:
        assert q._x_scale is not None


def test_quantize_net_entropy_mode():
    np.random.seed(1)
    net = _mlp()
    batches = [mx.nd.array(np.random.randn(8, 20).astype('float32'))
               for _ in range(3)]
    quantize_net(net, calib_data=batches, calib_mode='entropy')
    out = net(batches[0])
    assert np.isfinite(out.asnumpy()).all()
