"""ONNX export/import round-trips (reference python/mxnet/contrib/onnx
mx2onnx + onnx2mx; serialization here is the hand-rolled protobuf wire
codec in mxnet_amd/contrib/onnx/_proto.py)."""
import numpy as np
import torch

import mxnet_amd as mx
from mxnet_amd.gluon import nn
from mxnet_amd.contrib import onnx as mxonnx


def _convnet():
    torch.manual_seed(0)
    net = nn.HybridSequential()
    net.add(nn.Conv2D(8, 3, padding=1, in_channels=3),
            nn.BatchNorm(in_channels=8),
            nn.Activation('relu'),
            nn.MaxPool2D(2),
            nn.Flatten(),
            nn.Dense(10, in_units=8 * 4 * 4))
    net.initialize()
    net.hybridize()
    return net


def test_onnx_roundtrip_convnet(tmp_path):
    net = _convnet()
    x = mx.nd.from_torch(torch.randn(2, 3, 8, 8))
    y0 = net(x)
    net.export(str(tmp_path / 'm'))
    params = mx.nd.load(str(tmp_path / 'm-0000.params'))
    buf = mxonnx.export_model(str(tmp_path / 'm-symbol.json'), params,
                              [(2, 3, 8, 8)],
                              onnx_file=str(tmp_path / 'm.onnx'))
    assert len(buf) > 1000
    net2 = mxonnx.import_to_gluon(str(tmp_path / 'm.onnx'))
    y1 = net2(x)
    np.testing.assert_allclose(y1.asnumpy(), y0.asnumpy(),
                               rtol=1e-4, atol=1e-5)


def test_onnx_roundtrip_mlp_softmax(tmp_path):
    torch.manual_seed(1)
    net = nn.HybridSequential()
    net.add(nn.Dense(32, in_units=16, activation='tanh'),
            nn.Dense(8, in_units=32))
    net.initialize()
    net.hybridize()
    x = mx.nd.from_torch(torch.randn(4, 16))
    y0 = mx.nd.softmax(net(x), axis=-1)
    net.export(str(tmp_path / 'mlp'))
    params = mx.nd.load(str(tmp_path / 'mlp-0000.params'))
    buf = mxonnx.export_model(str(tmp_path / 'mlp-symbol.json'), params,
                              [(4, 16)])
    sym, args, auxs = mxonnx.import_model(buf)
    net2 = mxonnx.import_to_gluon(buf)
    y1 = mx.nd.softmax(net2(x), axis=-1)
    np.testing.assert_allclose(y1.asnumpy(), y0.asnumpy(),
                               rtol=1e-4, atol=1e-6)


def test_onnx_proto_structure(tmp_path):
    """The emitted bytes are a structurally valid ModelProto: parseable,
    graph field present, initializers carry raw tensor data that decodes
    to the original values."""
    from mxnet_amd.contrib.onnx import _proto as P
    net = _convnet()
    net(mx.nd.from_torch(torch.randn(1, 3, 8, 8)))
    net.export(str(tmp_path / 'p'))
    params = mx.nd.load(str(tmp_path / 'p-0000.params'))
    buf = mxonnx.export_model(str(tmp_path / 'p-symbol.json'), params,
                              [(1, 3, 8, 8)])
    model = P.parse(buf)
    assert P.as_int(model, 1) == 8            # ir_version
    assert P.as_str(model, 2) == 'mxnet_amd'  # producer
    graph = P.parse(P.as_bytes(model, 7))
    assert len(graph.get(1, [])) >= 6         # nodes
    names = set()
    for raw in graph.get(5, []):              # initializers
        f = P.parse(raw)
        names.add(P.as_str(f, 8))
        assert len(P.as_bytes(f, 9)) > 0
    assert '0.weight' in names and '1.gamma' in names


def test_onnx_roundtrip_resnet18(tmp_path):
    """Model-zoo ResNet-18 (fused BN+residual+relu hot path) exports and
    re-imports exactly — the residual keyword input becomes an Add node."""
    from mxnet_amd.gluon.model_zoo import vision
    torch.manual_seed(0)
    net = vision.resnet18_v1(classes=10)
    net.initialize()
    net.hybridize()
    x = mx.nd.from_torch(torch.randn(1, 3, 32, 32))
    y0 = net(x).asnumpy()
    net.export(str(tmp_path / 'r18'))
    params = mx.nd.load(str(tmp_path / 'r18-0000.params'))
    buf = mxonnx.export_model(str(tmp_path / 'r18-symbol.json'), params,
                              [(1, 3, 32, 32)])
    net2 = mxonnx.import_to_gluon(buf)
    np.testing.assert_allclose(net2(x).asnumpy(), y0, rtol=1e-4, atol=1e-4)


def test_onnx_embedding_gather(tmp_path):
    """Embedding exports as Gather and re-imports (reference mx2onnx
    embedding mapping)."""
    from mxnet_amd.gluon import nn as gnn
    from mxnet_amd import symbol as S
    torch.manual_seed(2)
    emb = gnn.Embedding(20, 8)
    dense = gnn.Dense(4, in_units=8, flatten=False)
    emb.initialize()
    dense.initialize()
    from mxnet_amd.gluon import nn
    net = nn.HybridSequential()
    net.add(emb, dense)
    net.hybridize()
    x = mx.nd.from_torch(torch.randint(0, 20, (2, 5)))
    y0 = net(x).asnumpy()
    net.export(str(tmp_path / 'e'))
    params = mx.nd.load(str(tmp_path / 'e-0000.params'))
    buf = mxonnx.export_model(str(tmp_path / 'e-symbol.json'), params,
                              [(2, 5)], in_types=[torch.int64])
    net2 = mxonnx.import_to_gluon(buf)
    np.testing.assert_allclose(net2(x).asnumpy(), y0, rtol=1e-4, atol=1e-5)


def test_onnx_export_native_params():
    """ONNX export accepts native-runtime parameter arrays (numpy
    bridge) — a native-trained model exports and re-imports."""
    import os
    import tempfile
    import numpy as np
    import mxnet_amd as mx
    from mxnet_amd.base import set_native
    from mxnet_amd.gluon import nn
    from mxnet_amd.contrib import onnx as onnx_mod
    prev = set_native(True)
    try:
        net = nn.HybridSequential()
        net.add(nn.Dense(8, activation='relu'), nn.Dense(4))
        net.initialize()
        x = mx.nd.array(np.random.RandomState(0).randn(2, 6)
                        .astype('float32'))
        net(x)
        d = tempfile.mkdtemp()
        pre = os.path.join(d, 'm')
        net.export(pre)
        params = mx.nd.load(pre + '-0000.params')
        assert next(iter(params.values())).is_native
        onnx_mod.export_model(pre + '-symbol.json', params, [(2, 6)],
                              onnx_file=os.path.join(d, 'm.onnx'))
        sym, arg, aux = onnx_mod.import_model(os.path.join(d, 'm.onnx'))
        assert len(arg) == 4
    finally:
        set_native(prev)
