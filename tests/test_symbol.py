"""Symbol tracing / JSON schema / SymbolBlock import tests."""
import json

import numpy as np

import mxnet_amd as mx
from mxnet_amd import nd, gluon
from mxnet_amd.gluon import nn


def _mlp():
    net = nn.HybridSequential()
    net.add(nn.Dense(8, activation='relu', in_units=4),
            nn.Dense(3, in_units=8))
    return net


def test_symbol_trace_and_json():
    net = _mlp()
    net.initialize()
    data = mx.sym.var('data')
    out = net(data)
    j = json.loads(out.tojson())
    assert set(j) >= {'nodes', 'arg_nodes', 'node_row_ptr', 'heads', 'attrs'}
    ops = [n['op'] for n in j['nodes']]
    assert 'FullyConnected' in ops
    assert 'Activation' in ops
    # all null-op nodes are inputs (arg_nodes)
    for i in j['arg_nodes']:
        assert j['nodes'][i]['op'] == 'null'
    # attrs stringified like dmlc::Parameter ("8", "True"...)
    fc = next(n for n in j['nodes'] if n['op'] == 'FullyConnected')
    assert fc['attrs']['num_hidden'] in ('8', '3')
    for inp in fc['inputs']:
        assert len(inp) == 3


def test_export_import_roundtrip(tmp_path):
    net = _mlp()
    net.initialize()
    x = nd.random_uniform(shape=(2, 4))
    y1 = net(x)
    path = str(tmp_path / 'model')
    net.export(path)
    blk = gluon.SymbolBlock.imports(path + '-symbol.json', ['data'],
                                    path + '-0000.params')
    y2 = blk(x)
    assert np.allclose(y1.asnumpy(), y2.asnumpy(), atol=1e-6)


def test_export_conv_model(tmp_path):
    net = nn.HybridSequential()
    net.add(nn.Conv2D(4, 3, padding=1, in_channels=2),
            nn.Activation('relu'),
            nn.MaxPool2D(2, 2),
            nn.Flatten(),
            nn.Dense(5))
    net.initialize()
    x = nd.random_uniform(shape=(2, 2, 8, 8))
    y1 = net(x)
    path = str(tmp_path / 'conv')
    net.export(path)
    blk = gluon.SymbolBlock.imports(path + '-symbol.json', ['data'],
                                    path + '-0000.params')
    y2 = blk(x)
    assert np.allclose(y1.asnumpy(), y2.asnumpy(), atol=1e-5)


def test_symbol_arith_eval():
    a = mx.sym.var('a')
    b = mx.sym.var('b')
    c = (a + b) * 2
    out = c.eval_dict({'a': nd.ones((2,)), 'b': nd.ones((2,))})
    assert np.allclose(out[0].asnumpy(), 4)


def test_list_arguments_order():
    net = _mlp()
    net.initialize()
    out = net(mx.sym.var('data'))
    args = out.list_arguments()
    assert args[0] == 'data'
    assert any('weight' in a for a in args)


def test_subgraph_partition():
    """Subgraph framework (reference build_subgraph.cc): pointwise chains
    collapse into a _fused_subgraph node carrying the sub-symbol."""
    import json
    from mxnet_amd import symbol as S
    from mxnet_amd.symbol.subgraph import partition_graph
    x = S.var('data')
    y = S.Activation(S.FullyConnected(x, num_hidden=8, name='fc'),
                     act_type='relu')
    y = S.tanh(y + 1.0)
    out = S.FullyConnected(y, num_hidden=4, name='fc2')
    p = partition_graph(out)
    conf = json.loads(p.tojson())
    ops = [n['op'] for n in conf['nodes'] if n['op'] != 'null']
    assert ops.count('_fused_subgraph') == 1
    fused = [n for n in conf['nodes'] if n['op'] == '_fused_subgraph'][0]
    assert fused['attrs']['ops'] == 'Activation,_plus_scalar,tanh'


def test_rtc_pointwise_fusion_pipeline():
    """partition_graph -> _fused_subgraph -> codegen + interpreted eval
    (reference pointwise_fusion_pass.cc + fused_op.cu; hiprtc here)."""
    import json
    import torch
    import mxnet_amd as mx
    from mxnet_amd import symbol as S
    from mxnet_amd.symbol.subgraph import partition_graph
    from mxnet_amd.contrib.fusion import codegen

    x = S.var('x')
    y = S.var('y')
    z = S.Activation(x * 2.0 + y, act_type='relu') * 0.5
    ps = partition_graph(z)
    d = json.loads(ps.tojson())
    fused = [n for n in d['nodes'] if n['op'] == '_fused_subgraph']
    assert len(fused) == 1
    sub = fused[0]['attrs']['subgraph']
    src, n_in = codegen(sub)
    assert n_in == 2 and '__global__' in src and 'fmaxf' in src
    # the generated source must compile under hiprtc
    mx.rtc.HipModule(src)
    mx.rtc.HipModule(codegen(sub, 'float16')[0])
    # interpreted execution matches eager composition
    xt, yt = torch.randn(4, 5), torch.randn(4, 5)
    out = ps.eval(x=mx.nd.from_torch(xt), y=mx.nd.from_torch(yt))[0]
    ref = torch.relu(xt * 2 + yt) * 0.5
    assert torch.allclose(out.handle, ref, atol=1e-6)
