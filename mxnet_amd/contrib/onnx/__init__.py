"""ONNX export / import (reference python/mxnet/contrib/onnx: mx2onnx
export_model + onnx2mx import_model).

The reference delegated serialization to the ``onnx`` python package;
this implementation writes/reads the protobuf wire format directly
(``_proto``) since the package is not available offline.  The graph
mapping operates on the same artifacts the reference used: the symbol
JSON produced by ``HybridBlock.export`` / ``Symbol.tojson`` plus the
parameter dict.

Supported op subset (NCHW graphs): Convolution, FullyConnected,
BatchNorm(+fused relu), Activation, Pooling (incl. global), Flatten,
Reshape, transpose, Concat, softmax/log_softmax, LayerNorm, Embedding,
Dropout (identity at inference), elemwise/broadcast add/sub/mul/div.
"""
import json
import struct

import torch

from . import _proto as P

__all__ = ['export_model', 'import_model', 'import_to_gluon']

_TORCH2ONNX = {torch.float32: P.FLOAT, torch.float16: P.FLOAT16,
               torch.bfloat16: P.BFLOAT16, torch.float64: P.DOUBLE,
               torch.int64: P.INT64, torch.int32: P.INT32,
               torch.int8: P.INT8, torch.uint8: P.UINT8,
               torch.bool: P.BOOL}
_ONNX2TORCH = {v: k for k, v in _TORCH2ONNX.items()}


def _tensor_bytes(t):
    t = t.detach().cpu().contiguous()
    return t.numpy().tobytes() if t.dtype != torch.bfloat16 else \
        t.view(torch.int16).numpy().tobytes()


def _tuple(s):
    v = eval(s, {'__builtins__': {}})  # "(3, 3)" / "3" attr strings
    return tuple(v) if isinstance(v, (tuple, list)) else (int(v),)


def _bool(s):
    return str(s) in ('True', 'true', '1')


def export_model(sym_json, params, in_shapes, in_types=None, onnx_file=None,
                 dynamic_batch=False):
    """Symbol JSON (dict/str/path) + params {name: NDArray/tensor} ->
    serialized ONNX ModelProto bytes (also written to ``onnx_file``).

    ``in_shapes``: list of input shapes (graph inputs in JSON order).
    """
    if isinstance(sym_json, str):
        sym_json = json.load(open(sym_json)) if sym_json.endswith('.json') \
            else json.loads(sym_json)
    if hasattr(sym_json, 'tojson'):
        sym_json = json.loads(sym_json.tojson())
    def _to_torch(v):
        if hasattr(v, 'is_native') and v.is_native:
            # native-runtime arrays bridge through numpy
            import numpy as _np
            a = v.asnumpy()
            return torch.from_numpy(_np.ascontiguousarray(a))
        return getattr(v, 'handle', v)
    params = {k.split(':', 1)[-1]: _to_torch(v) for k, v in params.items()}

    nodes = sym_json['nodes']
    out_name = {}                       # (node_idx, out_idx) -> onnx name
    onnx_nodes, initializers, g_inputs = [], [], []
    in_types = in_types or [torch.float32] * len(in_shapes)
    in_i = 0

    def name_of(ref):
        return out_name[(ref[0], ref[1])]

    def emit(op, ins, outs, nname, attrs=b''):
        onnx_nodes.append(P.node_proto(op, ins, outs, nname, attrs))

    for i, node in enumerate(nodes):
        op, nname = node['op'], node['name']
        attrs = node.get('attrs', {})
        ins = [name_of(r) for r in node.get('inputs', [])]
        out = nname
        if op == 'null':
            if nname in params:
                t = params[nname]
                initializers.append(P.tensor_proto(
                    nname, list(t.shape), _TORCH2ONNX[t.dtype],
                    _tensor_bytes(t)))
            else:
                shape = list(in_shapes[in_i])
                if dynamic_batch:
                    shape[0] = 0
                g_inputs.append(P.value_info(
                    nname, _TORCH2ONNX[in_types[in_i]], shape))
                in_i += 1
            out_name[(i, 0)] = nname
            continue
        if op == 'Convolution':
            kh, kw = _tuple(attrs['kernel'])
            sh, sw = _tuple(attrs.get('stride', '(1, 1)'))
            ph, pw = _tuple(attrs.get('pad', '(0, 0)'))
            dh, dw = _tuple(attrs.get('dilate', '(1, 1)'))
            if attrs.get('layout', 'NCHW') != 'NCHW':
                raise ValueError('ONNX export supports NCHW conv only')
            a = (P.attr_field(P.attr_ints('kernel_shape', [kh, kw])) +
                 P.attr_field(P.attr_ints('strides', [sh, sw])) +
                 P.attr_field(P.attr_ints('pads', [ph, pw, ph, pw])) +
                 P.attr_field(P.attr_ints('dilations', [dh, dw])) +
                 P.attr_field(P.attr_int('group',
                                         int(attrs.get('num_group', 1)))))
            emit('Conv', ins, [out], nname, a)
        elif op == 'FullyConnected':
            no_bias = _bool(attrs.get('no_bias', 'False'))
            flatten = _bool(attrs.get('flatten', 'True'))
            if flatten:
                emit('Flatten', [ins[0]], [out + '_flat'], nname + '_flat',
                     P.attr_field(P.attr_int('axis', 1)))
                a_in = out + '_flat'
                gemm_in = [a_in, ins[1]] + ([] if no_bias else [ins[2]])
                emit('Gemm', gemm_in, [out], nname,
                     P.attr_field(P.attr_int('transB', 1)))
            else:
                # [B, ..., in] x W^T: Transpose the weight initializer ref
                # via MatMul(x, W^T) -- emit a Transpose node on W.
                emit('Transpose', [ins[1]], [out + '_wt'], nname + '_wt',
                     P.attr_field(P.attr_ints('perm', [1, 0])))
                emit('MatMul', [ins[0], out + '_wt'],
                     [out if no_bias else out + '_mm'], nname + '_mm')
                if not no_bias:
                    emit('Add', [out + '_mm', ins[2]], [out], nname)
        elif op == 'BatchNorm':
            if int(attrs.get('axis', 1)) != 1:
                raise ValueError('ONNX export supports axis=1 BatchNorm only')
            a = (P.attr_field(P.attr_float('epsilon',
                                           float(attrs.get('eps', 1e-5)))) +
                 P.attr_field(P.attr_float('momentum',
                                           float(attrs.get('momentum', 0.9)))))
            fuse = _bool(attrs.get('fuse_relu', 'False'))
            # fused residual arrives as a keyword input (model-zoo
            # ResNet hot path): BN -> Add(residual) -> Relu
            kw_names = [k for k in
                        str(attrs.get('__kw_inputs__', '')).split(',') if k]
            res = ins[5] if 'residual' in kw_names and len(ins) > 5 else None
            bn_out = out + '_bn' if (fuse or res) else out
            emit('BatchNormalization', ins[:5], [bn_out], nname, a)
            cur = bn_out
            if res is not None:
                nxt = out + '_res' if fuse else out
                emit('Add', [cur, res], [nxt], nname + '_add')
                cur = nxt
            if fuse:
                emit('Relu', [cur], [out], nname + '_relu')
        elif op == 'Activation':
            act = {'relu': 'Relu', 'sigmoid': 'Sigmoid', 'tanh': 'Tanh',
                   'softrelu': 'Softplus', 'softsign': 'Softsign'}[
                       attrs['act_type']]
            emit(act, ins, [out], nname)
        elif op == 'Pooling':
            ptype = attrs.get('pool_type', 'max')
            if _bool(attrs.get('global_pool', 'False')):
                emit('GlobalMaxPool' if ptype == 'max' else
                     'GlobalAveragePool', ins, [out], nname)
            else:
                kh, kw = _tuple(attrs['kernel'])
                sh, sw = _tuple(attrs.get('stride', '(1, 1)'))
                ph, pw = _tuple(attrs.get('pad', '(0, 0)'))
                a = (P.attr_field(P.attr_ints('kernel_shape', [kh, kw])) +
                     P.attr_field(P.attr_ints('strides', [sh, sw])) +
                     P.attr_field(P.attr_ints('pads', [ph, pw, ph, pw])))
                if ptype == 'avg':
                    a += P.attr_field(P.attr_int(
                        'count_include_pad',
                        int(_bool(attrs.get('count_include_pad', 'True')))))
                emit('MaxPool' if ptype == 'max' else 'AveragePool',
                     ins, [out], nname, a)
        elif op == 'Flatten':
            emit('Flatten', ins, [out], nname,
                 P.attr_field(P.attr_int('axis', 1)))
        elif op in ('Reshape', 'reshape'):
            shp = list(_tuple(attrs['shape']))
            sname = nname + '_shape'
            initializers.append(P.tensor_proto(
                sname, [len(shp)], P.INT64,
                struct.pack(f'<{len(shp)}q', *shp)))
            emit('Reshape', [ins[0], sname], [out], nname)
        elif op == 'transpose':
            perm = list(_tuple(attrs['axes']))
            emit('Transpose', ins, [out], nname,
                 P.attr_field(P.attr_ints('perm', perm)))
        elif op == 'Concat':
            emit('Concat', ins, [out], nname,
                 P.attr_field(P.attr_int('axis', int(attrs.get('dim', 1)))))
        elif op in ('softmax', 'log_softmax'):
            emit('Softmax' if op == 'softmax' else 'LogSoftmax', ins, [out],
                 nname,
                 P.attr_field(P.attr_int('axis', int(attrs.get('axis', -1)))))
        elif op == 'LayerNorm':
            a = (P.attr_field(P.attr_int('axis', int(attrs.get('axis', -1)))) +
                 P.attr_field(P.attr_float('epsilon',
                                           float(attrs.get('eps', 1e-5)))))
            emit('LayerNormalization', ins[:3], [out], nname, a)
        elif op == 'Embedding':
            emit('Gather', [ins[1], ins[0]], [out], nname)
        elif op == 'Dropout':
            emit('Identity', ins[:1], [out], nname)
        elif op in ('elemwise_add', 'broadcast_add', 'add_n'):
            emit('Add', ins, [out], nname)
        elif op in ('elemwise_sub', 'broadcast_sub'):
            emit('Sub', ins, [out], nname)
        elif op in ('elemwise_mul', 'broadcast_mul'):
            emit('Mul', ins, [out], nname)
        elif op in ('elemwise_div', 'broadcast_div'):
            emit('Div', ins, [out], nname)
        elif op == 'relu':
            emit('Relu', ins, [out], nname)
        else:
            raise NotImplementedError(f'ONNX export: op {op}')
        out_name[(i, 0)] = out

    heads = [name_of(h) for h in sym_json['heads']]
    # output value_info: dtype of first input, unknown shape (rank-only not
    # required by the spec; emit no dims)
    g_outputs = [P.value_info(h, _TORCH2ONNX[in_types[0]], []) for h in heads]
    graph = P.graph_proto(onnx_nodes, 'mxnet_amd', initializers,
                          g_inputs, g_outputs)
    model = P.model_proto(graph)
    if onnx_file:
        with open(onnx_file, 'wb') as f:
            f.write(model)
    return model


# ---------------------------------------------------------------------------
# import
# ---------------------------------------------------------------------------


def _parse_attrs(node_fields):
    out = {}
    for raw in node_fields.get(5, []):
        a = P.parse(raw)
        name = P.as_str(a, 1)
        atype = P.as_int(a, 20)
        if atype == P.A_FLOAT:
            out[name] = P.as_float(a, 2)
        elif atype == P.A_INT:
            out[name] = P.as_sint(a, 3)
        elif atype == P.A_STRING:
            out[name] = P.as_bytes(a, 4).decode('utf-8')
        elif atype == P.A_INTS:
            out[name] = P.repeated_ints(a, 8)
        elif atype == P.A_FLOATS:
            out[name] = P.repeated_floats(a, 7)
    return out


def _parse_tensor(raw):
    f = P.parse(raw)
    dims = P.repeated_ints(f, 1)
    dt = P.as_int(f, 2)
    name = P.as_str(f, 8)
    rawd = P.as_bytes(f, 9)
    tdt = _ONNX2TORCH[dt]
    if rawd:
        import numpy as np
        if tdt is torch.bfloat16:
            t = torch.frombuffer(bytearray(rawd), dtype=torch.int16) \
                .view(torch.bfloat16)
        else:
            npdt = {P.FLOAT: np.float32, P.FLOAT16: np.float16,
                    P.DOUBLE: np.float64, P.INT64: np.int64,
                    P.INT32: np.int32, P.INT8: np.int8, P.UINT8: np.uint8,
                    P.BOOL: np.bool_}[dt]
            t = torch.from_numpy(
                np.frombuffer(rawd, dtype=npdt).copy())
    elif 4 in f:
        t = torch.tensor(P.repeated_floats(f, 4), dtype=torch.float32)
    elif 7 in f:
        t = torch.tensor(P.repeated_ints(f, 7), dtype=torch.int64)
    else:
        t = torch.zeros(0)
    return name, t.reshape(dims).to(tdt) if dims else t.to(tdt)


def import_model(model_file):
    """ONNX file/bytes -> (Symbol, arg_params, aux_params) — the
    reference onnx2mx import_model contract."""
    buf = model_file if isinstance(model_file, (bytes, bytearray)) \
        else open(model_file, 'rb').read()
    model = P.parse(buf)
    graph = P.parse(P.as_bytes(model, 7))

    inits = {}
    for raw in graph.get(5, []):
        name, t = _parse_tensor(raw)
        inits[name] = t
    g_inputs = []
    for raw in graph.get(11, []):
        vi = P.parse(raw)
        name = P.as_str(vi, 1)
        if name not in inits:
            g_inputs.append(name)

    from ...symbol import Symbol, var
    import mxnet_amd.symbol as sym_mod
    env = {n: var(n) for n in g_inputs}
    for n in inits:
        env[n] = var(n)

    def g(name):
        return env[name]

    for raw in graph.get(1, []):
        nf = P.parse(raw)
        ins = [v.decode('utf-8') for v in nf.get(1, [])]
        outs = [v.decode('utf-8') for v in nf.get(2, [])]
        op = P.as_str(nf, 4)
        attrs = _parse_attrs(nf)
        S = sym_mod
        if op == 'Conv':
            k = attrs.get('kernel_shape', [1, 1])
            st = attrs.get('strides', [1, 1])
            pd = attrs.get('pads', [0, 0, 0, 0])
            dl = attrs.get('dilations', [1, 1])
            num_filter = inits[ins[1]].shape[0] if ins[1] in inits else 0
            y = S.Convolution(g(ins[0]), g(ins[1]),
                              *( [g(ins[2])] if len(ins) > 2 else []),
                              kernel=tuple(k), stride=tuple(st),
                              pad=(pd[0], pd[1]), dilate=tuple(dl),
                              num_filter=num_filter,
                              num_group=attrs.get('group', 1),
                              no_bias=len(ins) < 3, layout='NCHW')
        elif op == 'Gemm':
            assert attrs.get('transB', 0) == 1, 'Gemm import: transB=1 only'
            nh = inits[ins[1]].shape[0] if ins[1] in inits else 0
            y = S.FullyConnected(g(ins[0]), g(ins[1]),
                                 *( [g(ins[2])] if len(ins) > 2 else []),
                                 num_hidden=nh, no_bias=len(ins) < 3,
                                 flatten=False)
        elif op == 'MatMul':
            y = S.dot(g(ins[0]), g(ins[1]))
        elif op == 'BatchNormalization':
            y = S.BatchNorm(g(ins[0]), g(ins[1]), g(ins[2]), g(ins[3]),
                            g(ins[4]), eps=attrs.get('epsilon', 1e-5),
                            momentum=attrs.get('momentum', 0.9), axis=1)
        elif op in ('Relu', 'Sigmoid', 'Tanh', 'Softplus', 'Softsign'):
            act = {'Relu': 'relu', 'Sigmoid': 'sigmoid', 'Tanh': 'tanh',
                   'Softplus': 'softrelu', 'Softsign': 'softsign'}[op]
            y = S.Activation(g(ins[0]), act_type=act)
        elif op in ('MaxPool', 'AveragePool'):
            pd = attrs.get('pads', [0, 0, 0, 0])
            y = S.Pooling(g(ins[0]),
                          kernel=tuple(attrs.get('kernel_shape', [1, 1])),
                          stride=tuple(attrs.get('strides', [1, 1])),
                          pad=(pd[0], pd[1]),
                          pool_type='max' if op == 'MaxPool' else 'avg',
                          count_include_pad=bool(
                              attrs.get('count_include_pad', 1)))
        elif op in ('GlobalMaxPool', 'GlobalAveragePool'):
            y = S.Pooling(g(ins[0]), kernel=(1, 1), global_pool=True,
                          pool_type='max' if op == 'GlobalMaxPool' else 'avg')
        elif op == 'Flatten':
            y = S.Flatten(g(ins[0]))
        elif op == 'Reshape':
            shp = tuple(inits[ins[1]].tolist())
            y = S.reshape(g(ins[0]), shape=shp)
        elif op == 'Transpose':
            y = S.transpose(g(ins[0]), axes=tuple(attrs['perm']))
        elif op == 'Concat':
            y = S.Concat(*[g(i) for i in ins], dim=attrs.get('axis', 1))
        elif op in ('Softmax', 'LogSoftmax'):
            fn = S.softmax if op == 'Softmax' else S.log_softmax
            y = fn(g(ins[0]), axis=attrs.get('axis', -1))
        elif op == 'LayerNormalization':
            y = S.LayerNorm(g(ins[0]), g(ins[1]), g(ins[2]),
                            axis=attrs.get('axis', -1),
                            eps=attrs.get('epsilon', 1e-5))
        elif op == 'Gather':
            w = inits.get(ins[0])
            y = S.Embedding(g(ins[1]), g(ins[0]),
                            input_dim=w.shape[0] if w is not None else 0,
                            output_dim=w.shape[1] if w is not None else 0)
        elif op == 'Identity':
            y = g(ins[0])
        elif op == 'Add':
            y = g(ins[0]) + g(ins[1])
        elif op == 'Sub':
            y = g(ins[0]) - g(ins[1])
        elif op == 'Mul':
            y = g(ins[0]) * g(ins[1])
        elif op == 'Div':
            y = g(ins[0]) / g(ins[1])
        else:
            raise NotImplementedError(f'ONNX import: op {op}')
        env[outs[0]] = y

    outs = []
    for raw in graph.get(12, []):
        vi = P.parse(raw)
        outs.append(env[P.as_str(vi, 1)])
    sym = outs[0] if len(outs) == 1 else sym_mod.Group(outs)

    from ...ndarray.ndarray import NDArray
    arg_params, aux_params = {}, {}
    for n, t in inits.items():
        if 'running_mean' in n or 'running_var' in n or 'moving_' in n:
            aux_params[n] = NDArray(t)
        else:
            arg_params[n] = NDArray(t)
    return sym, arg_params, aux_params


def import_to_gluon(model_file, ctx=None):
    """ONNX file -> ready-to-run Gluon SymbolBlock (reference
    contrib.onnx.import_to_gluon)."""
    sym, arg_params, aux_params = import_model(model_file)
    from ...gluon.block import SymbolBlock
    inputs = [v for v in sym.list_inputs()
              if v not in arg_params and v not in aux_params]
    from ...symbol import var
    net = SymbolBlock(sym, [var(i) for i in inputs])
    params = dict(arg_params)
    params.update(aux_params)
    for name, p in net.collect_params().items():
        if name in params:
            t = params[name].handle
            p.shape = tuple(t.shape)
            p.dtype = str(t.dtype).replace('torch.', '')
            p.initialize(ctx=ctx)
            p.set_data(params[name])
    return net
