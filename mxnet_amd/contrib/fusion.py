"""RTC pointwise fusion (reference src/operator/fusion/fused_op.cu:
NVRTC-compiled fused elementwise subgraphs; hiprtc here).

``partition_graph(sym)`` (symbol.subgraph) marks elementwise chains as
``_fused_subgraph`` nodes; this module executes them: on GPU each chain
is code-generated into ONE HIP kernel, compiled once with
``mx.rtc.HipModule`` (cached by sub-graph JSON + dtype) and launched on
the current stream; on CPU the sub-symbol is interpreted.
"""
import json

import torch

from ..ndarray.ndarray import NDArray

__all__ = ['execute_fused', 'codegen']

_KERNEL_CACHE = {}

_UNARY = {
    'relu': 'fmaxf({0}, 0.f)',
    'sigmoid': '1.f / (1.f + __expf(-{0}))',
    'tanh': 'tanhf({0})',
    'exp': '__expf({0})',
    'log': '__logf({0})',
    'sqrt': 'sqrtf({0})',
    'square': '({0} * {0})',
    'abs': 'fabsf({0})',
    'negative': '(-{0})',
    'softrelu': '__logf(1.f + __expf({0}))',
    'softsign': '({0} / (1.f + fabsf({0})))',
}
_BINARY = {
    'elemwise_add': '({0} + {1})', 'elemwise_sub': '({0} - {1})',
    'elemwise_mul': '({0} * {1})', 'elemwise_div': '({0} / {1})',
}
_SCALAR = {
    '_plus_scalar': '({0} + {s})', '_minus_scalar': '({0} - {s})',
    '_mul_scalar': '({0} * {s})', '_div_scalar': '({0} / {s})',
}


def codegen(sub_json, dtype='float32'):
    """Sub-symbol JSON -> (HIP source, n_inputs).  All math in fp32;
    I/O in the tensor dtype."""
    g = json.loads(sub_json) if isinstance(sub_json, str) else sub_json
    nodes = g['nodes']
    ctype = {'float32': 'float', 'float16': '_Float16'}[dtype]
    ins, exprs = [], {}
    body = []
    for i, n in enumerate(nodes):
        op, name = n['op'], n['name']
        attrs = n.get('attrs', {})
        srcs = [f'v{r[0]}' for r in n.get('inputs', [])]
        if op == 'null':
            idx = int(name[3:]) if name.startswith('_in') else len(ins)
            while len(ins) <= idx:
                ins.append(None)
            ins[idx] = i
            body.append(f'    float v{i} = (float)in{idx}[i];')
            continue
        if op == 'Activation':
            expr = _UNARY[attrs['act_type']].format(*srcs)
        elif op in _UNARY:
            expr = _UNARY[op].format(*srcs)
        elif op in _BINARY:
            expr = _BINARY[op].format(*srcs)
        elif op in _SCALAR:
            expr = _SCALAR[op].format(*srcs,
                                      s=f"{float(attrs['scalar'])!r}f")
        elif op == 'clip':
            lo, hi = float(attrs['a_min']), float(attrs['a_max'])
            expr = f'fminf(fmaxf({srcs[0]}, {lo}f), {hi}f)'
        else:
            raise NotImplementedError(f'fusion codegen: op {op}')
        body.append(f'    float v{i} = {expr};')
    out_v = f'v{len(nodes) - 1}'
    params = ', '.join(f'const {ctype}* in{k}' for k in range(len(ins)))
    src = f'''extern "C" __global__ void fused({params}, {ctype}* out, long n) {{
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {{
{chr(10).join(body)}
    out[i] = ({ctype}){out_v};
  }}
}}
'''
    return src, len(ins)


def _get_kernel(sub_json, dtype):
    key = (sub_json, dtype)
    if key not in _KERNEL_CACHE:
        from .. import rtc
        src, n_in = codegen(sub_json, dtype)
        mod = rtc.HipModule(src)
        sig = ', '.join(f'const float* in{k}' for k in range(n_in)) + \
            ', float* out, long n'
        _KERNEL_CACHE[key] = (mod.get_kernel('fused', sig), n_in)
    return _KERNEL_CACHE[key]


def execute_fused(*args, ops=None, subgraph=None, **_ignored):
    """Evaluator for ``_fused_subgraph`` nodes (registered in the symbol
    eval table)."""
    tensors = [a.handle if isinstance(a, NDArray) else a for a in args]
    t0 = tensors[0]
    dt = {torch.float32: 'float32', torch.float16: 'float16'}.get(t0.dtype)
    same = all(t.shape == t0.shape and t.dtype == t0.dtype
               for t in tensors)
    if t0.is_cuda and dt is not None and same and not torch.is_grad_enabled():
        kern, n_in = _get_kernel(subgraph, dt)
        assert n_in == len(tensors), 'fused subgraph arity mismatch'
        out = torch.empty_like(t0)
        n = t0.numel()
        grid = min((n + 255) // 256, 4096)
        kern.launch(tuple(tensors) + (out, n), None,
                    (int(grid), 1, 1), (256, 1, 1))
        return NDArray(out)
    # interpret the sub-symbol (CPU / autograd / mixed-shape fallback)
    from ..symbol import load_json
    sub = load_json(subgraph if isinstance(subgraph, str)
                    else json.dumps(subgraph))
    feed = {f'_in{k}': NDArray(t) for k, t in enumerate(tensors)}
    return sub.eval_dict(feed)[0]
