"""Symbol — the symbolic graph API + nnvm-JSON serialization.

Reference parity: python/mxnet/symbol/symbol.py and nnvm graph JSON
(writer c_api_symbolic.cc:518; schema in SURVEY.md Appendix A).  The graph
here is a lightweight python IR: tracing HybridBlocks (classic
``hybrid_forward(F, ...)``) builds it, ``save``/``tojson`` emit the exact
reference JSON schema, and ``eval_dict`` interprets it imperatively by
dispatching each node to the mx.nd op of the same name (our CachedOp
dynamic-mode equivalent, reference cached_op.cc:707 DynamicForward).
All op attrs serialize as python-repr strings ("(3, 3)", "64", "True")
exactly like dmlc::Parameter printing.
"""
import json

__all__ = ['Symbol', 'var', 'Variable', 'Group', 'load', 'load_json']


class _Node:
    __slots__ = ('op', 'name', 'attrs', 'inputs', 'num_outputs', 'aux')

    def __init__(self, op, name, attrs=None, inputs=None, num_outputs=1, aux=False):
        self.op = op                       # 'null' for variables
        self.name = name
        self.attrs = attrs or {}
        self.inputs = inputs or []         # list[(Node, out_idx)]
        self.num_outputs = num_outputs
        self.aux = aux


_NAME_COUNTER = {}


def _gen_name(op):
    i = _NAME_COUNTER.get(op, 0)
    _NAME_COUNTER[op] = i + 1
    return f'{op.lower()}{i}'


def _attr_str(v):
    """dmlc::Parameter-style stringification ('(3, 3)', '64', 'True')."""
    if isinstance(v, bool):
        return 'True' if v else 'False'
    if isinstance(v, (tuple, list)):
        return '(' + ', '.join(str(int(x)) for x in v) + ')'
    if isinstance(v, float):
        return repr(v)
    return str(v)


class Symbol:
    """Handle to one output of a graph node."""

    __slots__ = ('_node', '_index')

    def __init__(self, node, index=0):
        self._node = node
        self._index = index

    @property
    def name(self):
        return self._node.name

    # -- composition ----------------------------------------------------
    def __add__(self, other):
        return _apply('elemwise_add', [self, other]) if isinstance(other, Symbol) \
            else _apply('_plus_scalar', [self], scalar=other)

    def __radd__(self, other):
        return self.__add__(other)

    def __sub__(self, other):
        return _apply('elemwise_sub', [self, other]) if isinstance(other, Symbol) \
            else _apply('_minus_scalar', [self], scalar=other)

    def __mul__(self, other):
        return _apply('elemwise_mul', [self, other]) if isinstance(other, Symbol) \
            else _apply('_mul_scalar', [self], scalar=other)

    def __rmul__(self, other):
        return self.__mul__(other)

    def __truediv__(self, other):
        return _apply('elemwise_div', [self, other]) if isinstance(other, Symbol) \
            else _apply('_div_scalar', [self], scalar=other)

    def __neg__(self):
        return _apply('negative', [self])

    def reshape(self, shape, **kwargs):
        return _apply('reshape', [self], shape=shape)

    def transpose(self, axes=None):
        return _apply('transpose', [self], axes=axes)

    def flatten(self):
        return _apply('Flatten', [self])

    def expand_dims(self, axis):
        return _apply('expand_dims', [self], axis=axis)

    def squeeze(self, axis=None):
        return _apply('squeeze', [self], axis=axis)

    def sum(self, axis=None, keepdims=False):
        return _apply('sum', [self], axis=axis, keepdims=keepdims)

    def mean(self, axis=None, keepdims=False):
        return _apply('mean', [self], axis=axis, keepdims=keepdims)

    def astype(self, dtype):
        return _apply('Cast', [self], dtype=dtype)

    def slice_axis(self, axis, begin, end):
        return _apply('slice_axis', [self], axis=axis, begin=begin, end=end)

    def __getitem__(self, idx):
        if isinstance(idx, int) and self._node.num_outputs > 1:
            return Symbol(self._node, idx)
        raise NotImplementedError

    # -- graph walking ----------------------------------------------------
    def _topo(self):
        seen, order = set(), []

        def visit(node):
            if id(node) in seen:
                return
            seen.add(id(node))
            for inp, _ in node.inputs:
                visit(inp)
            order.append(node)

        visit(self._node)
        return order

    def list_arguments(self):
        return [n.name for n in self._topo() if n.op == 'null' and not n.aux]

    def list_auxiliary_states(self):
        return [n.name for n in self._topo() if n.op == 'null' and n.aux]

    def list_inputs(self):
        return [n.name for n in self._topo() if n.op == 'null']

    def list_outputs(self):
        return [f'{self._node.name}_output']

    def get_internals(self):
        return [Symbol(n) for n in self._topo()]

    # -- serialization (reference JSON schema) ---------------------------
    def tojson(self):
        nodes = self._topo()
        idx = {id(n): i for i, n in enumerate(nodes)}
        jnodes = []
        row_ptr = [0]
        for n in nodes:
            jn = {'op': n.op, 'name': n.name,
                  'inputs': [[idx[id(src)], oi, 0] for src, oi in n.inputs]}
            if n.attrs:
                jn['attrs'] = {k: _attr_str(v) for k, v in n.attrs.items()
                               if v is not None}
            jnodes.append(jn)
            row_ptr.append(row_ptr[-1] + n.num_outputs)
        heads = getattr(self, '_heads', None) or [(self._node, self._index)]
        return json.dumps({
            'nodes': jnodes,
            'arg_nodes': [i for i, n in enumerate(nodes) if n.op == 'null'],
            'node_row_ptr': row_ptr,
            'heads': [[idx[id(n)], oi, 0] for n, oi in heads],
            'attrs': {'mxnet_version': ['int', 20000]},
        }, indent=2)

    def save(self, fname):
        with open(fname, 'w') as f:
            f.write(self.tojson())

    # -- execution (CachedOp dynamic mode) -------------------------------
    def eval_dict(self, feed):
        """Interpret the graph with NDArray inputs; returns list of outputs."""
        from ..ndarray import ops as F
        values = {}
        for n in self._topo():
            if n.op == 'null':
                if n.name not in feed:
                    raise ValueError(f'missing input {n.name}')
                values[id(n)] = (feed[n.name],)
                continue
            args = [values[id(src)][oi] for src, oi in n.inputs]
            fn = _EVAL_TABLE.get(n.op) or getattr(F, n.op, None)
            if fn is None:
                raise NotImplementedError(f'symbol eval: op {n.op}')
            kwargs = dict(n.attrs)
            kw_in = kwargs.pop('__kw_inputs__', None)
            if kw_in:
                names = [k for k in str(kw_in).split(',') if k]
                for k, v in zip(names, args[len(args) - len(names):]):
                    kwargs[k] = v
                args = args[:len(args) - len(names)]
            out = fn(*args, **kwargs)
            values[id(n)] = out if isinstance(out, tuple) else \
                (tuple(out) if isinstance(out, list) else (out,))
        heads = getattr(self, '_heads', None) or [(self._node, self._index)]
        return [values[id(n)][oi] for n, oi in heads]

    def eval(self, ctx=None, **kwargs):
        outs = self.eval_dict(kwargs)
        return outs

    def bind(self, ctx, args, args_grad=None, **kwargs):
        raise NotImplementedError('legacy executor API: use SymbolBlock')

    def __repr__(self):
        return f'<Symbol {self.name}>'


def _scalar_op(opname):
    def fn(x, scalar=0.0, **kw):
        s = float(scalar)
        from ..ndarray.ndarray import NDArray
        if opname == '_plus_scalar':
            return NDArray(x._t + s)
        if opname == '_minus_scalar':
            return NDArray(x._t - s)
        if opname == '_mul_scalar':
            return NDArray(x._t * s)
        if opname == '_div_scalar':
            return NDArray(x._t / s)
    return fn


def _eval_fused_subgraph(*args, **kwargs):
    from ..contrib.fusion import execute_fused
    return execute_fused(*args, **kwargs)


_EVAL_TABLE = {
    '_fused_subgraph': _eval_fused_subgraph,
    '_plus_scalar': _scalar_op('_plus_scalar'),
    '_minus_scalar': _scalar_op('_minus_scalar'),
    '_mul_scalar': _scalar_op('_mul_scalar'),
    '_div_scalar': _scalar_op('_div_scalar'),
}


def _parse_attr(v):
    """Inverse of _attr_str for eval: parse '(3, 3)' / '64' / 'True'."""
    if not isinstance(v, str):
        return v
    s = v.strip()
    if s in ('True', 'False'):
        return s == 'True'
    if s == 'None':
        return None
    if s.startswith('(') or s.startswith('['):
        inner = s[1:-1].strip()
        if not inner:
            return ()
        return tuple(int(float(x)) for x in inner.split(','))
    try:
        return int(s)
    except ValueError:
        pass
    try:
        return float(s)
    except ValueError:
        return s


class Variable(Symbol):
    pass


def var(name, shape=None, dtype=None, aux=False, **kwargs):
    return Symbol(_Node('null', name, attrs={}, aux=aux))


def Group(symbols):
    """Multi-output symbol."""
    heads = [(s._node, s._index) for s in symbols]
    g = Symbol(symbols[-1]._node, symbols[-1]._index)
    g2 = Symbol.__new__(Symbol)
    g2._node = symbols[-1]._node
    g2._index = symbols[-1]._index
    # Symbol uses __slots__; carry heads on a subclass instead
    grp = _GroupSymbol(symbols[-1]._node, symbols[-1]._index)
    grp._heads = heads
    return grp


class _GroupSymbol(Symbol):
    __slots__ = ('_heads',)

    def _topo(self):
        seen, order = set(), []

        def visit(node):
            if id(node) in seen:
                return
            seen.add(id(node))
            for inp, _ in node.inputs:
                visit(inp)
            order.append(node)

        for n, _ in self._heads:
            visit(n)
        return order

    def list_outputs(self):
        return [f'{n.name}_output{i}' for n, i in self._heads]


def load_json(s):
    g = json.loads(s)
    nodes = []
    for jn in g['nodes']:
        attrs = {k: _parse_attr(v) for k, v in jn.get('attrs', {}).items()}
        n = _Node(jn['op'], jn['name'], attrs)
        nodes.append(n)
    for n, jn in zip(nodes, g['nodes']):
        n.inputs = [(nodes[i], oi) for i, oi, _ in jn['inputs']]
    heads = [(nodes[i], oi) for i, oi, _ in g['heads']]
    if len(heads) == 1:
        return Symbol(heads[0][0], heads[0][1])
    grp = _GroupSymbol(heads[-1][0], heads[-1][1])
    grp._heads = heads
    return grp


def load(fname):
    with open(fname) as f:
        return load_json(f.read())


# ---------------------------------------------------------------------------
# symbolic op constructors — mirror mx.nd names so hybrid_forward(F, ...) works
# ---------------------------------------------------------------------------

def _apply(op, inputs, name=None, num_outputs=1, **attrs):
    node = _Node(op, name or _gen_name(op),
                 {k: v for k, v in attrs.items() if v is not None},
                 [(s._node, s._index) for s in inputs], num_outputs)
    return Symbol(node)


def _make_op(opname, arity='var'):
    def op(*inputs, name=None, **attrs):
        syms = [i for i in inputs if isinstance(i, Symbol)]
        # Symbol-valued keyword args (e.g. BatchNorm residual=...) become
        # extra graph inputs; __kw_inputs__ records their keywords so the
        # evaluator can reconstruct the call (JSON-stringified like all
        # attrs, reference-compatible: unknown attrs are ignored there)
        kw_syms = [(k, v) for k, v in list(attrs.items())
                   if isinstance(v, Symbol)]
        for k, v in kw_syms:
            del attrs[k]
            syms.append(v)
        if kw_syms:
            attrs['__kw_inputs__'] = ','.join(k for k, _ in kw_syms)
        return _apply(opname, syms, name=name, **attrs)
    op.__name__ = opname
    return op


_OP_NAMES = [
    'FullyConnected', 'Convolution', 'Activation', 'Pooling', 'BatchNorm',
    'LayerNorm', 'Embedding', 'Dropout', 'LeakyReLU', 'RNN',
    'softmax', 'log_softmax', 'softmin', 'SoftmaxOutput',
    'Flatten', 'Concat', 'concat', 'flatten',
    'exp', 'log', 'sqrt', 'square', 'abs', 'sign', 'sin', 'cos', 'tanh',
    'sigmoid', 'relu', 'erf', 'negative', 'reciprocal',
    'elemwise_add', 'elemwise_sub', 'elemwise_mul', 'elemwise_div',
    'broadcast_add', 'broadcast_sub', 'broadcast_mul', 'broadcast_div',
    'broadcast_maximum', 'broadcast_minimum', 'add_n',
    'sum', 'mean', 'max', 'min', 'prod', 'norm', 'argmax', 'argmin',
    'reshape', 'transpose', 'expand_dims', 'squeeze', 'stack', 'split',
    'slice', 'slice_axis', 'slice_like', 'take', 'pick', 'one_hot', 'tile',
    'repeat', 'pad', 'broadcast_to', 'broadcast_like', 'broadcast_axis',
    'zeros_like', 'ones_like', 'Cast', 'cast', 'dot', 'batch_dot',
    'linalg_gemm2', 'SequenceMask', 'sequence_mask', 'where', 'clip', 'topk',
    'sort', 'argsort',
]
for _n in _OP_NAMES:
    globals().setdefault(_n, _make_op(_n))
