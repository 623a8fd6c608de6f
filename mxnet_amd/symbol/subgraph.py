"""Subgraph framework (reference src/operator/subgraph/: SubgraphProperty
registry + build_subgraph.cc partitioner).

The mechanism the reference keeps backend-agnostic: a registered
``SubgraphProperty`` selects connected op subsets of a Symbol graph; the
partitioner replaces each selected subset with a single fused node whose
attribute carries the sub-symbol.  Vendor backends (oneDNN/TensorRT in
the reference) are out of scope by design; the in-tree property fuses
elementwise chains (the reference's pointwise fusion pass shape).
"""
import json

from . import Symbol, _Node, load_json

__all__ = ['SubgraphProperty', 'register_subgraph_property',
           'partition_graph', 'ElementwiseFusionProperty']

_PROPERTIES = {}


def register_subgraph_property(name, prop):
    _PROPERTIES[name] = prop
    return prop


class SubgraphProperty:
    """Select ops for fusion (reference subgraph_property.h)."""

    name = 'base'

    def select(self, node):
        """Start a subgraph at this node?"""
        return False

    def select_input(self, node, input_node):
        """Grow the subgraph across this edge?"""
        return False

    def subgraph_op(self):
        return '_fused_subgraph'


class ElementwiseFusionProperty(SubgraphProperty):
    """Fuse chains of pointwise ops (reference pointwise_fusion_pass.cc)."""

    name = 'elemwise_fusion'
    _POINTWISE = {'Activation', 'relu', 'sigmoid', 'tanh', 'exp', 'log',
                  'sqrt', 'square', 'abs', 'negative', 'elemwise_add',
                  'elemwise_sub', 'elemwise_mul', 'elemwise_div',
                  '_plus_scalar', '_minus_scalar', '_mul_scalar',
                  '_div_scalar', 'clip'}

    def select(self, node):
        return node.op in self._POINTWISE

    def select_input(self, node, input_node):
        return input_node.op in self._POINTWISE


register_subgraph_property('elemwise_fusion', ElementwiseFusionProperty())


def _topo(sym):
    seen, order = set(), []

    def visit(n):
        if id(n) in seen:
            return
        seen.add(id(n))
        for src, _ in n.inputs:
            visit(src)
        order.append(n)
    visit(sym._node)
    return order


def partition_graph(sym, property='elemwise_fusion'):
    """Replace maximal selected subgraphs with fused nodes
    (reference build_subgraph.cc BuildSubgraph).  Returns a new Symbol;
    fused nodes carry op='_fused_subgraph' with the member op names in
    attrs['ops'] and the sub-symbol JSON in attrs['subgraph']."""
    prop = _PROPERTIES[property] if isinstance(property, str) else property
    order = _topo(sym)
    group = {}      # id(node) -> group index
    groups = []     # list of [nodes]
    for n in order:
        if not prop.select(n):
            continue
        # try to join a producer's group
        joined = None
        for src, _ in n.inputs:
            gi = group.get(id(src))
            if gi is not None and prop.select_input(n, src):
                joined = gi
                break
        if joined is None:
            joined = len(groups)
            groups.append([])
        groups[joined].append(n)
        group[id(n)] = joined

    # rebuild the graph bottom-up, replacing multi-node groups
    fused_of_group = {}
    mapping = {}

    def rebuild(n):
        if id(n) in mapping:
            return mapping[id(n)]
        new_inputs = [(rebuild(src), oi) for src, oi in n.inputs]
        gi = group.get(id(n))
        if gi is not None and len(groups[gi]) > 1:
            if gi not in fused_of_group:
                members = groups[gi]
                member_ids = {id(m) for m in members}
                # external inputs of the group, in first-use order
                ext = []
                for m in members:
                    for src, oi in m.inputs:
                        if id(src) not in member_ids and \
                                (id(src), oi) not in [(id(a), b) for a, b in ext]:
                            ext.append((src, oi))
                # self-contained sub-symbol: member clones over _in{k}
                # placeholder vars (externals must NOT leak whole
                # upstream graphs into the attribute)
                ext_key = {(id(a), b): k for k, (a, b) in enumerate(ext)}
                clones = {}
                for m in members:
                    mi = []
                    for src, oi in m.inputs:
                        if id(src) in member_ids:
                            mi.append((clones[id(src)], oi))
                        else:
                            mi.append((_Node('null',
                                             f'_in{ext_key[(id(src), oi)]}',
                                             {}, []), 0))
                    clones[id(m)] = _Node(m.op, m.name, dict(m.attrs), mi)
                sub_json = Symbol(clones[id(members[-1])]).tojson()
                fnode = _Node(
                    prop.subgraph_op(), f'fused_{prop.name}_{gi}',
                    {'ops': ','.join(m.op for m in members),
                     'subgraph': sub_json},
                    [(rebuild(src), oi) for src, oi in ext])
                fused_of_group[gi] = fnode
            # the group's last (output) member maps to the fused node;
            # inner members map there too (single-output chains)
            mapping[id(n)] = fused_of_group[gi]
            return mapping[id(n)]
        nn = _Node(n.op, n.name, dict(n.attrs), new_inputs, n.num_outputs,
                   getattr(n, 'aux', False))
        mapping[id(n)] = nn
        return nn

    return Symbol(rebuild(sym._node))
