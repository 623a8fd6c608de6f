"""Network visualization (reference python/mxnet/visualization.py):
``print_summary`` for Symbols and Gluon blocks; ``plot_network`` emits
graphviz source text (no graphviz runtime dependency needed to build it).
"""
import json

__all__ = ['print_summary', 'plot_network']


def _symbol_nodes(symbol):
    conf = json.loads(symbol.tojson())
    return conf['nodes'], conf.get('heads', [])


def print_summary(symbol, shape=None, line_length=120, positions=(.44, .64, .74, 1.)):
    """Layer-by-layer table of a Symbol graph (reference visualization.py:43)."""
    nodes, _ = _symbol_nodes(symbol)
    positions = [int(line_length * p) for p in positions]
    fields = ['Layer (type)', 'Output Shape', 'Param #', 'Previous Layer']
    line = ''
    for f, p in zip(fields, positions):
        line = (line + f)[:p].ljust(p)
    print('=' * line_length)
    print(line)
    print('=' * line_length)
    for node in nodes:
        op = node['op']
        if op == 'null':
            continue
        name = node['name']
        inputs = [nodes[i[0]]['name'] for i in node.get('inputs', [])
                  if nodes[i[0]]['op'] != 'null']
        row = [f'{name} ({op})', '', '', ','.join(inputs)]
        line = ''
        for f, p in zip(row, positions):
            line = (line + str(f))[:p].ljust(p)
        print(line)
    print('=' * line_length)


def plot_network(symbol, title='plot', save_format='pdf', shape=None,
                 node_attrs=None, hide_weights=True):
    """Build graphviz DOT source for a Symbol graph (reference
    visualization.py:216).  Returns the DOT text; callers with graphviz
    installed can render it."""
    nodes, heads = _symbol_nodes(symbol)
    lines = [f'digraph "{title}" {{', '  rankdir=BT;']
    for i, node in enumerate(nodes):
        op = node['op']
        name = node['name']
        if op == 'null':
            if hide_weights and (name.endswith('weight') or
                                 name.endswith('bias') or
                                 name.endswith('gamma') or
                                 name.endswith('beta') or 'running' in name):
                continue
            lines.append(f'  n{i} [label="{name}", shape=oval];')
        else:
            attrs = node.get('attrs', {})
            extra = ''
            if op == 'Convolution':
                extra = f"\\n{attrs.get('kernel', '')}/{attrs.get('stride', '')}" \
                        f", {attrs.get('num_filter', '')}"
            elif op == 'FullyConnected':
                extra = f"\\n{attrs.get('num_hidden', '')}"
            lines.append(f'  n{i} [label="{name}\\n{op}{extra}", shape=box];')
    emitted = {l.split(' ')[2][1:].rstrip('];') for l in lines if '[label' in l}
    for i, node in enumerate(nodes):
        if node['op'] == 'null':
            continue
        for inp in node.get('inputs', []):
            j = inp[0]
            src = nodes[j]
            if src['op'] == 'null' and hide_weights and (
                    src['name'].endswith(('weight', 'bias', 'gamma', 'beta'))
                    or 'running' in src['name']):
                continue
            lines.append(f'  n{j} -> n{i};')
    lines.append('}')
    return '\n'.join(lines)
