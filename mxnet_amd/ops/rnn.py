"""Fused-RNN op semantics (reference src/operator/rnn.cc / rnn-inl.h).

Weight layout matches the reference's fused ``RNN`` op: one flat parameter
vector per network — for each layer, i2h weights for all gates, then h2h
weights, then (after ALL layer weights) per-layer i2h biases then h2h
biases.  Gate order: LSTM [i, f, g(c~), o]; GRU [r, z, n].

Execution: time-step loop over fused cell kernels; the per-step gate GEMMs
run on the native MFMA GEMM on GPU (through ops.nn.fully_connected), the
pointwise cell math runs the fused LSTM-cell HIP kernel when available
(SURVEY §2.2 RNN row; BASELINE config 5).
"""
import torch

from .dispatch import hipops, use_hip
from . import nn as _nn


def _gate_count(mode):
    return {'rnn_relu': 1, 'rnn_tanh': 1, 'lstm': 4, 'gru': 3}[mode]


def slice_params(params, mode, input_size, hidden_size, num_layers,
                 bidirectional=False):
    """Split the flat param vector into per-layer (wi, wh, bi, bh)."""
    ng = _gate_count(mode)
    dirs = 2 if bidirectional else 1
    shapes = []
    for layer in range(num_layers):
        for d in range(dirs):
            isz = input_size if layer == 0 else hidden_size * dirs
            shapes.append(('wi', layer, d, (ng * hidden_size, isz)))
            shapes.append(('wh', layer, d, (ng * hidden_size, hidden_size)))
    for layer in range(num_layers):
        for d in range(dirs):
            shapes.append(('bi', layer, d, (ng * hidden_size,)))
            shapes.append(('bh', layer, d, (ng * hidden_size,)))
    out = {}
    off = 0
    for kind, layer, d, shp in shapes:
        n = 1
        for s in shp:
            n *= s
        out[(kind, layer, d)] = params[off:off + n].reshape(shp)
        off += n
    assert off == params.numel(), f'param size {params.numel()} != expected {off}'
    return out


def param_size(mode, input_size, hidden_size, num_layers, bidirectional=False):
    ng = _gate_count(mode)
    dirs = 2 if bidirectional else 1
    total = 0
    for layer in range(num_layers):
        isz = input_size if layer == 0 else hidden_size * dirs
        total += dirs * (ng * hidden_size * isz + ng * hidden_size * hidden_size)
    total += num_layers * dirs * 2 * ng * hidden_size
    return total


def _lstm_cell(x_gates, h, c, wh, bh):
    """One LSTM step given precomputed input gates. Fused HIP kernel on GPU;
    the recurrent gate GEMM runs the MFMA gemm_nt path."""
    gates = x_gates + _nn.fully_connected(h.contiguous(), wh, bh)
    if use_hip(gates):
        ext = hipops()
        if ext is not None and hasattr(ext, 'lstm_cell_fwd'):
            return ext.lstm_cell_fwd(gates, c)
    H = h.shape[-1]
    i, f, g, o = gates.split(H, dim=-1)
    i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
    g = torch.tanh(g)
    c_new = f * c + i * g
    h_new = o * torch.tanh(c_new)
    return h_new, c_new


def _gru_cell(x_gates, h, wh, bh):
    H = h.shape[-1]
    hg = _nn.fully_connected(h.contiguous(), wh, bh)
    xr, xz, xn = x_gates.split(H, dim=-1)
    hr, hz, hn = hg.split(H, dim=-1)
    r = torch.sigmoid(xr + hr)
    z = torch.sigmoid(xz + hz)
    n = torch.tanh(xn + r * hn)
    return (1 - z) * n + z * h


def _rnn_cell(x_gates, h, wh, bh, act):
    pre = x_gates + _nn.fully_connected(h.contiguous(), wh, bh)
    return torch.relu(pre) if act == 'relu' else torch.tanh(pre)


def rnn_forward(x, params, h0, c0, mode, hidden_size, num_layers,
                bidirectional=False, dropout=0.0, training=False):
    """x: [T, N, I] (TNC).  Returns (out [T,N,H*dirs], hn, cn)."""
    T, N, I = x.shape
    dirs = 2 if bidirectional else 1
    p = slice_params(params, mode, I, hidden_size, num_layers, bidirectional)
    h0 = h0.reshape(num_layers * dirs, N, hidden_size)
    if mode == 'lstm':
        c0 = c0.reshape(num_layers * dirs, N, hidden_size)
    hs, cs = [], []
    inp = x
    for layer in range(num_layers):
        outs_dir = []
        for d in range(dirs):
            wi, wh = p[('wi', layer, d)], p[('wh', layer, d)]
            bi, bh = p[('bi', layer, d)], p[('bh', layer, d)]
            idx = layer * dirs + d
            h = h0[idx]
            c = c0[idx] if mode == 'lstm' else None
            seq = inp if d == 0 else torch.flip(inp, dims=[0])
            # one big gate GEMM over all timesteps (MFMA-friendly shape)
            xg = _nn.fully_connected(seq.reshape(T * N, -1).contiguous(), wi, bi) \
                .reshape(T, N, -1)
            outs = []
            for t in range(T):
                if mode == 'lstm':
                    h, c = _lstm_cell(xg[t], h, c, wh, bh)
                elif mode == 'gru':
                    h = _gru_cell(xg[t], h, wh, bh)
                else:
                    h = _rnn_cell(xg[t], h, wh, bh,
                                  'relu' if mode == 'rnn_relu' else 'tanh')
                outs.append(h)
            out = torch.stack(outs, dim=0)
            if d == 1:
                out = torch.flip(out, dims=[0])
            outs_dir.append(out)
            hs.append(h)
            if mode == 'lstm':
                cs.append(c)
        inp = torch.cat(outs_dir, dim=-1) if dirs > 1 else outs_dir[0]
        if dropout > 0 and training and layer < num_layers - 1:
            inp = torch.nn.functional.dropout(inp, dropout, training)
    hn = torch.stack(hs, dim=0)
    cn = torch.stack(cs, dim=0) if mode == 'lstm' else None
    return inp, hn, cn


def rnn_ndarray(data, parameters, state, state_cell, mode, state_size,
                num_layers, bidirectional, p):
    """NDArray-level entry used by mx.nd.RNN."""
    from ..ndarray.ndarray import NDArray
    from .. import autograd as _ag
    x = data._t
    out, hn, cn = rnn_forward(
        x, parameters._t, state._t,
        state_cell._t if state_cell is not None else None,
        mode, state_size, num_layers, bidirectional, p,
        training=_ag.is_training())
    if mode == 'lstm':
        return NDArray(out), NDArray(hn), NDArray(cn)
    return NDArray(out), NDArray(hn)


# ---------------------------------------------------------------------------
# native-runtime path: the same fused-RNN semantics composed from native
# registry ops (FC GEMMs + elementwise + strided slices); every step is a
# tape node so backward comes from the own autograd, not torch.
# ---------------------------------------------------------------------------
def _n_slice_params(params, mode, input_size, hidden_size, num_layers,
                    bidirectional=False):
    """slice_params over a native flat vector: 1-D recorded slices (the
    scatter backward accumulates straight into the flat leaf)."""
    ng = _gate_count(mode)
    dirs = 2 if bidirectional else 1
    shapes = []
    for layer in range(num_layers):
        for d in range(dirs):
            isz = input_size if layer == 0 else hidden_size * dirs
            shapes.append(('wi', layer, d, (ng * hidden_size, isz)))
            shapes.append(('wh', layer, d, (ng * hidden_size, hidden_size)))
    for layer in range(num_layers):
        for d in range(dirs):
            shapes.append(('bi', layer, d, (ng * hidden_size,)))
            shapes.append(('bh', layer, d, (ng * hidden_size,)))
    out = {}
    off = 0
    for kind, layer, d, shp in shapes:
        n = 1
        for s in shp:
            n *= s
        out[(kind, layer, d)] = params[off:off + n].reshape(shp)
        off += n
    assert off == params.size
    return out


def _n_flip0(x):
    """Reverse along axis 0 as one strided gather (negative stride)."""
    T = x.shape[0]
    inner = 1
    for s in x.shape[1:]:
        inner *= s
    strides = [-inner] + [1] * 0
    # plan: [T, inner] over the flattened tail
    return x._invoke('_strided_copy', [x], {
        'shape': '(%d,%d,)' % (T, inner),
        'strides': '(%d,1,)' % (-inner),
        'offset': str((T - 1) * inner),
        'oshape': '(' + ','.join(str(s) for s in x.shape) + ',)'})


def _n_concat0(parts):
    """Stack [N,H] steps into [T,N,H] (one concat + reshape)."""
    first = parts[0]
    y = first._invoke('concat', parts, {'dim': '0'})
    return y.reshape((len(parts),) + tuple(first.shape))


def rnn_forward_native(x, params, h0, c0, mode, hidden_size, num_layers,
                       bidirectional=False, dropout=0.0, training=False):
    """Native rnn_forward: x [T,N,I] native NDArray; returns
    (out, hn, cn) native NDArrays (reference rnn-inl.h semantics)."""
    from ..ndarray import ops as F
    T, N, I = x.shape
    H = hidden_size
    dirs = 2 if bidirectional else 1
    p = _n_slice_params(params, mode, I, H, num_layers, bidirectional)
    h0 = h0.reshape(num_layers * dirs, N, H)
    if mode == 'lstm':
        c0 = c0.reshape(num_layers * dirs, N, H)

    def sig(z):
        return F.Activation(z, act_type='sigmoid')

    def tnh(z):
        return F.Activation(z, act_type='tanh')

    hs, cs = [], []
    inp = x
    for layer in range(num_layers):
        outs_dir = []
        for d in range(dirs):
            wi, wh = p[('wi', layer, d)], p[('wh', layer, d)]
            bi, bh = p[('bi', layer, d)], p[('bh', layer, d)]
            idx = layer * dirs + d
            h = h0[idx]
            c = c0[idx] if mode == 'lstm' else None
            seq = inp if d == 0 else _n_flip0(inp)
            xg = F.FullyConnected(seq.reshape(T * N, -1), wi, bi,
                                  flatten=False).reshape(T, N, -1)
            outs = []
            for t in range(T):
                if mode == 'lstm':
                    gates = xg[t] + F.FullyConnected(h, wh, bh,
                                                     flatten=False)
                    gi = sig(gates[:, 0:H])
                    gf = sig(gates[:, H:2 * H])
                    gg = tnh(gates[:, 2 * H:3 * H])
                    go = sig(gates[:, 3 * H:4 * H])
                    c = gf * c + gi * gg
                    h = go * tnh(c)
                elif mode == 'gru':
                    # reference gru gate math: r,z from x+h parts; the
                    # reset gate scales the h-part of the n gate
                    hg = F.FullyConnected(h, wh, bh, flatten=False)
                    xt = xg[t]
                    r = sig(xt[:, 0:H] + hg[:, 0:H])
                    z = sig(xt[:, H:2 * H] + hg[:, H:2 * H])
                    n_ = tnh(xt[:, 2 * H:3 * H] + r * hg[:, 2 * H:3 * H])
                    h = (z * -1.0 + 1.0) * n_ + z * h
                else:
                    gates = xg[t] + F.FullyConnected(h, wh, bh,
                                                     flatten=False)
                    act = 'relu' if mode == 'rnn_relu' else 'tanh'
                    h = F.Activation(gates, act_type=act)
                outs.append(h)
            out = _n_concat0(outs)
            if d == 1:
                out = _n_flip0(out)
            outs_dir.append(out)
            hs.append(h)
            if mode == 'lstm':
                cs.append(c)
        if dirs > 1:
            inp = outs_dir[0]._invoke('concat', outs_dir, {'dim': '2'})
        else:
            inp = outs_dir[0]
        if dropout > 0 and training and layer < num_layers - 1:
            inp = F.Dropout(inp, p=dropout)
    hn = _n_concat0(hs)
    cn = _n_concat0(cs) if mode == 'lstm' else None
    return inp, hn, cn
