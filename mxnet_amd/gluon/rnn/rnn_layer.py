"""Fused RNN layers (reference gluon/rnn/rnn_layer.py over the fused
`RNN` op, src/operator/rnn.cc)."""
import torch

from ..block import HybridBlock
from ..parameter import Parameter
from ...ndarray.ndarray import NDArray, zeros
from ...ops import rnn as _rnn_ops
from ... import initializer as init


class _RNNLayer(HybridBlock):
    def __init__(self, mode, hidden_size, num_layers=1, layout='TNC',
                 dropout=0.0, bidirectional=False, input_size=0,
                 i2h_weight_initializer=None, h2h_weight_initializer=None,
                 i2h_bias_initializer='zeros', h2h_bias_initializer='zeros',
                 dtype='float32', **kwargs):
        super().__init__(**kwargs)
        assert layout in ('TNC', 'NTC')
        self._mode = mode
        self._hidden_size = hidden_size
        self._num_layers = num_layers
        self._layout = layout
        self._dropout = dropout
        self._bidirectional = bidirectional
        self._input_size = input_size
        self._dtype = dtype
        n = _rnn_ops.param_size(mode, input_size, hidden_size, num_layers,
                                bidirectional) if input_size else 0
        self.parameters = Parameter('parameters', shape=(n,), dtype=dtype,
                                    init=i2h_weight_initializer or init.Uniform(0.1),
                                    allow_deferred_init=True)

    def infer_shape(self, x, *args):
        isz = x.shape[-1]
        self._input_size = isz
        n = _rnn_ops.param_size(self._mode, isz, self._hidden_size,
                                self._num_layers, self._bidirectional)
        self.parameters.shape = (n,)

    def state_info(self, batch_size=0):
        dirs = 2 if self._bidirectional else 1
        shape = (self._num_layers * dirs, batch_size, self._hidden_size)
        if self._mode == 'lstm':
            return [shape, shape]
        return [shape]

    def begin_state(self, batch_size=0, func=zeros, ctx=None, dtype=None,
                    **kwargs):
        states = []
        for shape in self.state_info(batch_size):
            states.append(func(shape, ctx=ctx, dtype=dtype or self._dtype,
                               **kwargs))
        return states

    def forward(self, x, states=None):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        params = self.parameters.data(ctx)
        if getattr(x, 'is_native', False):
            return self._forward_native(x, states, params, ctx)
        t = x._t
        if self._layout == 'NTC':
            t = t.transpose(0, 1).contiguous()
        N = t.shape[1]
        return_states = states is not None
        if states is None:
            # states follow the runtime compute dtype (fp16 after cast)
            states = self.begin_state(N, ctx=ctx, dtype=str(x.dtype)
                                      if hasattr(x, 'dtype') else None)
        if isinstance(states, NDArray):
            states = [states]
        from ... import autograd as _ag
        out, hn, cn = _rnn_ops.rnn_forward(
            t, params._t, states[0]._t,
            states[1]._t if len(states) > 1 else None,
            self._mode, self._hidden_size, self._num_layers,
            self._bidirectional, self._dropout,
            training=_ag.is_training())
        if self._layout == 'NTC':
            out = out.transpose(0, 1).contiguous()
        out = NDArray(out)
        if not return_states:
            return out
        new_states = [NDArray(hn)]
        if cn is not None:
            new_states.append(NDArray(cn))
        return out, new_states

    def _forward_native(self, x, states, params, ctx):
        """Native-runtime RNN: same fused-RNN semantics composed from
        native registry ops (ops/rnn.py rnn_forward_native) — backward
        runs on the own tape, no torch involved."""
        from ... import autograd as _ag
        t = x
        if self._layout == 'NTC':
            t = t.transpose((1, 0, 2))
        N = t.shape[1]
        return_states = states is not None
        if states is None:
            states = self.begin_state(N, ctx=ctx, dtype=str(x.dtype))
        if isinstance(states, NDArray):
            states = [states]
        out, hn, cn = _rnn_ops.rnn_forward_native(
            t, params, states[0],
            states[1] if len(states) > 1 else None,
            self._mode, self._hidden_size, self._num_layers,
            self._bidirectional, self._dropout,
            training=_ag.is_training())
        if self._layout == 'NTC':
            out = out.transpose((1, 0, 2))
        if not return_states:
            return out
        new_states = [hn]
        if cn is not None:
            new_states.append(cn)
        return out, new_states

    def __repr__(self):
        return (f'{type(self).__name__}({self._hidden_size}, '
                f'layers={self._num_layers}, layout={self._layout})')


class RNN(_RNNLayer):
    def __init__(self, hidden_size, num_layers=1, activation='relu', **kwargs):
        super().__init__('rnn_relu' if activation == 'relu' else 'rnn_tanh',
                         hidden_size, num_layers, **kwargs)


class LSTM(_RNNLayer):
    def __init__(self, hidden_size, num_layers=1, **kwargs):
        super().__init__('lstm', hidden_size, num_layers, **kwargs)


class GRU(_RNNLayer):
    def __init__(self, hidden_size, num_layers=1, **kwargs):
        super().__init__('gru', hidden_size, num_layers, **kwargs)
