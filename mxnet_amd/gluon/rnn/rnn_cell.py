"""RNN cells (reference gluon/rnn/rnn_cell.py).

Every cell composes dual-backend nd ops (FullyConnected on the MFMA
GEMM, Activation, slicing, concat), so the same code runs on the torch
frontend and on the native C++ runtime (own tape)."""
from ..block import HybridBlock
from ..parameter import Parameter
from ...ndarray.ndarray import NDArray, zeros, concat, stack
from ...ndarray import ops as F
from ... import initializer as init


def _sig(x):
    return F.Activation(x, act_type='sigmoid')


def _tanh(x):
    return F.Activation(x, act_type='tanh')


class RecurrentCell(HybridBlock):
    def __init__(self, **kwargs):
        super().__init__(**kwargs)

    def state_info(self, batch_size=0):
        raise NotImplementedError

    def begin_state(self, batch_size=0, func=zeros, ctx=None, dtype='float32',
                    **kwargs):
        return [func(info, ctx=ctx, dtype=dtype, **kwargs)
                for info in self.state_info(batch_size)]

    @staticmethod
    def _split_time(inputs, axis):
        """[.., T, ..] -> list of T arrays with the axis squeezed."""
        T = inputs.shape[axis]
        return [inputs.slice_axis(axis, t, t + 1).squeeze(axis=axis)
                for t in range(T)]

    def unroll(self, length, inputs, begin_state=None, layout='NTC',
               merge_outputs=None):
        axis = layout.find('T')
        if isinstance(inputs, NDArray):
            inputs = self._split_time(inputs, axis)
        batch = inputs[0].shape[0]
        states = begin_state or self.begin_state(
            batch, ctx=inputs[0].context, dtype=str(inputs[0].dtype))
        outputs = []
        for t in range(length):
            out, states = self(inputs[t], states)
            outputs.append(out)
        if merge_outputs:
            outputs = stack(outputs, axis=axis)
        return outputs, states


class RNNCell(RecurrentCell):
    def __init__(self, hidden_size, activation='tanh', input_size=0, **kwargs):
        super().__init__(**kwargs)
        self._hidden_size = hidden_size
        self._activation = activation
        self.i2h_weight = Parameter('i2h_weight', shape=(hidden_size, input_size),
                                    allow_deferred_init=True)
        self.h2h_weight = Parameter('h2h_weight', shape=(hidden_size, hidden_size))
        self.i2h_bias = Parameter('i2h_bias', shape=(hidden_size,), init=init.Zero())
        self.h2h_bias = Parameter('h2h_bias', shape=(hidden_size,), init=init.Zero())

    def state_info(self, batch_size=0):
        return [(batch_size, self._hidden_size)]

    def infer_shape(self, x, *a):
        self.i2h_weight.shape = (self._hidden_size, x.shape[-1])

    def forward(self, x, states):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        pre = F.FullyConnected(x, self.i2h_weight.data(ctx),
                               self.i2h_bias.data(ctx), flatten=False) + \
            F.FullyConnected(states[0], self.h2h_weight.data(ctx),
                             self.h2h_bias.data(ctx), flatten=False)
        out = _tanh(pre) if self._activation == 'tanh' \
            else F.Activation(pre, act_type=self._activation)
        return out, [out]


class LSTMCell(RecurrentCell):
    def __init__(self, hidden_size, input_size=0, **kwargs):
        super().__init__(**kwargs)
        self._hidden_size = hidden_size
        self.i2h_weight = Parameter('i2h_weight',
                                    shape=(4 * hidden_size, input_size),
                                    allow_deferred_init=True)
        self.h2h_weight = Parameter('h2h_weight',
                                    shape=(4 * hidden_size, hidden_size))
        self.i2h_bias = Parameter('i2h_bias', shape=(4 * hidden_size,),
                                  init=init.Zero())
        self.h2h_bias = Parameter('h2h_bias', shape=(4 * hidden_size,),
                                  init=init.Zero())

    def state_info(self, batch_size=0):
        return [(batch_size, self._hidden_size),
                (batch_size, self._hidden_size)]

    def infer_shape(self, x, *a):
        self.i2h_weight.shape = (4 * self._hidden_size, x.shape[-1])

    def forward(self, x, states):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        h, c = states
        gates = F.FullyConnected(x, self.i2h_weight.data(ctx),
                                 self.i2h_bias.data(ctx), flatten=False) + \
            F.FullyConnected(h, self.h2h_weight.data(ctx),
                             self.h2h_bias.data(ctx), flatten=False)
        H = self._hidden_size
        i = _sig(gates[:, 0:H])
        f = _sig(gates[:, H:2 * H])
        g = _tanh(gates[:, 2 * H:3 * H])
        o = _sig(gates[:, 3 * H:4 * H])
        c_new = f * c + i * g
        h_new = o * _tanh(c_new)
        return h_new, [h_new, c_new]


class GRUCell(RecurrentCell):
    def __init__(self, hidden_size, input_size=0, **kwargs):
        super().__init__(**kwargs)
        self._hidden_size = hidden_size
        self.i2h_weight = Parameter('i2h_weight',
                                    shape=(3 * hidden_size, input_size),
                                    allow_deferred_init=True)
        self.h2h_weight = Parameter('h2h_weight',
                                    shape=(3 * hidden_size, hidden_size))
        self.i2h_bias = Parameter('i2h_bias', shape=(3 * hidden_size,),
                                  init=init.Zero())
        self.h2h_bias = Parameter('h2h_bias', shape=(3 * hidden_size,),
                                  init=init.Zero())

    def state_info(self, batch_size=0):
        return [(batch_size, self._hidden_size)]

    def infer_shape(self, x, *a):
        self.i2h_weight.shape = (3 * self._hidden_size, x.shape[-1])

    def forward(self, x, states):
        self._finish_deferred(x)
        ctx = self._param_ctx((x,))
        h = states[0]
        xg = F.FullyConnected(x, self.i2h_weight.data(ctx),
                              self.i2h_bias.data(ctx), flatten=False)
        hg = F.FullyConnected(h, self.h2h_weight.data(ctx),
                              self.h2h_bias.data(ctx), flatten=False)
        H = self._hidden_size
        r = _sig(xg[:, 0:H] + hg[:, 0:H])
        z = _sig(xg[:, H:2 * H] + hg[:, H:2 * H])
        n = _tanh(xg[:, 2 * H:3 * H] + r * hg[:, 2 * H:3 * H])
        out = (z * -1.0 + 1.0) * n + z * h
        return out, [out]


class SequentialRNNCell(RecurrentCell):
    def __init__(self, **kwargs):
        super().__init__(**kwargs)

    def add(self, cell):
        self.register_child(cell)

    def state_info(self, batch_size=0):
        info = []
        for c in self._children.values():
            info.extend(c.state_info(batch_size))
        return info

    def forward(self, x, states):
        next_states = []
        p = 0
        for cell in self._children.values():
            n = len(cell.state_info())
            x, s = cell(x, states[p:p + n])
            next_states.extend(s)
            p += n
        return x, next_states


class DropoutCell(RecurrentCell):
    def __init__(self, rate, **kwargs):
        super().__init__(**kwargs)
        self._rate = rate

    def state_info(self, batch_size=0):
        return []

    def forward(self, x, states):
        from ... import autograd as _ag
        if self._rate > 0 and _ag.is_training():
            x = F.Dropout(x, p=self._rate)
        return x, states


class ZoneoutCell(RecurrentCell):
    def __init__(self, base_cell, zoneout_outputs=0.0, zoneout_states=0.0,
                 **kwargs):
        super().__init__(**kwargs)
        self.base_cell = base_cell
        self._zo, self._zs = zoneout_outputs, zoneout_states
        self._prev = None

    def state_info(self, batch_size=0):
        return self.base_cell.state_info(batch_size)

    def forward(self, x, states):
        out, next_states = self.base_cell(x, states)
        from ... import autograd as _ag
        if _ag.is_training():
            import numpy as _np

            def mix(new, old, p):
                if p == 0:
                    return new
                # bernoulli keep-old mask sampled host-side (parity with
                # reference zoneout; mask is not differentiated through)
                from ...ndarray.ndarray import array as _arr
                m = (_np.random.rand(*new.shape) < p).astype('float32')
                mask = _arr(m, ctx=new.context)
                if str(mask.dtype) != str(new.dtype):
                    mask = mask.astype(new.dtype)
                return mask * old + (mask * -1.0 + 1.0) * new
            out = mix(out, states[0], self._zo)
            next_states = [mix(n, o, self._zs)
                           for n, o in zip(next_states, states)]
        return out, next_states


class ResidualCell(RecurrentCell):
    def __init__(self, base_cell, **kwargs):
        super().__init__(**kwargs)
        self.base_cell = base_cell

    def state_info(self, batch_size=0):
        return self.base_cell.state_info(batch_size)

    def forward(self, x, states):
        out, next_states = self.base_cell(x, states)
        return out + x, next_states


class BidirectionalCell(RecurrentCell):
    def __init__(self, l_cell, r_cell, **kwargs):
        super().__init__(**kwargs)
        self.l_cell = l_cell
        self.r_cell = r_cell

    def state_info(self, batch_size=0):
        return (self.l_cell.state_info(batch_size)
                + self.r_cell.state_info(batch_size))

    def unroll(self, length, inputs, begin_state=None, layout='NTC',
               merge_outputs=None):
        axis = layout.find('T')
        if isinstance(inputs, NDArray):
            inputs = self._split_time(inputs, axis)
        batch = inputs[0].shape[0]
        states = begin_state or self.begin_state(
            batch, ctx=inputs[0].context, dtype=str(inputs[0].dtype))
        nl = len(self.l_cell.state_info())
        l_states, r_states = states[:nl], states[nl:]
        l_out, r_out = [], []
        for t in range(length):
            o, l_states = self.l_cell(inputs[t], l_states)
            l_out.append(o)
        for t in reversed(range(length)):
            o, r_states = self.r_cell(inputs[t], r_states)
            r_out.append(o)
        r_out.reverse()
        outputs = [concat([l, r], dim=-1) for l, r in zip(l_out, r_out)]
        if merge_outputs:
            outputs = stack(outputs, axis=axis)
        return outputs, l_states + r_states
