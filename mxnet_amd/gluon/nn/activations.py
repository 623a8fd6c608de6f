"""Advanced activation layers (reference gluon/nn/activations.py)."""
from ..block import HybridBlock
from ..parameter import Parameter
from ... import initializer as init


class LeakyReLU(HybridBlock):
    def __init__(self, alpha=0.01, **kwargs):
        super().__init__(**kwargs)
        self._alpha = alpha

    def hybrid_forward(self, F, x):
        return F.LeakyReLU(x, act_type='leaky', slope=self._alpha)


class PReLU(HybridBlock):
    def __init__(self, alpha_initializer=init.Constant(0.25), in_channels=1,
                 **kwargs):
        super().__init__(**kwargs)
        self.alpha = Parameter('alpha', shape=(in_channels,),
                               init=alpha_initializer)

    def hybrid_forward(self, F, x, alpha):
        if getattr(x, 'is_native', False):
            # prelu(x) = relu(x) - alpha * relu(-x); alpha broadcasts
            # over the channel axis (axis 1 for >1 channels)
            pos = F.Activation(x, act_type='relu')
            neg = F.Activation(x * -1.0, act_type='relu')
            if alpha.size > 1 and len(x.shape) > 1:
                ash = [1] * len(x.shape)
                ash[1] = alpha.size
                alpha = alpha.reshape(ash)
            return pos - neg * alpha
        import torch
        from ...ndarray.ndarray import NDArray
        return NDArray(torch.nn.functional.prelu(x._t, alpha._t))


class ELU(HybridBlock):
    def __init__(self, alpha=1.0, **kwargs):
        super().__init__(**kwargs)
        self._alpha = alpha

    def hybrid_forward(self, F, x):
        return F.LeakyReLU(x, act_type='elu', slope=self._alpha)


class SELU(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.LeakyReLU(x, act_type='selu')


class GELU(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.LeakyReLU(x, act_type='gelu')


class Swish(HybridBlock):
    def __init__(self, beta=1.0, **kwargs):
        super().__init__(**kwargs)

    def hybrid_forward(self, F, x):
        return F.Activation(x, act_type='silu')


class SiLU(Swish):
    pass
