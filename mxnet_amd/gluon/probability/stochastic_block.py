"""StochasticBlock (reference gluon/probability/block/stochastic_block.py):
a HybridBlock that can accumulate intermediate loss terms (e.g. KL
penalties from sampled latents) during forward.
"""
from ..block import HybridBlock
from ..nn import HybridSequential

__all__ = ['StochasticBlock', 'StochasticSequential']


class StochasticBlock(HybridBlock):
    def __init__(self, **kwargs):
        super().__init__(**kwargs)
        self._losses = []
        self._losscache = []

    @property
    def losses(self):
        return self._losses

    def add_loss(self, loss):
        self._losscache.append(loss)

    @staticmethod
    def collectLoss(forward_fn):
        """Decorator marking the forward whose losses are collected
        (reference stochastic_block.py:47)."""
        def wrapped(self, *args, **kwargs):
            self._losscache = []
            out = forward_fn(self, *args, **kwargs)
            self._losses = self._losscache
            return out
        return wrapped

    def __call__(self, *args, **kwargs):
        return super().__call__(*args, **kwargs)


class StochasticSequential(StochasticBlock):
    """Sequential container collecting child StochasticBlock losses."""

    def __init__(self, **kwargs):
        super().__init__(**kwargs)
        self._layers = []

    def add(self, *blocks):
        for b in blocks:
            self._layers.append(b)
            setattr(self, f'_layer{len(self._layers) - 1}', b)

    def forward(self, x):
        self._losses = []
        for layer in self._layers:
            x = layer(x)
            if isinstance(layer, StochasticBlock):
                self._losses.extend(layer.losses)
        return x
