"""DenseNet (reference gluon/model_zoo/vision/densenet.py)."""
import torch

from ...block import HybridBlock
from ... import nn
from ....ndarray.ndarray import NDArray

__all__ = ['DenseNet', 'densenet121', 'densenet161', 'densenet169',
           'densenet201']


class _DenseLayer(HybridBlock):
    def __init__(self, growth_rate, bn_size, dropout, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        self._dim = -1 if layout == 'NHWC' else 1
        self.body = nn.HybridSequential()
        self.body.add(nn.BatchNormReLU(axis=ax),
                      nn.Conv2D(bn_size * growth_rate, kernel_size=1,
                                use_bias=False, layout=layout),
                      nn.BatchNormReLU(axis=ax),
                      nn.Conv2D(growth_rate, kernel_size=3, padding=1,
                                use_bias=False, layout=layout))
        if dropout:
            self.body.add(nn.Dropout(dropout))

    def forward(self, x):
        out = self.body(x)
        from ....ndarray.ndarray import concat
        return concat([x, out], dim=self._dim)


def _make_transition(num_out, layout):
    ax = -1 if layout == 'NHWC' else 1
    out = nn.HybridSequential()
    out.add(nn.BatchNormReLU(axis=ax),
            nn.Conv2D(num_out, kernel_size=1, use_bias=False, layout=layout),
            nn.AvgPool2D(pool_size=2, strides=2, layout=layout))
    return out


class DenseNet(HybridBlock):
    def __init__(self, num_init_features, growth_rate, block_config,
                 bn_size=4, dropout=0, classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        self.features = nn.HybridSequential()
        self.features.add(nn.Conv2D(num_init_features, 7, 2, 3,
                                    use_bias=False, layout=layout),
                          nn.BatchNormReLU(axis=ax),
                          nn.MaxPool2D(3, 2, 1, layout=layout))
        num_features = num_init_features
        for i, num_layers in enumerate(block_config):
            block = nn.HybridSequential()
            for _ in range(num_layers):
                block.add(_DenseLayer(growth_rate, bn_size, dropout, layout))
            self.features.add(block)
            num_features += num_layers * growth_rate
            if i != len(block_config) - 1:
                num_features //= 2
                self.features.add(_make_transition(num_features, layout))
        self.features.add(nn.BatchNormReLU(axis=ax),
                          nn.GlobalAvgPool2D(layout=layout),
                          nn.Flatten())
        self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))


densenet_spec = {
    121: (64, 32, [6, 12, 24, 16]),
    161: (96, 48, [6, 12, 36, 24]),
    169: (64, 32, [6, 12, 32, 32]),
    201: (64, 32, [6, 12, 48, 32]),
}


def _densenet(n, **kwargs):
    f, g, cfg = densenet_spec[n]
    return DenseNet(f, g, cfg, **kwargs)


def densenet121(**kw):
    return _densenet(121, **kw)


def densenet161(**kw):
    return _densenet(161, **kw)


def densenet169(**kw):
    return _densenet(169, **kw)


def densenet201(**kw):
    return _densenet(201, **kw)
