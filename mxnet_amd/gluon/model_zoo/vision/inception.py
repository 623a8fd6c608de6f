"""Inception V3 (reference gluon/model_zoo/vision/inception.py)."""
import torch

from ...block import HybridBlock
from ... import nn
from ....ndarray.ndarray import NDArray

__all__ = ['Inception3', 'inception_v3']


def _conv(channels, kernel, stride=1, pad=0, layout='NCHW'):
    ax = -1 if layout == 'NHWC' else 1
    out = nn.HybridSequential()
    out.add(nn.Conv2D(channels, kernel, stride, pad, use_bias=False,
                      layout=layout),
            nn.BatchNormReLU(axis=ax, epsilon=0.001))
    return out


class _Branches(HybridBlock):
    """Run branches and concat on the channel axis."""

    def __init__(self, branches, layout, **kwargs):
        super().__init__(**kwargs)
        self._dim = -1 if layout == 'NHWC' else 1
        for i, b in enumerate(branches):
            self.register_child(b, f'b{i}')

    def forward(self, x):
        from ....ndarray.ndarray import concat
        return concat([b(x) for b in self._children.values()],
                      dim=self._dim)


def _seq(*blocks):
    s = nn.HybridSequential()
    s.add(*blocks)
    return s


def _make_A(pool_features, layout):
    return _Branches([
        _conv(64, 1, layout=layout),
        _seq(_conv(48, 1, layout=layout), _conv(64, 5, 1, 2, layout=layout)),
        _seq(_conv(64, 1, layout=layout), _conv(96, 3, 1, 1, layout=layout),
             _conv(96, 3, 1, 1, layout=layout)),
        _seq(nn.AvgPool2D(3, 1, 1, layout=layout),
             _conv(pool_features, 1, layout=layout)),
    ], layout)


def _make_B(layout):
    return _Branches([
        _conv(384, 3, 2, layout=layout),
        _seq(_conv(64, 1, layout=layout), _conv(96, 3, 1, 1, layout=layout),
             _conv(96, 3, 2, layout=layout)),
        _seq(nn.MaxPool2D(3, 2, layout=layout)),
    ], layout)


def _make_C(channels_7x7, layout):
    c = channels_7x7
    return _Branches([
        _conv(192, 1, layout=layout),
        _seq(_conv(c, 1, layout=layout), _conv(c, (1, 7), 1, (0, 3), layout=layout),
             _conv(192, (7, 1), 1, (3, 0), layout=layout)),
        _seq(_conv(c, 1, layout=layout), _conv(c, (7, 1), 1, (3, 0), layout=layout),
             _conv(c, (1, 7), 1, (0, 3), layout=layout),
             _conv(c, (7, 1), 1, (3, 0), layout=layout),
             _conv(192, (1, 7), 1, (0, 3), layout=layout)),
        _seq(nn.AvgPool2D(3, 1, 1, layout=layout), _conv(192, 1, layout=layout)),
    ], layout)


def _make_D(layout):
    return _Branches([
        _seq(_conv(192, 1, layout=layout), _conv(320, 3, 2, layout=layout)),
        _seq(_conv(192, 1, layout=layout),
             _conv(192, (1, 7), 1, (0, 3), layout=layout),
             _conv(192, (7, 1), 1, (3, 0), layout=layout),
             _conv(192, 3, 2, layout=layout)),
        _seq(nn.MaxPool2D(3, 2, layout=layout)),
    ], layout)


def _make_E(layout):
    return _Branches([
        _conv(320, 1, layout=layout),
        _seq(_conv(384, 1, layout=layout),
             _Branches([_conv(384, (1, 3), 1, (0, 1), layout=layout),
                        _conv(384, (3, 1), 1, (1, 0), layout=layout)], layout)),
        _seq(_conv(448, 1, layout=layout), _conv(384, 3, 1, 1, layout=layout),
             _Branches([_conv(384, (1, 3), 1, (0, 1), layout=layout),
                        _conv(384, (3, 1), 1, (1, 0), layout=layout)], layout)),
        _seq(nn.AvgPool2D(3, 1, 1, layout=layout), _conv(192, 1, layout=layout)),
    ], layout)


class Inception3(HybridBlock):
    def __init__(self, classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.features = nn.HybridSequential()
        self.features.add(
            _conv(32, 3, 2, layout=layout),
            _conv(32, 3, layout=layout),
            _conv(64, 3, 1, 1, layout=layout),
            nn.MaxPool2D(3, 2, layout=layout),
            _conv(80, 1, layout=layout),
            _conv(192, 3, layout=layout),
            nn.MaxPool2D(3, 2, layout=layout),
            _make_A(32, layout), _make_A(64, layout), _make_A(64, layout),
            _make_B(layout),
            _make_C(128, layout), _make_C(160, layout), _make_C(160, layout),
            _make_C(192, layout),
            _make_D(layout),
            _make_E(layout), _make_E(layout),
            nn.AvgPool2D(8, layout=layout),
            nn.Dropout(0.5),
            nn.Flatten())
        self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))


def inception_v3(pretrained=False, **kwargs):
    return Inception3(**kwargs)
