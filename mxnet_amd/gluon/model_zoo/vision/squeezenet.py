"""SqueezeNet (reference gluon/model_zoo/vision/squeezenet.py)."""
import torch

from ...block import HybridBlock
from ... import nn
from ....ndarray.ndarray import NDArray

__all__ = ['SqueezeNet', 'squeezenet1_0', 'squeezenet1_1']


class _Fire(HybridBlock):
    def __init__(self, squeeze, expand1x1, expand3x3, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self._dim = -1 if layout == 'NHWC' else 1
        self.squeeze = nn.Conv2D(squeeze, kernel_size=1, activation='relu',
                                 layout=layout)
        self.expand1 = nn.Conv2D(expand1x1, kernel_size=1, activation='relu',
                                 layout=layout)
        self.expand3 = nn.Conv2D(expand3x3, kernel_size=3, padding=1,
                                 activation='relu', layout=layout)

    def forward(self, x):
        x = self.squeeze(x)
        from ....ndarray.ndarray import concat
        return concat([self.expand1(x), self.expand3(x)], dim=self._dim)


class SqueezeNet(HybridBlock):
    def __init__(self, version='1.0', classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.features = nn.HybridSequential()
        if version == '1.0':
            self.features.add(
                nn.Conv2D(96, kernel_size=7, strides=2, activation='relu',
                          layout=layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(16, 64, 64, layout), _Fire(16, 64, 64, layout),
                _Fire(32, 128, 128, layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(32, 128, 128, layout), _Fire(48, 192, 192, layout),
                _Fire(48, 192, 192, layout), _Fire(64, 256, 256, layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(64, 256, 256, layout))
        else:
            self.features.add(
                nn.Conv2D(64, kernel_size=3, strides=2, activation='relu',
                          layout=layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(16, 64, 64, layout), _Fire(16, 64, 64, layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(32, 128, 128, layout), _Fire(32, 128, 128, layout),
                nn.MaxPool2D(3, 2, ceil_mode=True, layout=layout),
                _Fire(48, 192, 192, layout), _Fire(48, 192, 192, layout),
                _Fire(64, 256, 256, layout), _Fire(64, 256, 256, layout))
        self.features.add(nn.Dropout(0.5))
        self.output = nn.HybridSequential()
        self.output.add(nn.Conv2D(classes, kernel_size=1, activation='relu',
                                  layout=layout),
                        nn.GlobalAvgPool2D(layout=layout),
                        nn.Flatten())

    def forward(self, x):
        return self.output(self.features(x))


def squeezenet1_0(pretrained=False, **kwargs):
    return SqueezeNet('1.0', **kwargs)


def squeezenet1_1(pretrained=False, **kwargs):
    return SqueezeNet('1.1', **kwargs)
