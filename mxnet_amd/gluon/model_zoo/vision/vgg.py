"""VGG (reference gluon/model_zoo/vision/vgg.py)."""
from .... import initializer as init
from ...block import HybridBlock
from ... import nn

__all__ = ['VGG', 'vgg11', 'vgg13', 'vgg16', 'vgg19',
           'vgg11_bn', 'vgg13_bn', 'vgg16_bn', 'vgg19_bn']

vgg_spec = {
    11: ([1, 1, 2, 2, 2], [64, 128, 256, 512, 512]),
    13: ([2, 2, 2, 2, 2], [64, 128, 256, 512, 512]),
    16: ([2, 2, 3, 3, 3], [64, 128, 256, 512, 512]),
    19: ([2, 2, 4, 4, 4], [64, 128, 256, 512, 512]),
}


class VGG(HybridBlock):
    def __init__(self, layers, filters, classes=1000, batch_norm=False,
                 layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        self.features = nn.HybridSequential()
        for i, num in enumerate(layers):
            for _ in range(num):
                self.features.add(nn.Conv2D(
                    filters[i], kernel_size=3, padding=1, layout=layout,
                    use_bias=not batch_norm,
                    weight_initializer=init.Xavier(
                        rnd_type='gaussian', factor_type='out', magnitude=2)))
                if batch_norm:
                    self.features.add(nn.BatchNormReLU(axis=ax))
                else:
                    self.features.add(nn.Activation('relu'))
            self.features.add(nn.MaxPool2D(strides=2, layout=layout))
        self.features.add(
            nn.Flatten(),
            nn.Dense(4096, activation='relu',
                     weight_initializer=init.Normal(0.01)), nn.Dropout(0.5),
            nn.Dense(4096, activation='relu',
                     weight_initializer=init.Normal(0.01)), nn.Dropout(0.5))
        self.output = nn.Dense(classes, weight_initializer=init.Normal(0.01))

    def forward(self, x):
        return self.output(self.features(x))


def _vgg(num_layers, batch_norm=False, pretrained=False, **kwargs):
    if pretrained:
        raise RuntimeError('no network for pretrained weights')
    layers, filters = vgg_spec[num_layers]
    return VGG(layers, filters, batch_norm=batch_norm, **kwargs)


def vgg11(**kw):
    return _vgg(11, **kw)


def vgg13(**kw):
    return _vgg(13, **kw)


def vgg16(**kw):
    return _vgg(16, **kw)


def vgg19(**kw):
    return _vgg(19, **kw)


def vgg11_bn(**kw):
    return _vgg(11, batch_norm=True, **kw)


def vgg13_bn(**kw):
    return _vgg(13, batch_norm=True, **kw)


def vgg16_bn(**kw):
    return _vgg(16, batch_norm=True, **kw)


def vgg19_bn(**kw):
    return _vgg(19, batch_norm=True, **kw)
