"""AlexNet (reference gluon/model_zoo/vision/alexnet.py)."""
from .... import initializer as init
from ...block import HybridBlock
from ... import nn

__all__ = ['AlexNet', 'alexnet']


class AlexNet(HybridBlock):
    def __init__(self, classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.features = nn.HybridSequential()
        xav = init.Xavier(rnd_type='gaussian', factor_type='out',
                          magnitude=2)
        self.features.add(
            nn.Conv2D(64, kernel_size=11, strides=4, padding=2,
                      activation='relu', layout=layout,
                      weight_initializer=xav),
            nn.MaxPool2D(pool_size=3, strides=2, layout=layout),
            nn.Conv2D(192, kernel_size=5, padding=2, activation='relu',
                      layout=layout, weight_initializer=xav),
            nn.MaxPool2D(pool_size=3, strides=2, layout=layout),
            nn.Conv2D(384, kernel_size=3, padding=1, activation='relu',
                      layout=layout, weight_initializer=xav),
            nn.Conv2D(256, kernel_size=3, padding=1, activation='relu',
                      layout=layout, weight_initializer=xav),
            nn.Conv2D(256, kernel_size=3, padding=1, activation='relu',
                      layout=layout, weight_initializer=xav),
            nn.MaxPool2D(pool_size=3, strides=2, layout=layout),
            nn.Flatten(),
            nn.Dense(4096, activation='relu',
                     weight_initializer=init.Normal(0.01)),
            nn.Dropout(0.5),
            nn.Dense(4096, activation='relu',
                     weight_initializer=init.Normal(0.01)),
            nn.Dropout(0.5),
        )
        self.output = nn.Dense(classes, weight_initializer=init.Normal(0.01))

    def forward(self, x):
        return self.output(self.features(x))


def alexnet(pretrained=False, **kwargs):
    if pretrained:
        raise RuntimeError('no network for pretrained weights')
    return AlexNet(**kwargs)
