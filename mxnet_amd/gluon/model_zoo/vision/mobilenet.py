"""MobileNet V1/V2 (reference gluon/model_zoo/vision/mobilenet.py)."""
from ...block import HybridBlock
from ... import nn

__all__ = ['MobileNet', 'MobileNetV2', 'mobilenet1_0', 'mobilenet0_75',
           'mobilenet0_5', 'mobilenet0_25', 'mobilenet_v2_1_0']


def _add_conv(out, channels, kernel=1, stride=1, pad=0, num_group=1,
              active=True, layout='NCHW'):
    ax = -1 if layout == 'NHWC' else 1
    out.add(nn.Conv2D(channels, kernel, stride, pad, groups=num_group,
                      use_bias=False, layout=layout))
    if active:
        out.add(nn.BatchNormReLU(axis=ax))
    else:
        out.add(nn.BatchNorm(axis=ax))


class MobileNet(HybridBlock):
    def __init__(self, multiplier=1.0, classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.features = nn.HybridSequential()
        ch = [int(multiplier * c) for c in
              [32, 64, 128, 128, 256, 256, 512, 512, 512, 512, 512, 512,
               1024, 1024]]
        strides = [1, 2, 1, 2, 1, 2, 1, 1, 1, 1, 1, 2, 1]
        _add_conv(self.features, ch[0], 3, 2, 1, layout=layout)
        prev = ch[0]
        for c, s in zip(ch[1:], strides):
            _add_conv(self.features, prev, 3, s, 1, num_group=prev,
                      layout=layout)
            _add_conv(self.features, c, 1, 1, 0, layout=layout)
            prev = c
        self.features.add(nn.GlobalAvgPool2D(layout=layout), nn.Flatten())
        self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))


class _InvertedResidual(HybridBlock):
    def __init__(self, in_c, out_c, stride, expand, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self._same = stride == 1 and in_c == out_c
        mid = in_c * expand
        self.body = nn.HybridSequential()
        if expand != 1:
            _add_conv(self.body, mid, 1, layout=layout)
        _add_conv(self.body, mid, 3, stride, 1, num_group=mid, layout=layout)
        _add_conv(self.body, out_c, 1, active=False, layout=layout)

    def forward(self, x):
        out = self.body(x)
        return out + x if self._same else out


class MobileNetV2(HybridBlock):
    def __init__(self, multiplier=1.0, classes=1000, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.features = nn.HybridSequential()
        first = int(32 * multiplier)
        _add_conv(self.features, first, 3, 2, 1, layout=layout)
        cfg = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
               (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]
        in_c = first
        for t, c, n, s in cfg:
            out_c = int(c * multiplier)
            for i in range(n):
                self.features.add(_InvertedResidual(
                    in_c, out_c, s if i == 0 else 1, t, layout))
                in_c = out_c
        last = int(1280 * max(1.0, multiplier))
        _add_conv(self.features, last, 1, layout=layout)
        self.features.add(nn.GlobalAvgPool2D(layout=layout), nn.Flatten())
        self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))


def mobilenet1_0(**kw):
    return MobileNet(1.0, **kw)


def mobilenet0_75(**kw):
    return MobileNet(0.75, **kw)


def mobilenet0_5(**kw):
    return MobileNet(0.5, **kw)


def mobilenet0_25(**kw):
    return MobileNet(0.25, **kw)


def mobilenet_v2_1_0(**kw):
    return MobileNetV2(1.0, **kw)
