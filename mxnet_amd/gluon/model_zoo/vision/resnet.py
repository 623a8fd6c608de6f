"""ResNet V1/V1.5/V2 (reference gluon/model_zoo/vision/resnet.py).

The MI355X hot path builds with ``layout='NHWC'`` so every Conv/BN/Pool
runs the channel-contiguous MFMA/HIP kernels; BN+ReLU and BN+add+ReLU are
fused into the BatchNorm kernel (BatchNormReLU / residual input) — the
fusion the reference gets from its pointwise-fusion graph pass
(SURVEY §2.2 fusion row) is structural here.

ResNet-50 v1.5 = bottleneck with the stride on the 3x3 conv (the
BASELINE.json benchmark model).
"""
from ...block import HybridBlock
from ... import nn
from .... import initializer as init

__all__ = ['ResNetV1', 'ResNetV2', 'BasicBlockV1', 'BasicBlockV2',
           'BottleneckV1', 'BottleneckV2', 'ResNext', 'ResNextBlock',
           'resnet18_v1', 'resnet34_v1', 'resnet50_v1', 'resnet101_v1',
           'resnet152_v1', 'resnet18_v2', 'resnet34_v2', 'resnet50_v2',
           'resnet101_v2', 'resnet152_v2', 'resnext50_32x4d',
           'resnext101_32x4d', 'get_resnet']


def _conv3x3(channels, stride, in_channels, layout):
    return nn.Conv2D(channels, kernel_size=3, strides=stride, padding=1,
                     use_bias=False, in_channels=in_channels, layout=layout)


class BasicBlockV1(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self.conv1 = _conv3x3(channels, stride, in_channels, layout)
        self.bn1 = nn.BatchNormReLU(axis=-1 if layout == 'NHWC' else 1)
        self.conv2 = _conv3x3(channels, 1, channels, layout)
        self.bn2 = nn.BatchNorm(axis=-1 if layout == 'NHWC' else 1,
                                fuse_relu=True)
        if downsample:
            self.ds_conv = nn.Conv2D(channels, kernel_size=1, strides=stride,
                                     use_bias=False, in_channels=in_channels,
                                     layout=layout)
            self.ds_bn = nn.BatchNorm(axis=-1 if layout == 'NHWC' else 1)
        else:
            self.ds_conv = None

    def forward(self, x):
        residual = x
        out = self.bn1(self.conv1(x))
        out = self.conv2(out)
        if self.ds_conv is not None:
            residual = self.ds_bn(self.ds_conv(x))
        # fused add + BN + relu
        return self.bn2(out, residual)


class BottleneckV1(HybridBlock):
    """v1.5 bottleneck: stride lives on the 3x3."""

    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        mid = channels // 4
        self.conv1 = nn.Conv2D(mid, kernel_size=1, strides=1, use_bias=False,
                               in_channels=in_channels, layout=layout)
        self.bn1 = nn.BatchNormReLU(axis=ax)
        self.conv2 = _conv3x3(mid, stride, mid, layout)
        self.bn2 = nn.BatchNormReLU(axis=ax)
        self.conv3 = nn.Conv2D(channels, kernel_size=1, strides=1,
                               use_bias=False, in_channels=mid, layout=layout)
        self.bn3 = nn.BatchNorm(axis=ax, fuse_relu=True)
        if downsample:
            self.ds_conv = nn.Conv2D(channels, kernel_size=1, strides=stride,
                                     use_bias=False, in_channels=in_channels,
                                     layout=layout)
            self.ds_bn = nn.BatchNorm(axis=ax)
        else:
            self.ds_conv = None

    def forward(self, x):
        residual = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.conv3(out)
        if self.ds_conv is not None:
            residual = self.ds_bn(self.ds_conv(x))
        return self.bn3(out, residual)


class BasicBlockV2(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        self.bn1 = nn.BatchNormReLU(axis=ax)
        self.conv1 = _conv3x3(channels, stride, in_channels, layout)
        self.bn2 = nn.BatchNormReLU(axis=ax)
        self.conv2 = _conv3x3(channels, 1, channels, layout)
        if downsample:
            self.downsample = nn.Conv2D(channels, 1, stride, use_bias=False,
                                        in_channels=in_channels, layout=layout)
        else:
            self.downsample = None

    def forward(self, x):
        residual = x
        x = self.bn1(x)
        if self.downsample is not None:
            residual = self.downsample(x)
        x = self.conv1(x)
        x = self.bn2(x)
        x = self.conv2(x)
        return x + residual


class BottleneckV2(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        mid = channels // 4
        self.bn1 = nn.BatchNormReLU(axis=ax)
        self.conv1 = nn.Conv2D(mid, 1, 1, use_bias=False,
                               in_channels=in_channels, layout=layout)
        self.bn2 = nn.BatchNormReLU(axis=ax)
        self.conv2 = _conv3x3(mid, stride, mid, layout)
        self.bn3 = nn.BatchNormReLU(axis=ax)
        self.conv3 = nn.Conv2D(channels, 1, 1, use_bias=False,
                               in_channels=mid, layout=layout)
        if downsample:
            self.downsample = nn.Conv2D(channels, 1, stride, use_bias=False,
                                        in_channels=in_channels, layout=layout)
        else:
            self.downsample = None

    def forward(self, x):
        residual = x
        x = self.bn1(x)
        if self.downsample is not None:
            residual = self.downsample(x)
        x = self.conv1(x)
        x = self.bn2(x)
        x = self.conv2(x)
        x = self.bn3(x)
        x = self.conv3(x)
        return x + residual


class _ResNetBase(HybridBlock):
    def __init__(self, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        self._layout = layout

    def _maybe_to_layout(self, x):
        # Accept NCHW input even in NHWC mode (benchmark feeds native layout)
        if self._layout == 'NHWC' and x.shape[-1] not in (1, 3, 4):
            x = x.transpose((0, 2, 3, 1))
        return x


class ResNetV1(_ResNetBase):
    def __init__(self, block, layers, channels, classes=1000, thumbnail=False,
                 layout='NCHW', **kwargs):
        super().__init__(layout=layout, **kwargs)
        assert len(layers) == len(channels) - 1
        ax = -1 if layout == 'NHWC' else 1
        self.features = nn.HybridSequential()
        if thumbnail:
            self.features.add(_conv3x3(channels[0], 1, 0, layout))
        else:
            self.features.add(nn.Conv2D(channels[0], 7, 2, 3, use_bias=False,
                                        layout=layout))
            self.features.add(nn.BatchNormReLU(axis=ax))
            self.features.add(nn.MaxPool2D(3, 2, 1, layout=layout))
        in_c = channels[0]
        for i, num_layer in enumerate(layers):
            stride = 1 if i == 0 else 2
            self.features.add(self._make_layer(
                block, num_layer, channels[i + 1], stride, in_c, layout))
            in_c = channels[i + 1]
        self.features.add(nn.GlobalAvgPool2D(layout=layout))
        self.output = nn.Dense(classes, in_units=in_c)

    def _make_layer(self, block, num_layers, channels, stride, in_channels,
                    layout):
        layer = nn.HybridSequential()
        layer.add(block(channels, stride, channels != in_channels,
                        in_channels=in_channels, layout=layout))
        for _ in range(num_layers - 1):
            layer.add(block(channels, 1, False, in_channels=channels,
                            layout=layout))
        return layer

    def forward(self, x):
        x = self._maybe_to_layout(x)
        x = self.features(x)
        x = x.flatten()
        return self.output(x)


class ResNetV2(_ResNetBase):
    def __init__(self, block, layers, channels, classes=1000, thumbnail=False,
                 layout='NCHW', **kwargs):
        super().__init__(layout=layout, **kwargs)
        ax = -1 if layout == 'NHWC' else 1
        self.features = nn.HybridSequential()
        self.features.add(nn.BatchNorm(axis=ax, scale=False, center=False))
        if thumbnail:
            self.features.add(_conv3x3(channels[0], 1, 0, layout))
        else:
            self.features.add(nn.Conv2D(channels[0], 7, 2, 3, use_bias=False,
                                        layout=layout))
            self.features.add(nn.BatchNormReLU(axis=ax))
            self.features.add(nn.MaxPool2D(3, 2, 1, layout=layout))
        in_c = channels[0]
        for i, num_layer in enumerate(layers):
            stride = 1 if i == 0 else 2
            layer = nn.HybridSequential()
            layer.add(block(channels[i + 1], stride,
                            channels[i + 1] != in_c, in_channels=in_c,
                            layout=layout))
            for _ in range(num_layer - 1):
                layer.add(block(channels[i + 1], 1, False,
                                in_channels=channels[i + 1], layout=layout))
            self.features.add(layer)
            in_c = channels[i + 1]
        self.features.add(nn.BatchNormReLU(axis=ax))
        self.features.add(nn.GlobalAvgPool2D(layout=layout))
        self.output = nn.Dense(classes, in_units=in_c)

    def forward(self, x):
        x = self._maybe_to_layout(x)
        x = self.features(x)
        x = x.flatten()
        return self.output(x)


resnet_spec = {
    18: ('basic_block', [2, 2, 2, 2], [64, 64, 128, 256, 512]),
    34: ('basic_block', [3, 4, 6, 3], [64, 64, 128, 256, 512]),
    50: ('bottle_neck', [3, 4, 6, 3], [64, 256, 512, 1024, 2048]),
    101: ('bottle_neck', [3, 4, 23, 3], [64, 256, 512, 1024, 2048]),
    152: ('bottle_neck', [3, 8, 36, 3], [64, 256, 512, 1024, 2048]),
}
resnet_net_versions = [ResNetV1, ResNetV2]
resnet_block_versions = [
    {'basic_block': BasicBlockV1, 'bottle_neck': BottleneckV1},
    {'basic_block': BasicBlockV2, 'bottle_neck': BottleneckV2},
]


def get_resnet(version, num_layers, pretrained=False, ctx=None, **kwargs):
    block_type, layers, channels = resnet_spec[num_layers]
    net = resnet_net_versions[version - 1](
        resnet_block_versions[version - 1][block_type], layers, channels,
        **kwargs)
    if pretrained:
        raise RuntimeError('no network available for pretrained weights')
    return net


def resnet18_v1(**kwargs):
    return get_resnet(1, 18, **kwargs)


def resnet34_v1(**kwargs):
    return get_resnet(1, 34, **kwargs)


def resnet50_v1(**kwargs):
    return get_resnet(1, 50, **kwargs)


def resnet101_v1(**kwargs):
    return get_resnet(1, 101, **kwargs)


def resnet152_v1(**kwargs):
    return get_resnet(1, 152, **kwargs)


def resnet18_v2(**kwargs):
    return get_resnet(2, 18, **kwargs)


def resnet34_v2(**kwargs):
    return get_resnet(2, 34, **kwargs)


def resnet50_v2(**kwargs):
    return get_resnet(2, 50, **kwargs)


def resnet101_v2(**kwargs):
    return get_resnet(2, 101, **kwargs)


def resnet152_v2(**kwargs):
    return get_resnet(2, 152, **kwargs)


class ResNextBlock(HybridBlock):
    """ResNeXt bottleneck (reference resnext.py): grouped 3x3 conv —
    runs the per-group implicit-GEMM kernels on gfx950."""

    def __init__(self, channels, cardinality, bottleneck_width, stride,
                 downsample=False, in_channels=0, layout='NCHW', **kwargs):
        super().__init__(**kwargs)
        ax = -1 if layout == 'NHWC' else 1
        D = int(channels * bottleneck_width / 64) * cardinality // 2
        group_width = max(D, cardinality)
        self.conv1 = nn.Conv2D(group_width, kernel_size=1, use_bias=False,
                               in_channels=in_channels, layout=layout)
        self.bn1 = nn.BatchNormReLU(axis=ax)
        self.conv2 = nn.Conv2D(group_width, kernel_size=3, strides=stride,
                               padding=1, groups=cardinality, use_bias=False,
                               in_channels=group_width, layout=layout)
        self.bn2 = nn.BatchNormReLU(axis=ax)
        self.conv3 = nn.Conv2D(channels, kernel_size=1, use_bias=False,
                               in_channels=group_width, layout=layout)
        self.bn3 = nn.BatchNorm(axis=ax, fuse_relu=True)
        if downsample:
            self.ds_conv = nn.Conv2D(channels, kernel_size=1, strides=stride,
                                     use_bias=False, in_channels=in_channels,
                                     layout=layout)
            self.ds_bn = nn.BatchNorm(axis=ax)
        else:
            self.ds_conv = None

    def forward(self, x):
        residual = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.conv3(out)
        if self.ds_conv is not None:
            residual = self.ds_bn(self.ds_conv(x))
        return self.bn3(out, residual)


class ResNext(_ResNetBase):
    """ResNeXt (reference gluon/model_zoo/vision/resnext.py)."""

    def __init__(self, layers, cardinality=32, bottleneck_width=4,
                 classes=1000, layout='NCHW', **kwargs):
        super().__init__(layout=layout, **kwargs)
        channels = [256, 512, 1024, 2048]
        self.features = nn.HybridSequential()
        ax = -1 if layout == 'NHWC' else 1
        self.features.add(
            nn.Conv2D(64, kernel_size=7, strides=2, padding=3,
                      use_bias=False, in_channels=3, layout=layout),
            nn.BatchNormReLU(axis=ax),
            nn.MaxPool2D(pool_size=3, strides=2, padding=1, layout=layout))
        in_c = 64
        for i, num in enumerate(layers):
            stride = 1 if i == 0 else 2
            blk = nn.HybridSequential()
            blk.add(ResNextBlock(channels[i], cardinality, bottleneck_width,
                                 stride, True, in_channels=in_c,
                                 layout=layout))
            for _ in range(num - 1):
                blk.add(ResNextBlock(channels[i], cardinality,
                                     bottleneck_width, 1, False,
                                     in_channels=channels[i], layout=layout))
            self.features.add(blk)
            in_c = channels[i]
        self.features.add(nn.GlobalAvgPool2D(layout=layout))
        self.output = nn.Dense(classes)

    def forward(self, x):
        x = self._maybe_to_layout(x)
        x = self.features(x)
        x = x.flatten()
        return self.output(x)


def resnext50_32x4d(**kwargs):
    return ResNext([3, 4, 6, 3], cardinality=32, bottleneck_width=4, **kwargs)


def resnext101_32x4d(**kwargs):
    return ResNext([3, 4, 23, 3], cardinality=32, bottleneck_width=4, **kwargs)
