"""Model zoo (reference gluon/model_zoo/vision/__init__.py)."""
from .resnet import *  # noqa: F401,F403
from .resnet import get_resnet, resnet_spec
from .alexnet import alexnet, AlexNet
from .vgg import vgg11, vgg13, vgg16, vgg19, vgg11_bn, vgg13_bn, vgg16_bn, \
    vgg19_bn, VGG
from .squeezenet import squeezenet1_0, squeezenet1_1, SqueezeNet
from .mobilenet import (mobilenet1_0, mobilenet0_75, mobilenet0_5,
                        mobilenet0_25, mobilenet_v2_1_0, MobileNet,
                        MobileNetV2)
from .densenet import densenet121, densenet161, densenet169, densenet201, \
    DenseNet
from .inception import inception_v3, Inception3

_models = {}


def _register_models():
    from . import resnet as _r
    for v in (1, 2):
        for n in (18, 34, 50, 101, 152):
            _models[f'resnet{n}_v{v}'] = getattr(_r, f'resnet{n}_v{v}')
    _models.update({
        'alexnet': alexnet,
        'vgg11': vgg11, 'vgg13': vgg13, 'vgg16': vgg16, 'vgg19': vgg19,
        'vgg11_bn': vgg11_bn, 'vgg13_bn': vgg13_bn, 'vgg16_bn': vgg16_bn,
        'vgg19_bn': vgg19_bn,
        'squeezenet1.0': squeezenet1_0, 'squeezenet1.1': squeezenet1_1,
        'mobilenet1.0': mobilenet1_0, 'mobilenet0.75': mobilenet0_75,
        'mobilenet0.5': mobilenet0_5, 'mobilenet0.25': mobilenet0_25,
        'mobilenetv2_1.0': mobilenet_v2_1_0,
        'densenet121': densenet121, 'densenet161': densenet161,
        'densenet169': densenet169, 'densenet201': densenet201,
        'inceptionv3': inception_v3,
    })


_register_models()


def get_model(name, **kwargs):
    name = name.lower()
    if name not in _models:
        raise ValueError(f'unknown model {name}; available: {sorted(_models)}')
    return _models[name](**kwargs)
