"""Model zoo (parity with gluon/model_zoo/vision/__init__.py — get_model).

No-network environment: pretrained=True is unavailable; random init +
load_parameters from a local .params file instead."""
from .resnet import *  # noqa: F401,F403
from .alexnet import alexnet, AlexNet  # noqa: F401
from .vgg import (vgg11, vgg13, vgg16, vgg19, vgg11_bn, vgg13_bn, vgg16_bn,
                  vgg19_bn, VGG)  # noqa: F401
from .mobilenet import (mobilenet1_0, mobilenet0_75, mobilenet0_5,
                        mobilenet0_25, mobilenet_v2_1_0, MobileNet,
                        MobileNetV2)  # noqa: F401
from .squeezenet import squeezenet1_0, squeezenet1_1, SqueezeNet  # noqa: F401
from .densenet import (densenet121, densenet161, densenet169, densenet201,
                       DenseNet)  # noqa: F401
from .inception import inception_v3, Inception3  # noqa: F401
from . import resnet  # noqa: F401

_models = {}


def _register_models():
    import sys
    mod = sys.modules[__name__]
    for name in ["resnet18_v1", "resnet34_v1", "resnet50_v1", "resnet101_v1",
                 "resnet152_v1", "resnet18_v2", "resnet34_v2", "resnet50_v2",
                 "resnet101_v2", "resnet152_v2", "resnet50_v1_5", "alexnet",
                 "vgg11", "vgg13", "vgg16", "vgg19", "vgg11_bn", "vgg13_bn",
                 "vgg16_bn", "vgg19_bn", "mobilenet1.0", "mobilenet0.75",
                 "mobilenet0.5", "mobilenet0.25", "mobilenetv2_1.0",
                 "squeezenet1.0", "squeezenet1.1", "inceptionv3",
                 "densenet121",
                 "densenet161", "densenet169", "densenet201"]:
        attr = name.replace(".", "_").replace("mobilenetv2", "mobilenet_v2")
        if attr == "inceptionv3":
            attr = "inception_v3"
        if hasattr(mod, attr):
            _models[name] = getattr(mod, attr)


_register_models()


def get_model(name, **kwargs):
    name = name.lower()
    if name not in _models:
        raise ValueError(
            f"Model {name} is not supported. Available: {sorted(_models)}")
    return _models[name](**kwargs)
