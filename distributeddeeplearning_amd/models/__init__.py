from .resnet import (  # noqa: F401
    ResNet,
    BasicBlock,
    Bottleneck,
    resnet18,
    resnet34,
    resnet50,
    resnet101,
    resnet152,
)
from .vgg import VGG, vgg16  # noqa: F401

_REGISTRY = {
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
    "vgg16": vgg16,
}


def build_model(name: str, num_classes: int = 1000, **kw):
    try:
        return _REGISTRY[name](num_classes=num_classes, **kw)
    except KeyError:
        raise ValueError(f"unknown model {name!r}; available: {sorted(_REGISTRY)}")


def available_models():
    return sorted(_REGISTRY)
