from .resnet import resnet18, resnet34, resnet50  # noqa: F401
from .toy_net import Toy_Net  # noqa: F401
from .vgg import vgg16_bn  # noqa: F401

_REGISTRY = {
    "toy_net": Toy_Net,
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "vgg16": vgg16_bn,
}


def build_model(name: str, **kwargs):
    try:
        return _REGISTRY[name](**kwargs)
    except KeyError:
        raise ValueError(f"unknown model {name!r}; available: {sorted(_REGISTRY)}")


def register_model(name: str, ctor) -> None:
    _REGISTRY[name] = ctor
