"""Model zoo: the reference's three example families (res_cifar registry name
kept, example/ResNet18/models/__init__.py:1)."""
from .resnet_cifar import res_cifar, ResNetCifar
from .resnet import resnet50, ResNet
from .davidnet import davidnet, DavidNet

REGISTRY = {
    "res_cifar": res_cifar,
    "resnet18_cifar": res_cifar,
    "resnet50": resnet50,
    "davidnet": davidnet,
}


def build_model(name, **kwargs):
    return REGISTRY[name](**kwargs)


__all__ = ["res_cifar", "ResNetCifar", "resnet50", "ResNet", "davidnet",
           "DavidNet", "build_model", "REGISTRY"]
