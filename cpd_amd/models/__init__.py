"""Model zoo: the reference's three example families (res_cifar registry name
kept, example/ResNet18/models/__init__.py:1)."""
from .resnet_cifar import res_cifar, ResNetCifar
from .resnet import resnet50, ResNet
from .davidnet import davidnet, DavidNet

def resnet50_quant(num_classes=1000, fused_bn=False, exp=5, man=2):
    """BASELINE config 4: ResNet50 with every conv/linear GEMM routed through
    the (exp,man)-Kahan quantized accumulator (default e5m2)."""
    from cpd_amd.quant import quantize_model_gemms
    return quantize_model_gemms(resnet50(num_classes, fused_bn=fused_bn),
                                exp=exp, man=man)


def res_cifar_quant(num_classes=10, fused_bn=False, exp=5, man=2):
    from cpd_amd.quant import quantize_model_gemms
    return quantize_model_gemms(res_cifar(num_classes=num_classes,
                                          fused_bn=fused_bn),
                                exp=exp, man=man)


REGISTRY = {
    "res_cifar": res_cifar,
    "resnet18_cifar": res_cifar,
    "resnet50": resnet50,
    "resnet50_quant": resnet50_quant,
    "resnet18_cifar_quant": res_cifar_quant,
    "davidnet": davidnet,
}


def build_model(name, **kwargs):
    return REGISTRY[name](**kwargs)


__all__ = ["res_cifar", "ResNetCifar", "resnet50", "ResNet", "davidnet",
           "DavidNet", "build_model", "REGISTRY"]
