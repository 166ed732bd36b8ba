"""Swap a model's GEMM-bearing modules for their quantized-accumulator
equivalents (BASELINE config 4: the reference routes all model GEMMs through
the (exp,man)-Kahan accumulator via Quant_Linear/Quant_Conv,
quant_module.py:23-139 — the examples never do this; this converter makes it
a one-call model-level capability)."""
import torch.nn as nn

from .module import Quant_Conv, Quant_Linear

__all__ = ["quantize_model_gemms"]


def quantize_model_gemms(model, exp=8, man=23):
    """Replace every eligible nn.Conv2d / nn.Linear in ``model`` (in place)
    with Quant_Conv / Quant_Linear carrying the same weights.  Eligible:
    square-kernel, dilation=1, groups=1 convs (the reference's Quant_Conv
    contract).  Returns the model."""
    for name, child in list(model.named_children()):
        if isinstance(child, nn.Conv2d):
            kh, kw = child.kernel_size
            if (kh == kw and child.dilation == (1, 1) and child.groups == 1
                    and child.padding_mode == "zeros"):
                q = Quant_Conv(child.in_channels, child.out_channels, kh,
                               stride=child.stride[0],
                               padding=child.padding[0],
                               bias=child.bias is not None, exp=exp, man=man)
                q.weight.data.copy_(child.weight.data)
                if child.bias is not None:
                    q.bias.data.copy_(child.bias.data)
                setattr(model, name, q)
        elif isinstance(child, nn.Linear):
            q = Quant_Linear(child.in_features, child.out_features,
                             bias=child.bias is not None, exp=exp, man=man)
            q.weight.data.copy_(child.weight.data)
            if child.bias is not None:
                q.bias.data.copy_(child.bias.data)
            setattr(model, name, q)
        else:
            quantize_model_gemms(child, exp=exp, man=man)
    return model
