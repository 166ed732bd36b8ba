"""FP32 master-weight machinery (reference: prep_param_lists mix.py:53-63 and
the backward-accumulation trick mix.py:292-294; here a plain explicit copy —
no autograd leaf tricks)."""
import torch

__all__ = ["MasterParams"]


class MasterParams:
    """FP32 copies of a model's trainable parameters.

    The optimizer steps the masters; gradients flow model -> master via
    ``grads_from_model`` and updated weights master -> model via
    ``copy_to_model`` (keeps low-precision models stable; for fp32 models it
    reproduces the reference flagship's update path)."""

    def __init__(self, model):
        self.model_params = [p for p in model.parameters() if p.requires_grad]
        self.master_params = [p.detach().clone().float()
                              for p in self.model_params]
        for mp in self.master_params:
            mp.requires_grad_(True)

    def grads_from_model(self):
        for p, mp in zip(self.model_params, self.master_params):
            if p.grad is None:
                continue
            if mp.grad is None:
                mp.grad = p.grad.detach().float().clone()
            else:
                mp.grad.copy_(p.grad.detach())

    def copy_to_model(self):
        with torch.no_grad():
            for p, mp in zip(self.model_params, self.master_params):
                p.copy_(mp.to(p.dtype))

    def zero_grad(self):
        for mp in self.master_params:
            if mp.grad is not None:
                mp.grad.zero_()
