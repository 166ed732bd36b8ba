"""FP32 master-weight machinery (reference: prep_param_lists mix.py:53-63 and
the backward-accumulation trick mix.py:292-294; here a plain explicit copy —
no autograd leaf tricks — and per-optimizer-group, so schemes like BN-without-
weight-decay keep their group structure)."""
import torch

__all__ = ["MasterParams"]


class MasterParams:
    """FP32 copies of every parameter in an optimizer's param groups.

    Construction REPOINTS the optimizer's groups at the masters: the
    optimizer steps the masters; gradients flow model -> master via
    ``grads_from_model`` and updated weights master -> model via
    ``copy_to_model`` (keeps low-precision models stable; for fp32 models it
    reproduces the reference flagship's update path)."""

    def __init__(self, optimizer):
        self.pairs = []  # (model_param, master_param)
        for group in optimizer.param_groups:
            masters = []
            for p in group["params"]:
                mp = p.detach().clone().float().requires_grad_(True)
                masters.append(mp)
                self.pairs.append((p, mp))
                if p in optimizer.state:  # e.g. momentum loaded from a
                    optimizer.state[mp] = optimizer.state.pop(p)  # checkpoint
            group["params"] = masters

    def grads_from_model(self):
        for p, mp in self.pairs:
            if p.grad is None:
                continue
            if mp.grad is None:
                mp.grad = p.grad.detach().float().clone()
            else:
                mp.grad.copy_(p.grad.detach())

    def copy_to_model(self):
        with torch.no_grad():
            for p, mp in self.pairs:
                p.copy_(mp.to(p.dtype))

    def zero_grad(self):
        for _, mp in self.pairs:
            if mp.grad is not None:
                mp.grad.zero_()
