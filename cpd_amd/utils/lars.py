"""LARS optimizer (layer-wise adaptive rate scaling).

Update rule matches the reference's inline implementation (mix.py:297-310):
    local_lr = ||w|| / (||g|| + wd * ||w||) * trust_coefficient
    buf      = momentum * buf + lr * local_lr * (g + wd * w)
    w       -= buf
"""
import torch
from torch.optim.optimizer import Optimizer

__all__ = ["LARS"]


class LARS(Optimizer):
    def __init__(self, params, lr, momentum=0.9, weight_decay=1e-4,
                 trust_coefficient=0.001):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        trust_coefficient=trust_coefficient)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            wd = group["weight_decay"]
            mom = group["momentum"]
            lr = group["lr"]
            tc = group["trust_coefficient"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                w_norm = p.norm(2)
                g_norm = g.norm(2)
                denom = g_norm + wd * w_norm
                local_lr = tc * w_norm / denom if denom > 0 else \
                    torch.ones_like(denom)
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p)
                buf = state["momentum_buffer"]
                buf.mul_(mom).add_(g.add(p, alpha=wd), alpha=float(lr * local_lr))
                p.sub_(buf)
        return loss
