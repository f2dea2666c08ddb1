"""LAMB optimizer (replacement for the reference's Apex FusedLAMB,
``/root/reference/main.py:324-326``).  Plain PyTorch implementation of the
LAMB update (You et al., arXiv:1904.00962): Adam-style moments, then a
per-layer trust ratio ``||p|| / ||update||`` scales the step."""

import torch
from torch.optim.optimizer import Optimizer

__all__ = ["LAMB"]


class LAMB(Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                state["step"] += 1
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                bias1 = 1 - beta1 ** state["step"]
                bias2 = 1 - beta2 ** state["step"]
                update = (exp_avg / bias1) / (
                    (exp_avg_sq / bias2).sqrt().add_(group["eps"]))
                if group["weight_decay"] > 0:
                    update.add_(p, alpha=group["weight_decay"])
                w_norm = p.norm()
                u_norm = update.norm()
                trust = torch.where(
                    (w_norm > 0) & (u_norm > 0),
                    w_norm / u_norm, torch.ones_like(w_norm))
                p.add_(update, alpha=-group["lr"] * float(trust))
        return loss
