"""Layer/param helpers replicating the reference's ``helpers.layers``
submodule contract (reconstructed from call sites — SURVEY.md section 2.2)."""

import math
from typing import Optional

import torch
import torch.nn as nn

__all__ = ["add_weight_decay", "init_weights", "polyak_ema_parameters",
           "get_polyak_prediction", "number_of_parameters"]


def add_weight_decay(model: nn.Module, weight_decay: float):
    """Two param groups: weights with decay (``ignore: False`` — LARS adapts
    them) and bias/norm params with no decay and ``ignore: True`` (LARS skips
    adaptation; ``/root/reference/optimizers/lars.py:88-100`` reads the key).
    Norm/bias params are identified as 1-d-or-lower tensors (BN weight/bias
    and every bias), the standard SimCLR/BYOL recipe."""
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        (no_decay if p.ndim <= 1 else decay).append(p)
    return [
        {"params": decay, "weight_decay": weight_decay, "ignore": False},
        {"params": no_decay, "weight_decay": 0.0, "ignore": True},
    ]


def init_weights(model: nn.Module, init: Optional[str] = None) -> nn.Module:
    """Optional named weight-init; ``None`` keeps the construction-time
    (pytorch-default/kaiming) init, like the reference default."""
    if init is None:
        return model

    def _apply(m):
        w = getattr(m, "weight", None)
        if w is None or w.ndim < 2:
            return
        if init == "xavier_uniform":
            nn.init.xavier_uniform_(w)
        elif init == "xavier_normal":
            nn.init.xavier_normal_(w)
        elif init == "kaiming_uniform":
            nn.init.kaiming_uniform_(w, a=math.sqrt(5))
        elif init == "kaiming_normal":
            nn.init.kaiming_normal_(w, mode="fan_out", nonlinearity="relu")
        elif init == "orthogonal":
            nn.init.orthogonal_(w)
        else:
            raise ValueError(f"unknown weight init {init!r}")
        b = getattr(m, "bias", None)
        if b is not None:
            nn.init.zeros_(b)

    with torch.no_grad():
        model.apply(_apply)
    return model


@torch.no_grad()
def polyak_ema_parameters(model: nn.Module, decay: float) -> None:
    """Simple polyak average of model params (distinct from BYOL's CosEMA
    target), kept on the module as ``_polyak_shadow``; first call initialises
    the shadow to the current params."""
    module = model.module if hasattr(model, "module") else model
    shadow = getattr(module, "_polyak_shadow", None)
    if shadow is None:
        shadow = {name: p.detach().clone()
                  for name, p in module.named_parameters()}
        module._polyak_shadow = shadow
        return
    for name, p in module.named_parameters():
        shadow[name].mul_(decay).add_(p.detach(), alpha=1.0 - decay)


@torch.no_grad()
def get_polyak_prediction(model: nn.Module, pred_fn):
    """Evaluate ``pred_fn`` with the polyak-averaged weights swapped in
    (swap -> call -> restore)."""
    module = model.module if hasattr(model, "module") else model
    shadow = getattr(module, "_polyak_shadow", None)
    if shadow is None:
        return pred_fn()
    backup = {name: p.detach().clone()
              for name, p in module.named_parameters()}
    for name, p in module.named_parameters():
        p.copy_(shadow[name])
    try:
        out = pred_fn()
    finally:
        for name, p in module.named_parameters():
            p.copy_(backup[name])
    return out


def number_of_parameters(model: nn.Module) -> int:
    return sum(p.numel() for p in model.parameters() if p.requires_grad)
