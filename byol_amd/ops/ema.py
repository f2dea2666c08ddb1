"""Flat-parameter EMA update (the BYOL target-network update).

One fused lerp over the contiguous flat parameter buffer:
``mean <- (1-decay)*x + decay*mean``.  Replaces the reference's
pack-params-then-blend (K7/K8 in SURVEY.md: ~5 full parameter traversals per
step just for the EMA bookkeeping; here it is 2 reads + 1 write in one
float4-vectorized HIP kernel).  Reference semantics:
``/root/reference/main.py:158-161``.
"""

import torch

from . import require_extension


@torch.no_grad()
def flat_ema_update(mean: torch.Tensor, x: torch.Tensor, decay: float,
                    decay_dev: torch.Tensor = None) -> None:
    """One fused lerp.  ``decay_dev`` (1-float device scalar) overrides the
    host ``decay`` — used under hipGraph capture, where the replay wrapper
    rewrites the scalar between replays."""
    assert mean.is_contiguous() and x.is_contiguous()
    assert mean.numel() == x.numel()
    if mean.is_cuda:
        ext = require_extension("flat_ema_update")
        if decay_dev is not None:
            ext.flat_ema_update_dev(mean, x, decay_dev)
        else:
            ext.flat_ema_update(mean, x, float(decay))
    else:
        # oracle path: identical math via one ATen lerp
        mean.lerp_(x, 1.0 - float(decay))


@torch.no_grad()
def flat_ema_update_reference(mean: torch.Tensor, x: torch.Tensor,
                              decay: float) -> None:
    """Pure-PyTorch oracle for tests (same math, no extension)."""
    mean.mul_(decay).add_(x, alpha=1.0 - decay)
