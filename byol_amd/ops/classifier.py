"""Fused cross-entropy + top-k for the linear probe.

One kernel computes the CE loss AND top-1/top-5 accuracy (the reference
spends ~6 ATen kernels: softmax/CE + topk + eq + sums,
``/root/reference/main.py:596-598``); backward is one elementwise kernel.
Returns (mean_loss, acc1_percent, acc5_percent) with loss differentiable.
CPU/fallback path composes F.cross_entropy + metrics.topk (the oracle).
"""

from typing import Tuple

import torch
import torch.nn.functional as F

from . import extension, has_extension
from ..engine import metrics as _metrics

__all__ = ["cross_entropy_topk"]


class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        C = extension()
        m, n = logits.shape
        out = torch.zeros(3, device=logits.device, dtype=torch.float32)
        row_stats = torch.empty(m, 2, device=logits.device,
                                dtype=torch.float32)
        C.ce_topk_fwd(logits, labels, out, row_stats)
        ctx.save_for_backward(logits, labels, row_stats)
        loss = out[0] / m
        acc1 = out[1] * (100.0 / m)
        acc5 = out[2] * (100.0 / m)
        ctx.mark_non_differentiable(acc1, acc5)
        return loss, acc1, acc5

    @staticmethod
    def backward(ctx, grad_loss, _g1, _g5):
        C = extension()
        logits, labels, row_stats = ctx.saved_tensors
        dlogits = torch.empty_like(logits)
        C.ce_bwd(logits, labels, row_stats, grad_loss.contiguous(), dlogits)
        return dlogits, None


def cross_entropy_topk(logits: torch.Tensor, labels: torch.Tensor,
                       ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    if (logits.is_cuda and logits.dtype == torch.float32
            and logits.dim() == 2 and logits.is_contiguous()
            and has_extension()):
        return _FusedCE.apply(logits, labels)
    loss = F.cross_entropy(input=logits, target=labels)
    acc1, acc5 = _metrics.topk(output=logits, target=labels, topk=(1, 5))
    return loss, acc1, acc5
