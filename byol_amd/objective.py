"""BYOL loss with the reference's exact numerics.

``regression_loss(x, y) = -2 * sum(x*y, dim=-1) / (||x|| * ||y||)`` where the
norms are WHOLE-TENSOR Frobenius norms (a deliberate faithful-to-reference
deviation from the paper's per-row normalisation), and the symmetric loss is
``mean(regression_loss(p1, z2.detach()) + regression_loss(p2, z1.detach()))``
— see ``/root/reference/objective.py:6-25``.

On GPU the forward+backward are each ONE fused HIP kernel (row dots, the two
Frobenius reductions and the mean in a single pass over the B x D tensors)
via :class:`FusedBYOLLoss`; on CPU the same math runs in plain PyTorch and
serves as the numerics oracle.
"""

import torch

from .ops import has_extension, require_extension

__all__ = ["regression_loss", "loss_function"]


def regression_loss(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    norm_x, norm_y = x.norm(), y.norm()
    return -2.0 * torch.sum(x * y, dim=-1) / (norm_x * norm_y)


def _loss_reference(p1, p2, z1, z2):
    loss_ab = regression_loss(p1, z2.detach())
    loss_ba = regression_loss(p2, z1.detach())
    return torch.mean(loss_ab + loss_ba)


class _FusedBYOLLoss(torch.autograd.Function):
    """Fused forward/backward on GPU.

    Forward returns mean_i[-2*dot(p1_i,z2_i)/(|p1||z2|) - 2*dot(p2_i,z1_i)/(|p2||z1|)].
    Backward w.r.t. x in a term with detached y:
      d/dx_kl = (-2*go/B) * [ y_kl/(Nx*Ny) - S * x_kl / (Nx^3 * Ny) ],
    with S = sum_i dot(x_i, y_i).
    """

    @staticmethod
    def forward(ctx, p1, p2, z1, z2):
        ext = require_extension("fused BYOL loss")
        loss, stats = ext.byol_loss_forward(p1, p2, z1, z2)
        # stats: [S_a, Np1, Nz2, S_b, Np2, Nz1]
        ctx.save_for_backward(p1, p2, z1, z2, stats)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        p1, p2, z1, z2, stats = ctx.saved_tensors
        ext = require_extension("fused BYOL loss")
        g1, g2 = ext.byol_loss_backward(p1, p2, z1, z2, stats,
                                        grad_out.contiguous())
        return g1, g2, None, None


def loss_function(online_prediction1: torch.Tensor,
                  online_prediction2: torch.Tensor,
                  target_projection1: torch.Tensor,
                  target_projection2: torch.Tensor) -> torch.Tensor:
    if (online_prediction1.is_cuda and has_extension()
            and online_prediction1.dtype == torch.float32
            and online_prediction1.dim() == 2):
        return _FusedBYOLLoss.apply(
            online_prediction1.contiguous(), online_prediction2.contiguous(),
            target_projection1.detach().contiguous(),
            target_projection2.detach().contiguous())
    if online_prediction1.is_cuda:
        # GPU without the extension: fail loudly (policy in byol_amd.ops),
        # except for dtypes the fused kernel does not cover.
        if online_prediction1.dtype == torch.float32:
            require_extension("fused BYOL loss")
        # non-fp32 (autocast bf16) path falls through to composed ATen ops
    return _loss_reference(online_prediction1, online_prediction2,
                           target_projection1, target_projection2)
