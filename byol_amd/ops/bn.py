"""Fused BatchNorm(+residual)(+ReLU) module.

GPU fp32 path: the hand-written NHWC HIP kernels in ``csrc/bn_fused.hip`` —
2 launches forward (stats -> apply with residual-add and ReLU folded in) and
2 backward, vs the ~8 MIOpen/ATen kernels the composed graph costs
(profiles/r01_bench_bs512_fp32_nhwc_kernels.md: ~190 ms of a 494 ms ResNet-50
BYOL step was BN+ReLU+add elementwise traffic).

SyncBN: with ``sync`` enabled, cross-replica batch statistics via ONE packed
2C-float all-reduce between the stats and apply kernels (forward) and
between reduce and apply (backward) — the reference reaches this through
``nn.SyncBatchNorm.convert_sync_batchnorm`` (``/root/reference/main.py:433``).
Equal per-rank element counts are assumed (the engine shards with
``drop_last=True``).

Everything else (CPU, non-fp32, NCHW, odd channel counts) runs a composed
PyTorch path with identical semantics — the numerics oracle.

Layout contract of the kernels: rows x channels with channels contiguous,
i.e. 4-D channels_last tensors or plain 2-D [B, C] (the projector/predictor
BatchNorm1d — same kernels).
"""

import os
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from . import extension, has_extension

__all__ = ["FusedBatchNorm"]


def _world(process_group) -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(process_group)
    return 1


def _rows_view(t: torch.Tensor, c: int) -> torch.Tensor:
    """Zero-copy [M, C] view of a channels-contiguous tensor."""
    if t.dim() == 4:
        return t.permute(0, 2, 3, 1).reshape(-1, c)
    return t.reshape(-1, c)


def _bf16_enabled() -> bool:
    # bf16-I/O BN kernels: validated round 2 (numerics vs fp32 oracle +
    # bench: 2805 img/s vs 2323 composed at bs=1024 bf16,
    # profiles/r02_validation.md) — default ON; BYOL_BF16_BN=0 disables
    return os.environ.get("BYOL_BF16_BN", "1") == "1"


def _hip_eligible(x: torch.Tensor, c: int) -> bool:
    ok_dtype = (x.dtype == torch.float32
                or (x.dtype == torch.bfloat16 and _bf16_enabled()))
    if not (x.is_cuda and ok_dtype and c % 4 == 0):
        return False
    if x.dim() == 4 and not x.is_contiguous(
            memory_format=torch.channels_last):
        return False
    if x.dim() == 2 and not x.is_contiguous():
        return False
    if x.dim() not in (2, 4):
        return False
    # layout/dtype qualify for the HIP path: the extension is now
    # REQUIRED — a silent ATen fallback on a GPU box would hide a broken
    # build (fail-loud policy, byol_amd.ops)
    from . import require_extension
    require_extension("fused BatchNorm")
    return True


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                training, momentum, eps, relu, sync, process_group):
        C = extension()
        c = weight.numel()
        xv = _rows_view(x, c)
        m = xv.shape[0]
        world = _world(process_group) if sync else 1

        bf16 = x.dtype == torch.bfloat16
        if training:
            nslots = 64  # atomic-contention fanout (see csrc/bn_fused.hip)
            acc = torch.zeros(nslots * 2 * c, device=x.device,
                              dtype=torch.float32)
            if bf16:
                C.bn_stats_bf16(xv, acc, m, c, nslots)
            else:
                C.bn_stats(xv, acc, m, c, nslots)
            acc2c = torch.empty(2 * c, device=x.device, dtype=torch.float32)
            C.bn_reduce_slots(acc, acc2c, nslots)
            count = float(m)
            if world > 1:
                dist.all_reduce(acc2c, group=process_group)
                count = float(m * world)
            mean = torch.empty(c, device=x.device, dtype=torch.float32)
            invstd = torch.empty_like(mean)
            C.bn_finalize(acc2c, mean, invstd, running_mean, running_var,
                          count, eps, momentum, c,
                          1 if running_mean is not None else 0)
        else:
            mean = running_mean
            invstd = torch.rsqrt(running_var + eps)
            count = float(m)

        y = torch.empty_like(x)
        yv = _rows_view(y, c)
        rv = _rows_view(residual, c) if residual is not None else None
        if bf16:
            C.bn_apply_bf16(xv, rv, mean, invstd, weight, bias, yv, m, c,
                            1 if relu else 0)
        else:
            C.bn_apply(xv, rv, mean, invstd, weight, bias, yv, m, c,
                       1 if relu else 0)
        ctx.save_for_backward(x, y, weight, mean, invstd)
        ctx.relu = relu
        ctx.has_residual = residual is not None
        ctx.count = count
        ctx.sync_world = world
        ctx.process_group = process_group
        ctx.c = c
        return y

    @staticmethod
    def backward(ctx, dy):
        C = extension()
        x, y, weight, mean, invstd = ctx.saved_tensors
        c = ctx.c
        bf16 = x.dtype == torch.bfloat16
        if bf16 and dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dy = dy.contiguous(memory_format=torch.channels_last) \
            if dy.dim() == 4 else dy.contiguous()
        dyv = _rows_view(dy, c)
        xv = _rows_view(x, c)
        yv = _rows_view(y, c)
        m = xv.shape[0]
        nslots = 64
        red_s = torch.zeros(nslots * 2 * c, device=x.device,
                            dtype=torch.float32)
        if bf16:
            C.bn_bwd_reduce_bf16(dyv, yv, xv, mean, invstd, red_s, m, c,
                                 1 if ctx.relu else 0, nslots)
        else:
            C.bn_bwd_reduce(dyv, yv, xv, mean, invstd, red_s, m, c,
                            1 if ctx.relu else 0, nslots)
        red = torch.empty(2 * c, device=x.device, dtype=torch.float32)
        C.bn_reduce_slots(red_s, red, nslots)
        # weight/bias grads are LOCAL sums (DDP averages them afterwards,
        # same contract as nn.SyncBatchNorm)
        db = red[:c].clone()
        dw = red[c:].clone()
        if ctx.sync_world > 1:
            dist.all_reduce(red, group=ctx.process_group)
        dx = torch.empty_like(x)
        dxv = _rows_view(dx, c)
        dres = None
        drv = None
        if ctx.has_residual:
            dres = torch.empty_like(x)
            drv = _rows_view(dres, c)
        if bf16:
            C.bn_bwd_apply_bf16(dyv, yv, xv, mean, invstd, weight, red,
                                dxv, drv, 1.0 / ctx.count, m, c,
                                1 if ctx.relu else 0)
        else:
            C.bn_bwd_apply(dyv, yv, xv, mean, invstd, weight, red, dxv,
                           drv, 1.0 / ctx.count, m, c, 1 if ctx.relu else 0)
        return (dx, dres, dw, db, None, None, None, None, None, None, None,
                None)


class FusedBatchNorm(nn.Module):
    """BatchNorm (1d rows x channels or 2d NHWC) with optional fused
    residual-add and ReLU.  State names match ``nn.BatchNorm2d`` so
    checkpoints and weight-decay grouping are unchanged."""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.sync = False
        self.process_group = None
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def extra_repr(self):
        return (f"{self.num_features}, eps={self.eps}, "
                f"momentum={self.momentum}, relu={self.relu}, "
                f"sync={self.sync}")

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.training:
            with torch.no_grad():
                self.num_batches_tracked += 1
        # eval-mode forwards that still need autograd (e.g. linear-probe
        # fine-tuning) take the composed path: _FusedBNFunction.backward
        # implements only the training-mode (batch-stats) gradient, which
        # is wrong when running stats are constants
        needs_eval_grad = (not self.training and torch.is_grad_enabled()
                           and (x.requires_grad
                                or self.weight.requires_grad))
        if not needs_eval_grad and _hip_eligible(x, self.num_features) and \
                (residual is None or _hip_eligible(residual,
                                                   self.num_features)):
            return _FusedBNFunction.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, self.training, self.momentum, self.eps,
                self.relu, self.sync, self.process_group)
        return self._composed(x, residual)

    # -- composed oracle path (CPU / non-fp32 / NCHW) ---------------------
    def _composed(self, x, residual):
        world = _world(self.process_group) if self.sync else 1
        if self.training and world > 1:
            from ..parallel.sync_bn import _SyncBNFunction
            y, mean, var, count = _SyncBNFunction.apply(
                x, self.weight, self.bias, self.eps, self.process_group,
                world)
            with torch.no_grad():
                n = float(count)
                unbiased = var * (n / max(n - 1.0, 1.0))
                self.running_mean.mul_(1 - self.momentum).add_(
                    mean, alpha=self.momentum)
                self.running_var.mul_(1 - self.momentum).add_(
                    unbiased, alpha=self.momentum)
        else:
            y = F.batch_norm(x, self.running_mean, self.running_var,
                             self.weight, self.bias, self.training,
                             self.momentum, self.eps)
        if residual is not None:
            y = y + residual
        if self.relu:
            y = F.relu(y, inplace=True)
        return y
