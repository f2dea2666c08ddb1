"""MFMA 1x1 convolution (f32, NHWC) — hand-written gfx950 kernels with
per-shape dispatch against MIOpen.

A stride-1 1x1 conv in NHWC is exactly a GEMM over M = N*H*W rows;
``csrc/conv1x1.hip`` implements fwd/dgrad/wgrad on the f32-input MFMA
(``v_mfma_f32_32x32x2_f32``, the exact-f32 matrix path — 155 TF measured on
this chip).  ``tools/conv_microbench.py`` measures these against MIOpen's
igemm solvers shape-by-shape; dispatch is controlled by:

* env ``BYOL_MFMA_CONV1X1`` = "0" (off), "1" (always), "auto" (default:
  the measured per-shape tables below — fwd 8/12, dgrad 11/12, wgrad 3/12
  ResNet-50 shapes as of round 2), or "autotune" (runtime per-shape
  ours-vs-MIOpen selection, cudnn.benchmark-style; cache via
  ``BYOL_AUTOTUNE_CACHE``).  ``--conv-dispatch`` maps to these.
"""

import os
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import extension, has_extension

__all__ = ["MFMAConv1x1", "mfma_conv_mode"]


def mfma_conv_mode() -> str:
    return os.environ.get("BYOL_MFMA_CONV1X1", "auto")


# Shapes (K=Cin, N=Cout) where the MFMA FORWARD kernel measured faster
# than MIOpen on MI355X at bs-512-class M (tools/conv_microbench.py;
# profiles/r02_validation.md — the 128x64 fast tile added the two N=64
# shapes in round 2).  BYOL_MFMA_CONV1X1=1 forces our kernels everywhere
# (tests).
_AUTO_SHAPES = {
    (64, 64), (64, 256), (256, 64), (256, 128),
    (128, 512), (512, 128), (512, 256), (256, 1024),
}

# Shapes where OUR dgrad kernel beats MIOpen's split backward (r2 call 3:
# 11 of 12 ResNet-50 shapes — e.g. 0.202 vs 0.238 ms on 56/64/64, 0.896
# vs 1.015 on 28/512/256).  In auto mode backward runs our dgrad on these
# + per-shape wgrad routing below.
_AUTO_DGRAD = {
    (64, 64), (64, 256), (256, 64), (256, 128),
    (128, 512), (512, 128), (512, 256), (256, 1024),
    (1024, 256), (1024, 512), (2048, 512),
}

# Shapes where OUR wgrad (v3 glds pipeline) beats MIOpen wrw (r2 call 4:
# 0.536 vs 0.559 on 56/64/256, 0.540 vs 0.558 on 56/256/64, 0.912 vs
# 0.938 on 56/256/128); the rest stay with MIOpen (3-8% faster there).
_AUTO_WGRAD = {
    (64, 256), (256, 64), (256, 128),
}


def _eligible(x: torch.Tensor, weight: torch.Tensor) -> bool:
    if not (x.is_cuda and x.dtype == torch.float32 and has_extension()):
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    k = weight.shape[1]
    if k % 32 != 0:
        return False
    mode = mfma_conv_mode()
    if mode == "0":
        return False
    if mode in ("1", "autotune"):
        return True
    shape = (k, weight.shape[0])
    return shape in _AUTO_SHAPES or shape in _AUTO_DGRAD


def _rows(t: torch.Tensor, c: int) -> torch.Tensor:
    return t.permute(0, 2, 3, 1).reshape(-1, c)


# ---------------------------------------------------------------------------
# per-forward weight-layout cache
#
# The MFMA kernels want W pre-arranged ([N,K]+[K,N] for 1x1, [9][K][N] for
# 3x3); rebuilding those copies on every conv call cost ~0.5 ms/call on the
# big shapes.  BYOL calls every conv FOUR times per step with only two
# distinct weight tensors (online flat views x2 passes, EMA views x2), so a
# cache that BYOL.forward clears on entry halves the copies and can never go
# stale: all parameter mutation (LARS step, EMA lerp, DDP broadcast) happens
# outside model.forward.  Entries hold a strong ref to the weight and check
# both identity and _version, so standalone use (tests, microbench) is also
# safe whenever in-place updates go through ATen.
# ---------------------------------------------------------------------------

_weight_cache: dict = {}


def clear_weight_cache() -> None:
    _weight_cache.clear()


def _cached_layout(weight: torch.Tensor, kind: str, build):
    key = (id(weight), kind)
    hit = _weight_cache.get(key)
    if hit is not None and hit[0] is weight and hit[1] == weight._version:
        return hit[2]
    value = build()
    _weight_cache[key] = (weight, weight._version, value)
    return value


class _Conv1x1Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        C = extension()
        b, k, h, w = x.shape
        n = weight.shape[0]
        m = b * h * w
        # [N,K] + pre-transposed [K,N] (feeds the glds fast path), cached
        # across the 2 passes sharing this weight tensor within one step
        wv, wt = _cached_layout(
            weight, "1x1",
            lambda: (lambda v: (v, v.t().contiguous()))(
                weight.reshape(n, k).contiguous()))
        mode = mfma_conv_mode()
        use_ours = (mode == "1" or (k, n) in _AUTO_SHAPES
                    or (mode == "autotune"))
        if use_ours:
            y = torch.empty((b, n, h, w), device=x.device, dtype=x.dtype,
                            memory_format=torch.channels_last)
            C.conv1x1_fwd(_rows(x, k), wv, wt, _rows(y, n), m, k, n)
        else:
            # shape is here only for its backward win (_AUTO_DGRAD):
            # MIOpen forward
            y = torch.ops.aten.convolution(
                x, wv.view(n, k, 1, 1), None, [1, 1], [0, 0], [1, 1],
                False, [0, 0], 1)
        ctx.save_for_backward(x, wv)
        ctx.dims = (b, k, h, w, n, m)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wv = ctx.saved_tensors
        b, k, h, w, n, m = ctx.dims
        dy = dy.contiguous(memory_format=torch.channels_last)
        if mfma_conv_mode() == "1":
            # our dgrad/wgrad kernels (forced mode; numerics tests)
            C = extension()
            dx = torch.empty_like(x)
            C.conv1x1_dgrad(_rows(dy, n), wv, _rows(dx, k), m, n, k)
            if os.environ.get("BYOL_WGRAD", "atomic") == "v2":
                # v2 (measured slower; kept for A/Bs): partial slabs + reduce
                nchunks = C.wgrad_nchunks(m, n, k)
                partial = torch.empty(nchunks * n * k, device=x.device,
                                      dtype=x.dtype)
                dw = torch.empty(n, k, device=x.device, dtype=x.dtype)
                C.conv1x1_wgrad_v2(_rows(dy, n), _rows(x, k), partial, dw,
                                   m, n, k)
            else:
                dw = torch.zeros(n, k, device=x.device, dtype=x.dtype)
                C.conv1x1_wgrad(_rows(dy, n), _rows(x, k), dw, m, n, k)
            return dx, dw.view(n, k, 1, 1)
        # auto mode: per-shape measured routing (r2 calls 3/4) — our dgrad
        # on _AUTO_DGRAD, our wgrad v3 on _AUTO_WGRAD, MIOpen elsewhere
        use_dgrad = (k, n) in _AUTO_DGRAD and m % 128 == 0
        use_wgrad = ((k, n) in _AUTO_WGRAD and m % 32 == 0
                     and n % 64 == 0 and k % 64 == 0)
        if use_dgrad or use_wgrad:
            C = extension()
            if use_dgrad:
                dx = torch.empty_like(x)
                C.conv1x1_dgrad(_rows(dy, n), wv, _rows(dx, k), m, n, k)
            else:
                dx, _, _ = torch.ops.aten.convolution_backward(
                    dy, x, wv.view(n, k, 1, 1), [0], [1, 1], [0, 0],
                    [1, 1], False, [0, 0], 1, [True, False, False])
            if use_wgrad:
                dw = torch.zeros(n, k, device=x.device, dtype=x.dtype)
                C.conv1x1_wgrad(_rows(dy, n), _rows(x, k), dw, m, n, k)
                dw = dw.view(n, k, 1, 1)
            else:
                _, dw, _ = torch.ops.aten.convolution_backward(
                    dy, x, wv.view(n, k, 1, 1), [0], [1, 1], [0, 0],
                    [1, 1], False, [0, 0], 1, [False, True, False])
            return dx, dw
        dx, dw, _ = torch.ops.aten.convolution_backward(
            dy, x, wv.view(n, k, 1, 1), [0], [1, 1], [0, 0], [1, 1], False,
            [0, 0], 1, [True, True, False])
        return dx, dw


@torch.no_grad()
def _autotune_decide_1x1(x: torch.Tensor, weight: torch.Tensor) -> bool:
    """cudnn.benchmark-style runtime selection (BYOL_MFMA_CONV1X1=autotune):
    first sight of a (k, n, m) times our fwd kernel vs MIOpen, cached for
    the process (and BYOL_AUTOTUNE_CACHE if set)."""
    from .autotune import autotuner
    b, k, h, w = x.shape
    n = weight.shape[0]
    m = b * h * w
    tun = autotuner()
    key = ("c1f", k, n, m)
    hit = tun.decisions.get(key)
    if hit is not None:
        return hit
    C = extension()
    xd = x.detach()
    wd = weight.detach()
    xr = _rows(xd, k)
    wv = wd.reshape(n, k).contiguous()
    wt = wv.t().contiguous()
    yr = torch.empty(m, n, device=x.device, dtype=x.dtype)
    return tun.choose(
        key,
        ours=lambda: C.conv1x1_fwd(xr, wv, wt, yr, m, k, n),
        theirs=lambda: F.conv2d(xd, wd))


class MFMAConv1x1(nn.Conv2d):
    """Drop-in for stride-1 bias-free 1x1 nn.Conv2d; routes to the MFMA
    kernels when eligible, MIOpen otherwise.  Dispatch modes
    (BYOL_MFMA_CONV1X1): "auto" = measured static tables (default),
    "autotune" = runtime per-shape selection, "1" force, "0" off."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _eligible(x, self.weight):
            if (mfma_conv_mode() == "autotune"
                    and not _autotune_decide_1x1(x, self.weight)):
                return F.conv2d(x, self.weight, self.bias, self.stride,
                                self.padding, self.dilation, self.groups)
            return _Conv1x1Fn.apply(x, self.weight)
        return F.conv2d(x, self.weight, self.bias, self.stride,
                        self.padding, self.dilation, self.groups)


# ---------------------------------------------------------------------------
# 3x3 convolution: MFMA forward (implicit GEMM over 9 taps), MIOpen backward
# ---------------------------------------------------------------------------

_zero_pages: dict = {}


def _zero_page(device) -> torch.Tensor:
    """Persistent 16-B zero buffer the pad-free 3x3 kernel redirects
    out-of-bounds tap lanes to."""
    zp = _zero_pages.get(device)
    if zp is None:
        zp = torch.zeros(8, device=device, dtype=torch.float32)
        _zero_pages[device] = zp
    return zp


def mfma_conv3x3_mode() -> str:
    # "0" off, "1" force everywhere, "auto" per-shape table below.
    return os.environ.get("BYOL_MFMA_CONV3X3", "auto")


def _parse_3x3_table(env: str):
    out = set()
    for item in env.split(","):
        parts = item.strip().split("/")
        if len(parts) == 3:
            out.add((int(parts[0]), int(parts[1]), int(parts[2])))
    return out


# (K, N, stride) where the pad-free MFMA fwd measured at parity-or-better
# vs MIOpen (r2 call 10); override with BYOL_CONV3X3_AUTO="k/n/s,k/n/s".
_AUTO_3X3 = _parse_3x3_table(
    os.environ.get("BYOL_CONV3X3_AUTO", ""))


def _conv3x3_auto(k: int, n: int, stride: int) -> bool:
    mode = mfma_conv3x3_mode()
    if mode == "1":
        return True
    if mode == "auto":
        return (k, n, stride) in _AUTO_3X3
    return False


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride):
        C = extension()
        b, k, hi, wi = x.shape
        n = weight.shape[0]
        ho = (hi + 2 - 3) // stride + 1
        wo = (wi + 2 - 3) // stride + 1
        # prepermute W [N,K,3,3] -> [9][K][N] for coalesced B staging
        # (cached across the 2 passes sharing this weight within one step)
        wp = _cached_layout(
            weight, "3x3",
            lambda: weight.reshape(n, k, 9).permute(2, 1, 0).contiguous())
        y = torch.empty((b, n, ho, wo), device=x.device, dtype=x.dtype,
                        memory_format=torch.channels_last)
        m = b * ho * wo
        if (m % 128 == 0 and n % 64 == 0 and k % 32 == 0
                and os.environ.get("BYOL_CONV3X3_FAST", "1") == "1"):
            if os.environ.get("BYOL_CONV3X3_NOPAD", "1") == "1":
                # pad-free: OOB tap lanes read a 16-B zero page instead of
                # a padded input copy (saves the 0.07-0.28 ms pad pass)
                zpage = _zero_page(x.device)
                C.conv3x3_fwd_nopad(_rows(x, k), wp, _rows(y, n), zpage,
                                    b, hi, wi, ho, wo, k, n, stride)
            else:
                xpad = torch.empty(b * (hi + 2) * (wi + 2) * k,
                                   device=x.device, dtype=x.dtype)
                C.pad_nhwc(_rows(x, k), xpad, b, hi, wi, k)
                C.conv3x3_fwd_fast(xpad, wp, _rows(y, n), b, hi, wi, ho,
                                   wo, k, n, stride)
        else:
            C.conv3x3_fwd(_rows(x, k), wp, _rows(y, n), b, hi, wi, ho, wo,
                          k, n, stride)
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        s = ctx.stride
        b, k, hi, wi = x.shape
        n = weight.shape[0]
        ho = (hi + 2 - 3) // s + 1
        wo = (wi + 2 - 3) // s + 1
        m = b * ho * wo
        # our 9-tap wgrad (v3 m-pipeline per tap, padded input) — gated by
        # BYOL_C3WGRAD until the microbench proves per-shape wins
        use_wgrad = (os.environ.get("BYOL_C3WGRAD", "0") == "1"
                     and m % 32 == 0 and n % 64 == 0 and k % 64 == 0)
        # stride-1 dgrad == conv3x3(pad(dy), rot180(W)^T): reuses the fwd
        # fast kernel directly — gated by BYOL_C3DGRAD
        use_dgrad = (os.environ.get("BYOL_C3DGRAD", "0") == "1"
                     and s == 1 and (b * hi * wi) % 128 == 0
                     and k % 64 == 0 and n % 32 == 0)
        if use_wgrad or use_dgrad:
            C = extension()
            if use_dgrad:
                # Wp_d[tap][n][k] = W[n][k][2-ty][2-tx]
                wpd = _cached_layout(
                    weight, "3x3d",
                    lambda: weight.flip([2, 3]).reshape(n, k, 9)
                    .permute(2, 0, 1).contiguous())
                dypad = torch.empty(b * (hi + 2) * (wi + 2) * n,
                                    device=x.device, dtype=x.dtype)
                C.pad_nhwc(_rows(dy, n), dypad, b, hi, wi, n)
                dx = torch.empty_like(x)
                C.conv3x3_fwd_fast(dypad, wpd, _rows(dx, k), b, hi, wi,
                                   hi, wi, n, k, 1)
            else:
                dx, _, _ = torch.ops.aten.convolution_backward(
                    dy, x, weight, [0], [s, s], [1, 1], [1, 1], False,
                    [0, 0], 1, [True, False, False])
            if use_wgrad:
                xpad = torch.empty(b * (hi + 2) * (wi + 2) * k,
                                   device=x.device, dtype=x.dtype)
                C.pad_nhwc(_rows(x, k), xpad, b, hi, wi, k)
                dw9 = torch.zeros(9 * n * k, device=x.device, dtype=x.dtype)
                dw = torch.empty(n, k, 3, 3, device=x.device, dtype=x.dtype)
                C.conv3x3_wgrad(_rows(dy, n), xpad, dw9, dw.reshape(-1), b,
                                hi, wi, ho, wo, k, n, s)
            else:
                _, dw, _ = torch.ops.aten.convolution_backward(
                    dy, x, weight, [0], [s, s], [1, 1], [1, 1], False,
                    [0, 0], 1, [False, True, False])
            return dx, dw, None
        dx, dw, _ = torch.ops.aten.convolution_backward(
            dy, x, weight, [0], [s, s], [1, 1], [1, 1], False, [0, 0], 1,
            [True, True, False])
        return dx, dw, None


@torch.no_grad()
def _autotune_decide_3x3(x: torch.Tensor, weight: torch.Tensor,
                         stride: int) -> bool:
    from .autotune import autotuner
    b, k, h, w = x.shape
    n = weight.shape[0]
    tun = autotuner()
    key = ("c3f", k, n, stride, b * h * w)
    hit = tun.decisions.get(key)
    if hit is not None:
        return hit
    xd = x.detach()
    wd = weight.detach()
    return tun.choose(
        key,
        ours=lambda: _Conv3x3Fn.apply(xd, wd, stride),
        theirs=lambda: F.conv2d(xd, wd, stride=stride, padding=1))


class MFMAConv3x3(nn.Conv2d):
    """Drop-in for bias-free pad-1 3x3 nn.Conv2d; MFMA forward when enabled
    and eligible, MIOpen otherwise (and always for backward unless the
    BYOL_C3WGRAD/BYOL_C3DGRAD gates are set)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (x.is_cuda and x.dtype == torch.float32 and has_extension()
                and x.is_contiguous(memory_format=torch.channels_last)
                and self.weight.shape[1] % 32 == 0):
            mode = mfma_conv3x3_mode()
            take = _conv3x3_auto(self.weight.shape[1],
                                 self.weight.shape[0], self.stride[0])
            if not take and mode == "autotune":
                take = _autotune_decide_3x3(x, self.weight, self.stride[0])
            if take:
                return _Conv3x3Fn.apply(x, self.weight, self.stride[0])
        return F.conv2d(x, self.weight, self.bias, self.stride,
                        self.padding, self.dilation, self.groups)
