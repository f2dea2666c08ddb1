#!/usr/bin/env python3
"""Per-shape microbench: MFMA conv1x1 kernels vs MIOpen (F.conv2d), fwd and
bwd, on the ResNet-50 bs=512 1x1 shapes.  Also checks numerics vs MIOpen.
Run on an MI355X:  python tools/conv_microbench.py [--batch 512]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "NORMAL")
os.environ["BYOL_MFMA_CONV1X1"] = "1"

import torch
import torch.nn.functional as F

# (H, Cin, Cout) stride-1 1x1 shapes in ResNet-50 @224
SHAPES = [
    (56, 64, 64), (56, 64, 256), (56, 256, 64), (56, 256, 128),
    (28, 128, 512), (28, 512, 128), (28, 512, 256),
    (14, 256, 1024), (14, 1024, 256), (14, 1024, 512),
    (7, 512, 2048), (7, 2048, 512),
]
# (H_in, C, stride) 3x3 shapes in ResNet-50 @224 (Cin == Cout)
SHAPES_3X3 = [
    (56, 64, 1), (56, 128, 2), (28, 128, 1), (28, 256, 2),
    (14, 256, 1), (14, 512, 2), (7, 512, 1),
]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--iters", type=int, default=10)
    args, _ = ap.parse_known_args()
    from byol_amd.ops import require_extension
    from byol_amd.ops.conv import _Conv1x1Fn
    require_extension("microbench")
    torch.manual_seed(0)
    print(f"{'shape':>22} {'miopen f':>9} {'mfma f':>9} {'miopen b':>9} "
          f"{'mfma b':>9}  maxerr_f maxerr_dx maxerr_dw")
    for hgt, cin, cout in SHAPES:
        b = args.batch
        x = torch.randn(b, cin, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last)
        w = torch.randn(cout, cin, 1, 1, device="cuda") * 0.05

        # numerics
        x1 = x.clone().requires_grad_(True)
        w1 = w.clone().requires_grad_(True)
        y1 = F.conv2d(x1, w1)
        g = torch.randn_like(y1).to(memory_format=torch.channels_last)
        y1.backward(g)
        x2 = x.clone().requires_grad_(True)
        w2 = w.clone().requires_grad_(True)
        y2 = _Conv1x1Fn.apply(x2, w2)
        y2.backward(g)
        ef = (y1 - y2).abs().max().item()
        edx = (x1.grad - x2.grad).abs().max().item()
        edw = (w1.grad - w2.grad).abs().max().item() / max(
            w1.grad.abs().max().item(), 1e-6)

        # perf: forward (wrapper) + raw kernel (no alloc/transpose)
        tm_f = timeit(lambda: F.conv2d(x, w), args.iters)
        to_f = timeit(lambda: _Conv1x1Fn.apply(x, w), args.iters)
        C = require_extension("raw")
        m = b * hgt * hgt
        xr = x.permute(0, 2, 3, 1).reshape(-1, cin)
        wv = w.reshape(cout, cin).contiguous()
        wt = wv.t().contiguous()
        yr = torch.empty(m, cout, device="cuda")
        t_raw = timeit(lambda: C.conv1x1_fwd(xr, wv, wt, yr, m, cin, cout),
                       args.iters)

        # raw backward split: our dgrad / wgrad kernels vs MIOpen's split
        # convolution_backward (where exactly does backward lose?)
        dyr = g.permute(0, 2, 3, 1).reshape(-1, cout).contiguous()
        dxr = torch.empty(m, cin, device="cuda")
        t_dg = timeit(lambda: C.conv1x1_dgrad(dyr, wv, dxr, m, cout, cin),
                      args.iters)
        dwr = torch.zeros(cout, cin, device="cuda")
        def wg():
            dwr.zero_()
            C.conv1x1_wgrad(dyr, xr, dwr, m, cout, cin)
        t_wg = timeit(wg, args.iters)
        w4 = w.reshape(cout, cin, 1, 1)
        t_dg_m = timeit(lambda: torch.ops.aten.convolution_backward(
            g, x, w4, [0], [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
            [True, False, False]), args.iters)
        t_wg_m = timeit(lambda: torch.ops.aten.convolution_backward(
            g, x, w4, [0], [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
            [False, True, False]), args.iters)

        # perf: full fwd+bwd
        def bwd_miopen():
            xr = x.detach().requires_grad_(True)
            wr = w.detach().requires_grad_(True)
            F.conv2d(xr, wr).backward(g)

        def bwd_ours():
            xr = x.detach().requires_grad_(True)
            wr = w.detach().requires_grad_(True)
            _Conv1x1Fn.apply(xr, wr).backward(g)

        tm_b = timeit(bwd_miopen, args.iters)
        to_b = timeit(bwd_ours, args.iters)
        tag = "<<" if to_f < tm_f and to_b < tm_b else ""
        print(f"H{hgt:>3} K{cin:>5} N{cout:>5} "
              f"{tm_f:9.3f} {to_f:9.3f} raw{t_raw:8.3f} {tm_b:9.3f} "
              f"{to_b:9.3f}  {ef:.2e} {edx:.2e} {edw:.2e} {tag} "
              f"dg{t_dg:7.3f}/{t_dg_m:7.3f} wg{t_wg:7.3f}/{t_wg_m:7.3f}")


def main_3x3(batch=512, iters=8):
    from byol_amd.ops import require_extension
    from byol_amd.ops.conv import _Conv3x3Fn, _rows
    C = require_extension("3x3 bench")
    print(f"{'3x3 shape':>22} {'miopen f':>9} {'mfma f':>9} "
          f"{'wg ours':>8} {'wg miopen':>9}")
    for hgt, c, stride in SHAPES_3X3:
        x = torch.randn(batch, c, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last)
        w = torch.randn(c, c, 3, 3, device="cuda") * 0.05
        tm = timeit(lambda: F.conv2d(x, w, stride=stride, padding=1), iters)
        to = timeit(lambda: _Conv3x3Fn.apply(x, w, stride), iters)
        err = (F.conv2d(x, w, stride=stride, padding=1)
               - _Conv3x3Fn.apply(x, w, stride)).abs().max().item()
        # 3x3 wgrad A/B: ours (9-tap v3) vs MIOpen wrw-only
        ho = (hgt + 2 - 3) // stride + 1
        y = F.conv2d(x, w, stride=stride, padding=1)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        b = batch
        xpad = torch.empty(b * (hgt + 2) * (hgt + 2) * c, device="cuda")
        C.pad_nhwc(_rows(x, c), xpad, b, hgt, hgt, c)
        dw9 = torch.zeros(9 * c * c, device="cuda")
        dw = torch.empty(c, c, 3, 3, device="cuda")
        def wg_ours():
            dw9.zero_()
            C.conv3x3_wgrad(_rows(g, c), xpad, dw9, dw.reshape(-1), b, hgt,
                            hgt, ho, ho, c, c, stride)
        t_wg = timeit(wg_ours, iters)
        t_wg_m = timeit(lambda: torch.ops.aten.convolution_backward(
            g, x, w, [0], [stride, stride], [1, 1], [1, 1], False, [0, 0],
            1, [False, True, False]), iters)
        # stride-1 dgrad via fwd kernel vs MIOpen dgrad-only
        t_dgo = t_dgm = float("nan")
        if stride == 1:
            wpd = w.flip([2, 3]).reshape(c, c, 9).permute(2, 0, 1) \
                .contiguous()
            gpad = torch.empty(b * (hgt + 2) * (hgt + 2) * c,
                               device="cuda")
            dxo = torch.empty_like(x)
            def dg_ours():
                C.pad_nhwc(_rows(g, c), gpad, b, hgt, hgt, c)
                C.conv3x3_fwd_fast(gpad, wpd, _rows(dxo, c), b, hgt, hgt,
                                   hgt, hgt, c, c, 1)
            t_dgo = timeit(dg_ours, iters)
            t_dgm = timeit(lambda: torch.ops.aten.convolution_backward(
                g, x, w, [0], [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                [True, False, False]), iters)
            dx_ref, _, _ = torch.ops.aten.convolution_backward(
                g, x, w, [0], [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                [True, False, False])
            dg_ours()
            dgerr = (dxo - dx_ref).abs().max().item()
            assert dgerr < 1e-3, (hgt, c, dgerr)
        _, dw_ref, _ = torch.ops.aten.convolution_backward(
            g, x, w, [0], [stride, stride], [1, 1], [1, 1], False, [0, 0],
            1, [False, True, False])
        wgerr = (dw - dw_ref).abs().max().item() / max(
            dw_ref.abs().max().item(), 1e-6)
        tag = "<<" if to < tm else ""
        print(f"H{hgt:>3} C{c:>5} s{stride} {tm:9.3f} {to:9.3f} "
              f"{t_wg:8.3f} {t_wg_m:9.3f} dg {t_dgo:7.3f}/{t_dgm:7.3f}  "
              f"err={err:.2e} wgerr={wgerr:.2e} {tag}")


if __name__ == "__main__":
    main()
    if "--with-3x3" in sys.argv or os.environ.get("BENCH_3X3"):
        main_3x3()
