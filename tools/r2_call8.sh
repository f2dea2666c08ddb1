#!/bin/bash
set -x
cd /root/repo

# 1. 3x3 microbench: dg columns (dgrad-via-fwd A/B) + wgrad with block sweep
BENCH_3X3=1 timeout 300 python - <<'PY' 2>&1 | tail -10
import sys
sys.argv = ["x", "--batch", "512", "--iters", "8"]
sys.path.insert(0, "tools")
import conv_microbench as cm
cm.main_3x3(batch=512, iters=8)
PY

# 2. wgrad block-cap sweep on two losing shapes
for cap in 2048 8192; do
  BYOL_WGRAD_BLOCKS=$cap timeout 200 python - <<PY 2>&1 | tail -3
import sys, torch, time
sys.path.insert(0, ".")
from byol_amd.ops import require_extension
C = require_extension("x")
for (h, k, n) in [(28, 512, 128), (7, 2048, 512)]:
    m = 512 * h * h
    dy = torch.randn(m, n, device="cuda")
    x = torch.randn(m, k, device="cuda")
    dw = torch.zeros(n, k, device="cuda")
    for _ in range(3):
        C.conv1x1_wgrad(dy, x, dw, m, n, k)
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(8):
        C.conv1x1_wgrad(dy, x, dw, m, n, k)
    torch.cuda.synchronize()
    print(f"cap=$cap shape {h}/{k}/{n}: {(time.perf_counter()-t)/8*1000:.3f} ms")
PY
done

# 3. final regression: full GPU suite + headline bench
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -1
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null | tail -1
