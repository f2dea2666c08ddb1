#!/bin/bash
set -x
cd /root/repo
# 1. suite with the fixed nopad kernel
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -1
# 2. 3x3 A/B with correct staging
BENCH_3X3=1 timeout 250 python - <<'PY' 2>&1 | tail -9
import sys
sys.path.insert(0, "tools")
import conv_microbench as cm
cm.main_3x3(batch=512, iters=10)
PY
# 3. headline with tentative 3x3 flips (the two expected parity-or-better
#    shapes) vs default
BYOL_CONV3X3_AUTO="128/128/2,128/128/1" timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null | tail -1
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null | tail -1
