#!/bin/bash
set -x
cd /root/repo
# 1. suite (nopad path now default in forced-mode 3x3 tests)
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -1
# 2. 3x3 A/B: nopad (default) vs padded vs MIOpen
BENCH_3X3=1 timeout 250 python - <<'PY' 2>&1 | tail -9
import sys
sys.path.insert(0, "tools")
import conv_microbench as cm
cm.main_3x3(batch=512, iters=8)
PY
BYOL_CONV3X3_NOPAD=0 BENCH_3X3=1 timeout 250 python - <<'PY' 2>&1 | tail -8
import sys
sys.path.insert(0, "tools")
import conv_microbench as cm
cm.main_3x3(batch=512, iters=8)
PY
# 3. final headline
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null | tail -1
