#!/bin/bash
set -x
cd /root/repo
# 1. quick regression
timeout 300 python -m pytest tests/test_ops_gpu.py tests/test_fused_bn.py -q -m gpu 2>&1 | grep -E "passed|failed" | tail -2
# 2. BN microbench incl. apply v1/v2 A/B
timeout 420 python tools/bn_microbench.py 2>&1 | tail -22
# 3. bf16 bench with room for MIOpen bf16 finds
timeout 500 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype bf16 --batch-per-gpu 1024 2>/dev/null | tail -1
