#!/bin/bash
set -x
cd /root/repo

# 1. GPU suite incl. conv3x3 wgrad numerics + bf16 BN v2 port
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -2

# 2. 3x3 microbench with the wgrad A/B columns
BENCH_3X3=1 timeout 350 python tools/conv_microbench.py --batch 512 --iters 8 --with-3x3 2>&1 | tail -9

# 3. bf16 bench (bf16 BN v2 port) — capture full JSON
timeout 240 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype bf16 --batch-per-gpu 1024 2>/dev/null | tail -1

# 4. fp32 headline confirm
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null | tail -1
