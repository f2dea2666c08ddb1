#!/bin/bash
# Round-2 opening GPU call: validate the pre-authored round-1 candidates and
# collect the measurements that gate their dispatch.  Run via gpurun:
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/round2_validation.sh'
set -x
cd /root/repo

# 1. regression: full GPU suite (all round-1-validated kernels)
timeout 400 python -m pytest tests -q -m gpu 2>&1 | tail -3

# 2. pending candidates: wgrad v2 + conv3x3 glds fast path
BYOL_TEST_PENDING=1 timeout 300 python -m pytest tests/test_ops_gpu.py -q \
    -m gpu -k "wgrad_v2 or conv3x3_fast or aug_sample_v2 or bf16_fused_bn" 2>&1 | tail -3

# 3. per-shape measurements (1x1 incl. BYOL_WGRAD=v2 A/B, 3x3 fast/slow A/B)
timeout 400 python tools/conv_microbench.py --batch 512 --iters 8 2>&1 | tail -16
BYOL_WGRAD=v2 timeout 400 python tools/conv_microbench.py --batch 512 \
    --iters 8 2>&1 | tail -16
BENCH_3X3=1 timeout 300 python tools/conv_microbench.py --batch 512 \
    --iters 8 --with-3x3 2>&1 | tail -10
BYOL_CONV3X3_FAST=0 BENCH_3X3=1 timeout 300 python tools/conv_microbench.py \
    --batch 512 --iters 8 --with-3x3 2>&1 | tail -10

# 4. current headline number + bf16 config with the bf16 BN kernels
timeout 300 python bench.py --gpus 1 --steps 8 --warmup 3 2>&1 | tail -1
BYOL_BF16_BN=1 timeout 300 python bench.py --gpus 1 --steps 6 --warmup 2 \
    --dtype bf16 --batch-per-gpu 1024 2>&1 | tail -1
