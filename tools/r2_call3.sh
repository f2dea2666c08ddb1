#!/bin/bash
set -x
cd /root/repo

# 1. full GPU suite (incl. blur, graph parity, RCCL smoke, fast64 numerics
#    via forced-mode conv tests)
timeout 500 python -m pytest tests -q -m gpu 2>&1 | tail -6

# 2. conv microbench: base (fast64 engages on N=64 shapes now)
BENCH_3X3=1 timeout 400 python tools/conv_microbench.py --batch 512 --iters 8 --with-3x3 2>&1 | tail -24

# 3. XCD swizzle variant
BYOL_CONV_XSWZ=1 BENCH_3X3=1 timeout 350 python tools/conv_microbench.py --batch 512 --iters 8 --with-3x3 2>&1 | tail -24

# 4. setprio variant
BYOL_CONV_PRIO=1 BENCH_3X3=1 timeout 350 python tools/conv_microbench.py --batch 512 --iters 8 --with-3x3 2>&1 | tail -24

# 5. BN stats v1 vs v2 grid sweep
timeout 300 python tools/bn_microbench.py 2>&1 | tail -10

# 6. pipeline-region bench with aug v2 + HIP blur defaults
timeout 300 python bench.py --gpus 1 --steps 8 --warmup 2 --timed-region pipeline 2>&1 | tail -1
