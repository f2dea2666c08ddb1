#!/bin/bash
set -x
cd /root/repo

# 1. regression: full GPU suite with the new dispatch
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -2

# 2. wgrad v3 A/B (wg column: ours/miopen)
BYOL_WGRAD=v3 timeout 350 python tools/conv_microbench.py --batch 512 --iters 8 2>&1 | tail -14

# 3. BN stats/bwd-reduce v1 vs v2 sweep
timeout 300 python tools/bn_microbench.py 2>&1 | tail -10

# 4. headline bench with new dispatch (fwd fast64 shapes + our dgrad)
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tail -1

# 5. ResNet-200 re-bench (BASELINE config 5)
timeout 300 python bench.py --gpus 1 --steps 5 --warmup 2 --arch resnet200 --batch-per-gpu 256 2>&1 | tail -1

# 6. kernel-level evidence for the new dispatch
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof -- bash -c "cd $GRAFT_REPO_ROOT && python bench.py --gpus 1 --steps 3 --warmup 2" > /tmp/prof_run.log 2>&1
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
grep -E "conv1x1|conv3x3|bn_|igemm|naive|ck|flat_ema|lars_|byol_loss|ce_topk|aug_|Cijk|elementwise" /tmp/prof_run.log | head -5
find /tmp/prof -name "*stats*" -o -name "*.csv" 2>/dev/null | head -5
for f in $(find /tmp/prof -name "*kernel_stats*" 2>/dev/null | head -2); do
  cp "$f" gpurun_out/; done
tail -60 /tmp/prof_run.log | head -20
