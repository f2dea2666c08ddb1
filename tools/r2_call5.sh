#!/bin/bash
set -x
cd /root/repo

# 1. regression with BN v2 + wgrad v3 defaults
timeout 400 python -m pytest tests -q -m gpu 2>&1 | grep -E "passed|failed" | tail -2

# 2. headline bench (cumulative: conv dispatch + BN v2 + wgrad routing)
timeout 240 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tail -1

# 3. bf16 config
timeout 240 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype bf16 --batch-per-gpu 1024 2>&1 | tail -1

# 4. kernel profile of the new step
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace -d /tmp/prof -- bash -c "cd $GRAFT_REPO_ROOT && python bench.py --gpus 1 --steps 3 --warmup 2" > /tmp/prof_run.log 2>&1
cd "$GRAFT_REPO_ROOT"
DB=$(find /tmp/prof -name "*results.db" | head -1)
python tools/analyze_prof.py "$DB" 45 2>&1 | tee gpurun_out/r2_profile_kernels.txt | head -50
