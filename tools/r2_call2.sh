#!/bin/bash
# Round-2 GPU call 2: hipGraph validation + A/B, bf16 default BN bench,
# pipeline-mode bench, aug v1/v2 A/B.
set -x
cd /root/repo

# 1. full GPU suite incl. hipGraph parity + RCCL smoke + promoted tests
timeout 500 python -m pytest tests -q -m gpu 2>&1 | tail -4

# 2. fp32 headline: hipGraph on (auto) vs off
timeout 300 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tail -1
timeout 300 python bench.py --gpus 1 --steps 10 --warmup 3 --hip-graph off 2>&1 | tail -1

# 3. bf16 bs=1024 (bf16 BN now default) graph on/off
timeout 300 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype bf16 --batch-per-gpu 1024 2>&1 | tail -1
timeout 300 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype bf16 --batch-per-gpu 1024 --hip-graph off 2>&1 | tail -1

# 4. whole-pipeline timed region (loader-fed)
timeout 300 python bench.py --gpus 1 --steps 8 --warmup 2 --timed-region pipeline 2>&1 | tail -1

# 5. aug v1 vs v2 perf
timeout 200 python - <<'PY'
import torch, time, os
torch.cuda.init()
x = torch.rand(512, 3, 256, 256, device="cuda")
for v2 in ("0", "1"):
    os.environ["BYOL_AUG_V2"] = v2
    from byol_amd.data.gpu_augment import GPUTwoViewAugment
    aug = GPUTwoViewAugment(224, seed=1)
    for _ in range(3):
        aug(x)
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(10):
        aug(x)
    torch.cuda.synchronize()
    print(f"aug v2={v2}: {(time.perf_counter()-t)/10*1000:.2f} ms per 2-view bs=512 call")
PY
