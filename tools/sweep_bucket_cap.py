#!/usr/bin/env python3
"""Bucket-cap sweep for the first multi-GPU opportunity (MULTI_GPU_PLAN.md).

Run on an N-GPU node:

    for cap in 8 16 32 64 128; do
      python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
          --master-addr 127.0.0.1 tools/sweep_bucket_cap.py --cap $cap
    done

Prints ms/step (max over ranks) per cap; pick the knee and set
``--bucket-cap-mb`` / bench default accordingly.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cap", type=float, default=32.0)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch-per-gpu", type=int, default=512)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    torch.cuda.set_device(local_rank)
    dist.init_process_group("nccl")

    from byol_amd import layers
    from byol_amd.models.byol import BYOL
    from byol_amd.objective import loss_function
    from byol_amd.optim.lars import LARS
    from byol_amd.parallel.ddp import FlatDDP

    torch.manual_seed(1 + rank)
    model = BYOL(arch="resnet50", base_network_output_size=2048,
                 projection_output_size=256, classifier_output_size=1000,
                 total_training_steps=100000).cuda()
    model.base_network.to(memory_format=torch.channels_last)
    model.finalize()
    model.train()
    inner = torch.optim.SGD(layers.add_weight_decay(model, 1e-6), lr=0.1,
                            momentum=0.9)
    opt = LARS(inner, eps=0.0)
    opt.attach_flat_space(model.flat_space)
    net = FlatDDP(model, bucket_cap_mb=args.cap)

    B = args.batch_per_gpu
    a1 = torch.rand(B, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last)
    a2 = torch.rand(B, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last)

    def step():
        out = net(a1, a2)
        loss = loss_function(out["online_prediction1"].float(),
                             out["online_prediction2"].float(),
                             out["target_projection1"].float(),
                             out["target_projection2"].float())
        opt.zero_grad()
        loss.backward()
        net.finish_grad_sync()
        opt.step()

    for _ in range(args.warmup):
        step()
    dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dist.barrier()
    el = torch.tensor([time.perf_counter() - t0], device="cuda")
    dist.all_reduce(el, op=dist.ReduceOp.MAX)
    if rank == 0:
        print(f"cap={args.cap} MiB: "
              f"{float(el) / args.steps * 1000:.2f} ms/step "
              f"(world {dist.get_world_size()})")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
