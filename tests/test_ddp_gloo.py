"""Multi-process distributed correctness on CPU/gloo (world_size=2):
FlatDDP gradient averaging == single-process full-batch gradients, and
custom SyncBatchNorm == single-process full-batch BatchNorm."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn

from byol_amd.models.byol import BYOL

WORLD = 2


def _init(rank, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _small_byol(seed=0):
    torch.manual_seed(seed)
    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=8, classifier_output_size=3,
             total_training_steps=10, head_latent_size=16)
    return m


def _ddp_grad_worker(rank, port, out_dir):
    _init(rank, port)
    from byol_amd.parallel.ddp import FlatDDP
    model = _small_byol(seed=rank)  # different seeds; broadcast must align
    model.finalize()
    ddp = FlatDDP(model, bucket_cap_mb=0.05)
    torch.manual_seed(123)
    x1 = torch.rand(4, 3, 32, 32)
    x2 = torch.rand(4, 3, 32, 32)
    # shard the batch across ranks
    a1, a2 = x1[rank * 2:(rank + 1) * 2], x2[rank * 2:(rank + 1) * 2]
    ddp.train()
    out = ddp(a1, a2)
    loss = out["online_prediction1"].square().mean() + \
        out["linear_preds"].square().mean()
    model.flat_space.zero_grads()
    loss.backward()
    ddp.finish_grad_sync()
    torch.save({"grads": model.flat_space.flat_grads,
                "params": model.flat_space.flat_params},
               os.path.join(out_dir, f"rank{rank}.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_flat_ddp_grads_match_full_batch(tmp_path):
    port = 29511
    mp.spawn(_ddp_grad_worker, args=(port, str(tmp_path)), nprocs=WORLD,
             join=True)
    g0 = torch.load(tmp_path / "rank0.pt", weights_only=False)
    g1 = torch.load(tmp_path / "rank1.pt", weights_only=False)
    # ranks agree after all-reduce
    assert torch.allclose(g0["grads"], g1["grads"], atol=1e-6)
    assert torch.equal(g0["params"], g1["params"])  # broadcast worked

    # single-process oracle: same model (seed 0 = rank 0 pre-broadcast),
    # mean of per-shard losses == average of shard grads
    model = _small_byol(seed=0)
    model.finalize()
    model.train()
    torch.manual_seed(123)
    x1 = torch.rand(4, 3, 32, 32)
    x2 = torch.rand(4, 3, 32, 32)
    total = 0
    for r in range(WORLD):
        mr = _small_byol(seed=0)
        mr.finalize()
        mr.train()
        out = mr(x1[r * 2:(r + 1) * 2], x2[r * 2:(r + 1) * 2])
        loss = out["online_prediction1"].square().mean() + \
            out["linear_preds"].square().mean()
        mr.flat_space.zero_grads()
        loss.backward()
        total = total + mr.flat_space.flat_grads / WORLD
    assert torch.allclose(g0["grads"], total, atol=1e-5), \
        (g0["grads"] - total).abs().max()


def _syncbn_worker(rank, port, out_dir):
    _init(rank, port)
    from byol_amd.parallel.sync_bn import SyncBatchNorm
    torch.manual_seed(0)
    bn = SyncBatchNorm(6)
    torch.manual_seed(42)
    x_full = torch.randn(8, 6, 5, 5)
    x = x_full[rank * 4:(rank + 1) * 4].clone().requires_grad_(True)
    y = bn(x)
    # upstream grad: deterministic function of the FULL batch position
    gy = torch.linspace(-1, 1, y.numel()).view_as(y) + rank
    y.backward(gy)
    torch.save({"y": y.detach(), "gx": x.grad,
                "gw": bn.weight.grad, "gb": bn.bias.grad,
                "rm": bn.running_mean, "rv": bn.running_var},
               os.path.join(out_dir, f"bn{rank}.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sync_bn_matches_full_batch_bn(tmp_path):
    port = 29512
    mp.spawn(_syncbn_worker, args=(port, str(tmp_path)), nprocs=WORLD,
             join=True)
    r0 = torch.load(tmp_path / "bn0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "bn1.pt", weights_only=False)

    # oracle: plain BatchNorm2d over the concatenated batch
    torch.manual_seed(0)
    bn = nn.BatchNorm2d(6)
    torch.manual_seed(42)
    x_full = torch.randn(8, 6, 5, 5).requires_grad_(True)
    y = bn(x_full)
    gy0 = torch.linspace(-1, 1, 4 * 6 * 25).view(4, 6, 5, 5)
    gy = torch.cat([gy0, gy0 + 1], 0)
    y.backward(gy)

    got_y = torch.cat([r0["y"], r1["y"]], 0)
    assert torch.allclose(got_y, y.detach(), atol=1e-5)
    got_gx = torch.cat([r0["gx"], r1["gx"]], 0)
    assert torch.allclose(got_gx, x_full.grad, atol=1e-5)
    # per-rank dw/db are LOCAL sums; their mean (DDP semantics) equals the
    # full-batch grad / world
    assert torch.allclose((r0["gw"] + r1["gw"]) / 2, bn.weight.grad / 2,
                          atol=1e-4)
    assert torch.allclose(r0["rm"], bn.running_mean, atol=1e-5)
    assert torch.allclose(r0["rv"], bn.running_var, atol=1e-4)


def _stress4_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=4)
    from byol_amd import layers
    from byol_amd.optim.lars import LARS
    from byol_amd.parallel.ddp import FlatDDP

    model = _small_byol(seed=rank)
    model.finalize()
    ddp = FlatDDP(model, bucket_cap_mb=0.02)  # many tiny buckets
    inner = torch.optim.SGD(layers.add_weight_decay(model, 1e-6), lr=0.05,
                            momentum=0.9)
    opt = LARS(inner, eps=0.0)
    opt.attach_flat_space(model.flat_space)
    ddp.train()
    torch.manual_seed(7)
    x1 = torch.rand(8, 3, 32, 32)
    x2 = torch.rand(8, 3, 32, 32)
    for step in range(3):
        a1 = x1[rank * 2:(rank + 1) * 2]
        a2 = x2[rank * 2:(rank + 1) * 2]
        out = ddp(a1, a2)
        loss = out["online_prediction1"].square().mean() + \
            out["linear_preds"].square().mean()
        opt.zero_grad()
        loss.backward()
        ddp.finish_grad_sync()
        opt.step()
    torch.save(model.flat_space.flat_params.clone(),
               os.path.join(out_dir, f"p{rank}.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_flat_ddp_four_rank_stress(tmp_path):
    """4 ranks x 3 full steps (forward+backward+LARS) over many tiny
    buckets: parameters must stay bit-identical across replicas (the
    replicated-state invariant the 8-GPU run depends on)."""
    port = 29517
    mp.spawn(_stress4_worker, args=(port, str(tmp_path)), nprocs=4,
             join=True)
    ps = [torch.load(os.path.join(str(tmp_path), f"p{r}.pt"),
                     weights_only=True) for r in range(4)]
    for r in range(1, 4):
        assert torch.equal(ps[0], ps[r]), \
            f"rank {r} diverged: max|d|=" \
            f"{(ps[0] - ps[r]).abs().max().item()}"
