"""End-to-end distributed training parity (BASELINE config 3 semantics on
CPU/gloo): 2 ranks x half batch with FlatDDP + SyncBN + LARS must track a
single process training on the full batch with plain BN — global batch
statistics and averaged gradients make the two mathematically identical."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from byol_amd import layers
from byol_amd.models.byol import BYOL
from byol_amd.optim.lars import LARS

WORLD = 2
STEPS = 2


def _build(seed=0, sync=False):
    torch.manual_seed(seed)
    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=8, classifier_output_size=3,
             total_training_steps=50, head_latent_size=16)
    if sync:
        from byol_amd.parallel.sync_bn import convert_sync_batchnorm
        m = convert_sync_batchnorm(m)
    return m.finalize()


def _make_optimizer(model):
    inner = torch.optim.SGD(layers.add_weight_decay(model, 1e-4), lr=0.02,
                            momentum=0.9)
    opt = LARS(inner, eps=0.0)
    opt.attach_flat_space(model.flat_space)
    return opt


def _batches():
    torch.manual_seed(777)
    return [(torch.rand(8, 3, 32, 32), torch.rand(8, 3, 32, 32),
             torch.randint(3, (8,))) for _ in range(STEPS)]


def _train_step(model, ddp, opt, a1, a2, labels):
    """NOTE: the real BYOL loss normalizes by whole-tensor Frobenius norms
    of the LOCAL batch (reference semantics, objective.py), so rank-mean of
    losses != full-batch loss BY DESIGN; this parity test uses a
    shard-linear surrogate loss so the single-process full-batch oracle is
    mathematically identical."""
    import torch.nn.functional as F
    model.train()
    out = (ddp or model)(a1, a2)
    loss = (out["online_prediction1"].square().mean()
            + out["online_prediction2"].square().mean())
    loss = loss + F.cross_entropy(out["linear_preds"],
                                  torch.cat([labels, labels]))
    opt.zero_grad()
    loss.backward()
    if ddp is not None:
        ddp.finish_grad_sync()
    opt.step()


def _worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    from byol_amd.parallel.ddp import FlatDDP
    model = _build(seed=rank, sync=True)  # different seeds; broadcast fixes
    ddp = FlatDDP(model, bucket_cap_mb=0.25)
    opt = _make_optimizer(model)
    for a1, a2, labels in _batches():
        sh = slice(rank * 4, (rank + 1) * 4)
        _train_step(model, ddp, opt, a1[sh], a2[sh], labels[sh])
    if rank == 0:
        torch.save({"params": model.flat_space.flat_params.detach(),
                    "ema": model.target_network.mean},
                   os.path.join(out_dir, "ddp.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ddp_syncbn_training_matches_full_batch(tmp_path):
    mp.spawn(_worker, args=(29523, str(tmp_path)), nprocs=WORLD, join=True)
    got = torch.load(tmp_path / "ddp.pt", weights_only=False)

    model = _build(seed=0, sync=False)
    opt = _make_optimizer(model)
    for a1, a2, labels in _batches():
        _train_step(model, None, opt, a1, a2, labels)

    diff = (got["params"] - model.flat_space.flat_params).abs().max()
    assert diff < 2e-5, float(diff)
    ediff = (got["ema"] - model.target_network.mean).abs().max()
    assert ediff < 2e-5, float(ediff)
