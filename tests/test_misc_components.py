"""Coverage for the smaller components: LAMB, polyak helpers, grapher,
PhaseTimer, launch-mode detection, FlatDDP buffer-broadcast parity flag."""

import json
import os

import pytest
import torch
import torch.nn as nn


def test_lamb_single_step_matches_manual():
    from byol_amd.optim.lamb import LAMB
    torch.manual_seed(0)
    p = nn.Parameter(torch.randn(4, 3))
    opt = LAMB([p], lr=0.1, betas=(0.9, 0.999), eps=1e-6, weight_decay=0.01)
    g = torch.randn(4, 3)
    p.grad = g.clone()
    p0 = p.detach().clone()
    opt.step()
    # manual first step
    exp_avg = 0.1 * g
    exp_sq = 0.001 * g * g
    update = (exp_avg / 0.1) / ((exp_sq / 0.001).sqrt() + 1e-6)
    update = update + 0.01 * p0
    trust = p0.norm() / update.norm()
    want = p0 - 0.1 * trust * update
    assert torch.allclose(p.detach(), want, atol=1e-6)


def test_polyak_helpers_swap_restore():
    from byol_amd.layers import get_polyak_prediction, polyak_ema_parameters
    torch.manual_seed(1)
    m = nn.Linear(3, 3)
    polyak_ema_parameters(m, 0.5)  # init shadow = current
    with torch.no_grad():
        for p in m.parameters():
            p.add_(1.0)
    polyak_ema_parameters(m, 0.5)  # shadow = 0.5*old + 0.5*new
    cur = [p.detach().clone() for p in m.parameters()]
    x = torch.randn(2, 3)
    with torch.no_grad():
        direct = m(x)
    out = get_polyak_prediction(m, lambda: m(x))
    assert not torch.allclose(out, direct)  # used averaged weights
    for p, c in zip(m.parameters(), cur):
        assert torch.equal(p.detach(), c)  # restored


def test_grapher_jsonl_backend(tmp_path):
    from byol_amd.engine.grapher import Grapher
    g = Grapher("jsonl", logdir=str(tmp_path))
    g.add_scalar("train_loss", 1.5, 3)
    g.add_text("config", "hello", 0)
    g.add_image("imgs", torch.rand(3, 4, 4), 0)
    g.save()
    g.close()
    lines = [json.loads(x) for x in
             open(tmp_path / "events.jsonl").read().splitlines()]
    kinds = {x["kind"] for x in lines}
    assert kinds == {"scalar", "text", "image"}
    assert lines[0]["value"] == 1.5


def test_phase_timer_cpu():
    from byol_amd.profiling import PhaseTimer
    t = PhaseTimer(enabled=True, use_cuda=False)
    t.start("data")
    t.start("forward")  # implicit stop of data
    t.stop()
    t.add_samples(8)
    line = t.epoch_summary()
    assert "data=" in line and "forward=" in line and "images/sec=" in line


def test_launch_torchrun_env_detection(monkeypatch):
    from byol_amd.parallel.launch import launch
    seen = {}

    def fake_run(rank, args):
        seen["rank"] = rank
        seen["world"] = args.num_replicas

    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("WORLD_SIZE", "8")
    monkeypatch.setenv("LOCAL_RANK", "3")
    import types
    args = types.SimpleNamespace(num_replicas=1, distributed_rank=0)
    launch(fake_run, args)
    assert seen == {"rank": 3, "world": 8}


def _bcast_worker(rank):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29533"
    dist.init_process_group("gloo", rank=rank, world_size=2)
    from byol_amd.models.byol import BYOL
    from byol_amd.parallel.ddp import FlatDDP
    torch.manual_seed(rank)
    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=8, classifier_output_size=3,
             total_training_steps=10, head_latent_size=16).finalize()
    ddp = FlatDDP(m, broadcast_buffers=True)
    # desync a buffer on rank 1; forward must re-sync it from rank 0
    if rank == 1:
        with torch.no_grad():
            m.base_network.bn1.running_mean.fill_(42.0)
    ddp.train()
    ddp(torch.rand(2, 3, 32, 32), torch.rand(2, 3, 32, 32))
    rm = m.base_network.bn1.running_mean
    assert float(rm.abs().max()) < 40.0, "buffer broadcast did not run"
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_flat_ddp_broadcast_buffers_flag():
    """broadcast_buffers=True (reference-parity mode) re-broadcasts BN
    running stats every forward."""
    import torch.multiprocessing as mp
    mp.spawn(_bcast_worker, nprocs=2, join=True)


def test_conv3x3_auto_table_parsing():
    from byol_amd.ops.conv import _parse_3x3_table
    assert _parse_3x3_table("") == set()
    assert _parse_3x3_table("128/128/2,64/64/1") == {(128, 128, 2),
                                                     (64, 64, 1)}
    assert _parse_3x3_table(" 256/256/1 ") == {(256, 256, 1)}


def test_conv_dispatch_tables_consistent():
    """Every wgrad-table shape is also dgrad-routed (the backward branch
    assumes it), and tables only contain MFMA-eligible geometry."""
    from byol_amd.ops.conv import _AUTO_DGRAD, _AUTO_SHAPES, _AUTO_WGRAD
    for k, n in _AUTO_SHAPES | _AUTO_DGRAD | _AUTO_WGRAD:
        assert k % 32 == 0 and n % 32 == 0, (k, n)
    for k, n in _AUTO_WGRAD:
        assert n % 64 == 0 and k % 64 == 0, (k, n)


def test_half_dtype_fp16_cpu_plumbing(tmp_path):
    """--half --half-dtype fp16 on CPU: scaler stays disabled, the engine
    runs a debug step without error (the fp16 math itself is GPU-only)."""
    import byol_amd.config as config
    from byol_amd.engine import trainer

    args = config.parse_args([
        "--task", "synthetic_multi_augment_image_folder",
        "--arch", "resnet18", "--representation-size", "512",
        "--batch-size", "8", "--epochs", "1", "--num-replicas", "1",
        "--no-cuda", "--half", "--half-dtype", "fp16", "--debug-step",
        "--image-size-override", "32", "--synthetic-classes", "4",
        "--workers-per-replica", "0",
        "--model-dir", str(tmp_path / "m"),
        "--log-dir", str(tmp_path / "l"),
    ])
    args.cuda = False
    args.distributed_rank = 0
    loader, model, grapher = trainer.build_loader_model_grapher(args)
    from byol_amd.optim import build_optimizer
    opt, sched = build_optimizer(model, args)
    trainer.train(1, model, opt, loader.train_loader, grapher, args)


def test_autotuner_caches_and_persists(tmp_path):
    """Decision caching + JSON persistence with an injected fake timer."""
    from byol_amd.ops.autotune import Autotuner

    calls = []

    def fake_timer(fn):
        calls.append(fn)
        fn()
        return 1.0 if getattr(fn, "_fast", False) else 2.0

    path = str(tmp_path / "tune.json")
    tun = Autotuner(cache_path=path, timer=fake_timer)

    def ours():
        pass
    ours._fast = True

    def theirs():
        pass

    assert tun.choose(("c1f", 64, 64, 1024), ours, theirs) is True
    assert len(calls) == 2
    # cached: no more timing
    assert tun.choose(("c1f", 64, 64, 1024), ours, theirs) is True
    assert len(calls) == 2

    # persisted decisions reload
    tun2 = Autotuner(cache_path=path, timer=fake_timer)
    assert tun2.decisions[("c1f", 64, 64, 1024)] is True
    assert tun2.choose(("c1f", 64, 64, 1024), theirs, ours) is True
    assert len(calls) == 2


def test_conv_dispatch_flag_sets_env(monkeypatch):
    import os

    import byol_amd.config as config

    monkeypatch.delenv("BYOL_MFMA_CONV1X1", raising=False)
    monkeypatch.delenv("BYOL_MFMA_CONV3X3", raising=False)
    config.parse_args(["--conv-dispatch", "autotune"])
    assert os.environ["BYOL_MFMA_CONV1X1"] == "autotune"
    assert os.environ["BYOL_MFMA_CONV3X3"] == "autotune"
    config.parse_args(["--conv-dispatch", "force"])
    assert os.environ["BYOL_MFMA_CONV1X1"] == "1"


def test_topk_metric_cpu_oracle():
    """topk vs a brute-force ranking check, incl. k > num_classes clamp."""
    import torch

    from byol_amd.engine import metrics

    torch.manual_seed(3)
    logits = torch.randn(64, 10)
    labels = torch.randint(10, (64,))
    t1, t5 = metrics.topk(logits, labels, topk=(1, 5))
    order = logits.argsort(dim=1, descending=True)
    hit1 = (order[:, 0] == labels).float().mean() * 100
    hit5 = (order[:, :5] == labels[:, None]).any(dim=1).float().mean() * 100
    assert torch.isclose(t1, hit1)
    assert torch.isclose(t5, hit5)
    # tiny class count: k clamps to num_classes
    small = torch.randn(8, 3)
    lab = torch.randint(3, (8,))
    _, t5s = metrics.topk(small, lab, topk=(1, 5))
    assert t5s == 100.0
