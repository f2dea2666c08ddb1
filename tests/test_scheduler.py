"""Warmup->cosine scheduling semantics (/root/reference/optimizers/scheduler.py)."""

import math
import types

import pytest
import torch
import torch.nn as nn

from byol_amd.optim.builder import build_lr_schedule, build_optimizer


def make_args(**kw):
    base = dict(lr=0.2, lr_update_schedule="cosine", warmup=3, epochs=13,
                optimizer="lars_momentum", weight_decay=1e-6,
                batch_size=64, num_replicas=1)
    base.update(kw)
    return types.SimpleNamespace(**base)


def test_warmup_ramps_then_cosine():
    model = nn.Linear(4, 4)
    args = make_args()
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    sched = build_lr_schedule(opt, args)
    lrs = [opt.param_groups[0]["lr"]]
    for _ in range(args.epochs):
        sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    # warmup: 0, 1/3, 2/3 then cosine from 1.0 over T_max = epochs - warmup
    assert lrs[0] == pytest.approx(0.0)
    assert lrs[1] == pytest.approx(1.0 / 3)
    assert lrs[2] == pytest.approx(2.0 / 3)
    assert lrs[3] == pytest.approx(1.0)
    t_max = args.epochs - args.warmup
    # first cosine step after handoff
    want = 0.5 * (1 + math.cos(math.pi * 1 / t_max))
    assert lrs[4] == pytest.approx(want, rel=1e-5)


def test_fixed_schedule():
    model = nn.Linear(4, 4)
    args = make_args(lr_update_schedule="fixed", warmup=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.5)
    sched = build_lr_schedule(opt, args)
    for _ in range(5):
        sched.step()
    assert opt.param_groups[0]["lr"] == pytest.approx(0.5)


def test_linear_lr_scaling_for_momentum_family():
    model = nn.Linear(4, 4)
    args = make_args(batch_size=64, num_replicas=8)  # global 512
    opt, _ = build_optimizer(model, args)
    # lr = 0.2 * 512/256 = 0.4
    assert opt.param_groups[0]["initial_lr"] == pytest.approx(0.4)


def test_no_scaling_for_adam():
    model = nn.Linear(4, 4)
    args = make_args(optimizer="adam", batch_size=64, num_replicas=8)
    opt, _ = build_optimizer(model, args)
    assert opt.param_groups[0]["initial_lr"] == pytest.approx(0.2)


def test_scheduler_state_roundtrip():
    """Resume restores optimizer AND scheduler state (the checkpoint bundle
    does both — CosineAnnealingLR's recursive formula needs the optimizer's
    current lr)."""
    model = nn.Linear(4, 4)
    args = make_args()
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    sched = build_lr_schedule(opt, args)
    for _ in range(5):
        sched.step()
    sd = sched.state_dict()
    opt_sd = opt.state_dict()

    # order matters (and matches CheckpointBundle.load): construct the
    # schedulers first (construction mutates the optimizer's lr), then
    # restore optimizer state, then scheduler state
    opt2 = torch.optim.SGD(nn.Linear(4, 4).parameters(), lr=1.0)
    sched2 = build_lr_schedule(opt2, make_args())
    opt2.load_state_dict(opt_sd)
    sched2.load_state_dict(sd)
    assert sched2.warmup.complete == sched.warmup.complete
    sched.step()
    sched2.step()
    assert sched2.get_last_lr() == pytest.approx(sched.get_last_lr())
    assert opt2.param_groups[0]["lr"] == pytest.approx(
        opt.param_groups[0]["lr"])


def test_scheduler_resume_mid_warmup():
    """Resume inside the warmup phase: the rewritten LinearWarmup must
    restore its step counter and keep ramping from where it stopped."""
    model = nn.Linear(4, 4)
    args = make_args(warmup=4, epochs=12)
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    sched = build_lr_schedule(opt, args)
    sched.step()  # -> 1/4
    sd = sched.state_dict()
    opt_sd = opt.state_dict()

    opt2 = torch.optim.SGD(model.parameters(), lr=1.0)
    sched2 = build_lr_schedule(opt2, args)
    opt2.load_state_dict(opt_sd)
    sched2.load_state_dict(sd)
    assert sched2.warmup.complete is False
    assert sched2.warmup.last_epoch == 1
    sched2.step()
    assert opt2.param_groups[0]["lr"] == pytest.approx(2.0 / 4)
    sched2.step()
    assert opt2.param_groups[0]["lr"] == pytest.approx(3.0 / 4)
    sched2.step()
    assert opt2.param_groups[0]["lr"] == pytest.approx(1.0)
    assert sched2.warmup.complete is True
