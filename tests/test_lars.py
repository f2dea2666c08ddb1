"""LARS wrapper semantics vs a hand-written oracle of the reference math
(/root/reference/optimizers/lars.py:84-127 + torch SGD momentum)."""

import copy

import torch
import torch.nn as nn

from byol_amd import layers
from byol_amd.optim.lars import LARS


def tiny_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.BatchNorm1d(16), nn.ReLU(),
                         nn.Linear(16, 4))


def test_add_weight_decay_groups():
    m = tiny_model()
    groups = layers.add_weight_decay(m, 1e-4)
    assert len(groups) == 2
    assert groups[0]["ignore"] is False and groups[0]["weight_decay"] == 1e-4
    assert groups[1]["ignore"] is True and groups[1]["weight_decay"] == 0.0
    # 2 weight matrices adapt; 2 linear biases + BN weight/bias don't
    assert len(groups[0]["params"]) == 2
    assert len(groups[1]["params"]) == 4


def manual_lars_step(params_groups, lr, momentum, trust, eps, state):
    """Oracle: reference LARS.apply_adaptive_lrs + SGD momentum step."""
    for group in params_groups:
        wd = group["weight_decay"]
        ignore = group.get("ignore", None)
        for p in group["params"]:
            if p.grad is None:
                continue
            g = p.grad.clone()
            if wd > 0:
                g = g + wd * p.detach()
            if ignore is not None and not ignore:
                pn, gn = p.detach().norm(), g.norm()
                alr = 1.0
                if pn > 0 and gn > 0:
                    alr = float(trust * pn / (gn + eps))
                g = g * alr
            buf = state.get(id(p))
            if buf is None:
                buf = g.clone()
            else:
                buf = momentum * buf + g
            state[id(p)] = buf
            with torch.no_grad():
                p.add_(buf, alpha=-lr)


def test_lars_matches_oracle_over_steps():
    m1 = tiny_model()
    m2 = copy.deepcopy(m1)
    wd, lr, trust = 1e-4, 0.05, 0.001

    groups1 = layers.add_weight_decay(m1, wd)
    opt = LARS(torch.optim.SGD(groups1, lr=lr, momentum=0.9), eps=0.0,
               trust_coef=trust)
    groups2 = layers.add_weight_decay(m2, wd)
    state = {}

    for step in range(4):
        torch.manual_seed(100 + step)
        x = torch.randn(6, 8)
        loss1 = m1(x).square().mean()
        opt.zero_grad()
        loss1.backward()
        opt.step()

        loss2 = m2(x).square().mean()
        for p in m2.parameters():
            p.grad = None
        loss2.backward()
        manual_lars_step(groups2, lr, 0.9, trust, 0.0, state)

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_lars_state_dict_roundtrip():
    m = tiny_model()
    opt = LARS(torch.optim.SGD(layers.add_weight_decay(m, 1e-4), lr=0.1,
                               momentum=0.9), eps=0.0)
    x = torch.randn(4, 8)
    m(x).sum().backward()
    opt.step()
    sd = opt.state_dict()
    opt2 = LARS(torch.optim.SGD(layers.add_weight_decay(m, 1e-4), lr=0.1,
                                momentum=0.9), eps=0.0)
    opt2.load_state_dict(sd)
    assert len(opt2.state_dict()["state"]) == len(sd["state"])
