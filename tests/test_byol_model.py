"""BYOL model semantics vs the reference's pack/swap/restore design."""

import copy
import math

import pytest
import torch
import torch.nn as nn

from byol_amd.models.byol import BYOL, CosEMA


def small_byol(total_steps=10, classes=7):
    torch.manual_seed(0)
    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=32, classifier_output_size=classes,
             total_training_steps=total_steps, head_latent_size=64)
    return m.finalize()


def test_forward_returns_13_keys():
    m = small_byol()
    m.train()
    out = m(torch.rand(4, 3, 32, 32), torch.rand(4, 3, 32, 32))
    expected = {
        "linear_preds",
        "online_representation1", "online_projection1", "online_prediction1",
        "online_representation2", "online_projection2", "online_prediction2",
        "target_representation1", "target_projection1", "target_prediction1",
        "target_representation2", "target_projection2", "target_prediction2",
    }
    assert set(out.keys()) == expected
    assert out["linear_preds"].shape == (8, 7)  # both views in train
    m.eval()
    out = m(torch.rand(4, 3, 32, 32), torch.rand(4, 3, 32, 32))
    assert out["linear_preds"].shape == (4, 7)  # one view in eval


def test_total_param_count_resnet50_matches_reference():
    torch.manual_seed(0)
    m = BYOL(arch="resnet50", base_network_output_size=2048,
             projection_output_size=256, classifier_output_size=1000,
             total_training_steps=10)
    n = sum(p.numel() for p in m.parameters())
    # 37,116,456 params for the default config
    assert n == 37_116_456, n


def test_cos_ema_schedule_and_priming():
    m = small_byol(total_steps=10)
    ema = m.target_network
    # priming advanced step to 1 and mean = (1 - base_decay) * params
    assert ema.step == 1
    flat = m.flat_space.flat_params
    expected = (1 - ema.base_decay) * flat
    assert torch.allclose(ema.mean, expected, atol=1e-6)
    # decay formula at k: 1 - (1-base) * (cos(pi k / K)+1)/2
    for k in (0, 3, 10):
        ema.step = k
        want = 1 - (1 - ema.base_decay) * (
            math.cos(math.pi * k / ema.total_steps) + 1) / 2
        assert ema.current_decay() == pytest.approx(want)


def test_ema_only_steps_in_training_mode():
    m = small_byol()
    step0 = m.target_network.step
    m.eval()
    m(torch.rand(2, 3, 32, 32), torch.rand(2, 3, 32, 32))
    assert m.target_network.step == step0
    m.train()
    m(torch.rand(2, 3, 32, 32), torch.rand(2, 3, 32, 32))
    assert m.target_network.step == step0 + 1


def test_ema_update_matches_manual_blend():
    m = small_byol(total_steps=100)
    m.train()
    ema = m.target_network
    mean_before = ema.mean.clone()
    decay = ema.current_decay()
    flat_before = m.flat_space.flat_params.clone()
    m(torch.rand(2, 3, 32, 32), torch.rand(2, 3, 32, 32))
    want = (1 - decay) * flat_before + decay * mean_before
    assert torch.allclose(ema.mean, want, atol=1e-6)


def test_target_prediction_equals_pack_swap_oracle():
    """Our zero-copy functional_call target pass must equal the reference's
    vector_to_parameters swap (/root/reference/main.py:214-227) bit-for-bit
    on the same module values."""
    m = small_byol()
    m.eval()  # freeze BN stats so the oracle double-run is comparable
    x = torch.rand(4, 3, 32, 32)

    # oracle: clone model, load EMA vector into params the reference way
    oracle = copy.deepcopy(m)
    params = [p for p in oracle.parameters()]
    nn.utils.vector_to_parameters(oracle.target_network.mean, params)
    with torch.no_grad():
        want_repr = oracle.base_network(x).view(-1, 512)
        want_proj = oracle.head(want_repr)
        want_pred = oracle.predictor(want_proj)

    got_repr, got_proj, got_pred = m.target_prediction(x)
    assert torch.equal(got_repr, want_repr)
    assert torch.equal(got_proj, want_proj)
    assert torch.equal(got_pred, want_pred)


def test_target_pass_updates_bn_running_stats_in_train_mode():
    """Reference quirk preserved: target passes run BN in train mode and
    update running stats (the same module objects are used)."""
    m = small_byol()
    m.train()
    rm_before = m.base_network.bn1.running_mean.clone()
    m.target_prediction(torch.rand(4, 3, 32, 32))
    assert not torch.equal(rm_before, m.base_network.bn1.running_mean)


def test_flat_space_views_track_param_updates():
    m = small_byol()
    flat = m.flat_space.flat_params
    p = m.linear_classifier.weight
    with torch.no_grad():
        p.add_(1.0)
    name_map = dict((name, (off, n))
                    for name, off, n, _ in m.flat_space.layout)
    off, n = name_map["linear_classifier.weight"]
    assert torch.equal(flat[off:off + n].view(p.shape), p)


def test_grads_accumulate_into_flat_buffer():
    m = small_byol()
    m.train()
    out = m(torch.rand(2, 3, 32, 32), torch.rand(2, 3, 32, 32))
    loss = out["online_prediction1"].sum() + out["linear_preds"].sum()
    loss.backward()
    g = m.flat_space.flat_grads
    assert g.abs().sum() > 0
    # every param's .grad is a view of flat_grads
    for (name, off, n, shape), p in zip(m.flat_space.layout, m.parameters()):
        assert p.grad.data_ptr() == g[off:off + n].data_ptr()


def test_state_dict_roundtrip_preserves_flat_space():
    m = small_byol()
    sd = copy.deepcopy(m.state_dict())
    m2 = small_byol()
    with torch.no_grad():
        m2.flat_space.flat_params.mul_(2.0)
    m2.load_state_dict(sd)
    assert torch.allclose(m2.flat_space.flat_params,
                          m.flat_space.flat_params)
    # still views after load
    p = m2.linear_classifier.weight
    assert p.data_ptr() >= m2.flat_space.flat_params.data_ptr()
    assert m2.target_network.step == m.target_network.step


def test_flat_space_verify_catches_detached_params():
    """Module.to(memory_format=...) AFTER finalize() rebinds param.data off
    the flat buffer; verify() must catch it (the fused EMA/LARS/DDP ops
    would otherwise silently update a buffer the model no longer reads)."""
    import pytest
    import torch

    from byol_amd.models.byol import BYOL

    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=32, classifier_output_size=4,
             total_training_steps=10, head_latent_size=64)
    m.finalize()
    m.flat_space.verify()  # intact after finalize
    m.base_network.to(memory_format=torch.channels_last)  # the footgun
    with pytest.raises(RuntimeError, match="detached"):
        m.flat_space.verify()
