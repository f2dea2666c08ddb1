import pytest
import torch

from byol_amd.models.resnet import build_encoder


def test_resnet18_shapes():
    enc = build_encoder("resnet18")
    x = torch.randn(2, 3, 32, 32)
    out = enc(x)
    assert out.shape == (2, 512, 1, 1)


def test_resnet50_shapes_and_param_count():
    enc = build_encoder("resnet50")
    x = torch.randn(2, 3, 64, 64)
    out = enc(x)
    assert out.shape == (2, 2048, 1, 1)
    # torchvision resnet50 minus fc = 23,508,032 params (SURVEY.md sizing)
    n = sum(p.numel() for p in enc.parameters())
    assert n == 23_508_032, n


def test_resnet50_backward():
    enc = build_encoder("resnet50")
    x = torch.randn(2, 3, 32, 32, requires_grad=True)
    enc(x).sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


@pytest.mark.parametrize("arch,expansion_out", [
    ("resnet34", 512), ("resnet101", 2048), ("resnet200", 2048),
    ("wide_resnet50_2", 2048), ("resnext50_32x4d", 2048)])
def test_other_archs_build(arch, expansion_out):
    enc = build_encoder(arch)
    assert enc.out_channels == expansion_out
    out = enc(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, expansion_out, 1, 1)


def test_variant_param_counts_match_torchvision():
    # torchvision reference counts minus the fc layer
    want = {"wide_resnet50_2": 68_883_240 - 2_049_000,
            "resnext50_32x4d": 25_028_904 - 2_049_000}
    for arch, n_want in want.items():
        enc = build_encoder(arch)
        n = sum(p.numel() for p in enc.parameters())
        assert n == n_want, (arch, n, n_want)
