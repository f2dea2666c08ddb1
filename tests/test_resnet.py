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
    ("resnet34", 512), ("resnet101", 2048), ("resnet200", 2048)])
def test_other_archs_build(arch, expansion_out):
    enc = build_encoder(arch)
    assert enc.out_channels == expansion_out
