"""FusedBatchNorm semantics (composed path here; HIP path in test_ops_gpu).

Oracle: nn.BatchNorm2d/1d + explicit add + relu."""

import torch
import torch.nn as nn

from byol_amd.ops.bn import FusedBatchNorm


def test_bn2d_forward_backward_matches_torch():
    torch.manual_seed(0)
    fused = FusedBatchNorm(6, relu=False)
    ref = nn.BatchNorm2d(6)
    x1 = torch.randn(4, 6, 5, 5, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = fused(x1)
    y2 = ref(x2)
    assert torch.allclose(y1, y2, atol=1e-6)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)
    assert torch.allclose(fused.weight.grad, ref.weight.grad, atol=1e-5)
    assert torch.allclose(fused.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(fused.running_var, ref.running_var, atol=1e-6)


def test_bn_relu_residual_fusion_semantics():
    torch.manual_seed(1)
    fused = FusedBatchNorm(8, relu=True)
    ref = nn.BatchNorm2d(8)
    x1 = torch.randn(3, 8, 4, 4, requires_grad=True)
    r1 = torch.randn(3, 8, 4, 4, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    r2 = r1.detach().clone().requires_grad_(True)
    y1 = fused(x1, residual=r1)
    y2 = torch.relu(ref(x2) + r2)
    assert torch.allclose(y1, y2, atol=1e-6)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)
    assert torch.allclose(r1.grad, r2.grad, atol=1e-6)


def test_bn1d_rows_path():
    torch.manual_seed(2)
    fused = FusedBatchNorm(16, relu=True)
    ref = nn.BatchNorm1d(16)
    x1 = torch.randn(32, 16, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = fused(x1)
    y2 = torch.relu(ref(x2))
    assert torch.allclose(y1, y2, atol=1e-6)
    y1.sum().backward()
    y2.sum().backward()
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)


def test_eval_mode_uses_running_stats():
    torch.manual_seed(3)
    fused = FusedBatchNorm(4)
    ref = nn.BatchNorm2d(4)
    x = torch.randn(8, 4, 3, 3)
    fused(x)
    ref(x)
    fused.eval()
    ref.eval()
    x2 = torch.randn(2, 4, 3, 3)
    assert torch.allclose(fused(x2), ref(x2), atol=1e-6)


def test_state_dict_names_match_torch_bn():
    fused = FusedBatchNorm(4)
    keys = set(fused.state_dict().keys())
    assert keys == {"weight", "bias", "running_mean", "running_var",
                    "num_batches_tracked"}


def test_convert_sync_flips_flag():
    from byol_amd.parallel import convert_sync_batchnorm
    from byol_amd.models.resnet import build_encoder
    enc = build_encoder("resnet18")
    enc = convert_sync_batchnorm(enc)
    assert enc.bn1.sync is True
