"""BYOL loss semantics: whole-tensor Frobenius norms + row dots (the
reference's exact math, /root/reference/objective.py:6-25)."""

import torch

from byol_amd.objective import loss_function, regression_loss


def test_regression_loss_uses_frobenius_norms():
    torch.manual_seed(0)
    x = torch.randn(5, 8)
    y = torch.randn(5, 8)
    got = regression_loss(x, y)
    want = -2 * (x * y).sum(-1) / (x.norm() * y.norm())
    assert torch.allclose(got, want)


def test_loss_function_symmetry_and_detach():
    torch.manual_seed(1)
    p1 = torch.randn(4, 16, requires_grad=True)
    p2 = torch.randn(4, 16, requires_grad=True)
    z1 = torch.randn(4, 16, requires_grad=True)
    z2 = torch.randn(4, 16, requires_grad=True)
    loss = loss_function(p1, p2, z1, z2)
    want = torch.mean(-2 * (p1 * z2).sum(-1) / (p1.norm() * z2.norm())
                      - 2 * (p2 * z1).sum(-1) / (p2.norm() * z1.norm()))
    assert torch.allclose(loss, want)
    loss.backward()
    assert p1.grad is not None and p2.grad is not None
    # targets are detached: no grads flow
    assert z1.grad is None and z2.grad is None


def test_loss_gradients_match_autograd_of_manual_formula():
    torch.manual_seed(2)
    p1 = torch.randn(6, 12, requires_grad=True)
    p2 = torch.randn(6, 12, requires_grad=True)
    z1 = torch.randn(6, 12)
    z2 = torch.randn(6, 12)
    loss_function(p1, p2, z1, z2).backward()

    p1b = p1.detach().clone().requires_grad_(True)
    p2b = p2.detach().clone().requires_grad_(True)
    manual = torch.mean(
        -2 * (p1b * z2).sum(-1) / (p1b.norm() * z2.norm())
        - 2 * (p2b * z1).sum(-1) / (p2b.norm() * z1.norm()))
    manual.backward()
    assert torch.allclose(p1.grad, p1b.grad, atol=1e-6)
    assert torch.allclose(p2.grad, p2b.grad, atol=1e-6)


def test_fused_backward_formula_matches_autograd_cpu():
    """The closed-form gradient the HIP kernel implements, checked on CPU:
    d/dp1 = (-2*go/B) * [ z2/(N1*N2) - S * p1 / (N1^3 * N2) ]."""
    torch.manual_seed(3)
    B, D = 5, 9
    p1 = torch.randn(B, D, requires_grad=True)
    z2 = torch.randn(B, D)
    loss = torch.mean(-2 * (p1 * z2).sum(-1) / (p1.norm() * z2.norm()))
    loss.backward()
    n1, n2 = p1.detach().norm(), z2.norm()
    S = (p1.detach() * z2).sum()
    closed = (-2.0 / B) * (z2 / (n1 * n2) - S * p1.detach() / (n1 ** 3 * n2))
    assert torch.allclose(p1.grad, closed, atol=1e-6)
