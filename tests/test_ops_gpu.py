"""GPU numerics: every HIP kernel vs its plain-PyTorch fp32 oracle."""

import copy

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from byol_amd.ops import require_extension
    return require_extension("gpu tests")


def test_extension_loaded_from_tree(ext):
    import byol_amd._C as C
    assert "byol_amd" in C.__file__, C.__file__


def test_flat_ema_update_matches_oracle(ext):
    from byol_amd.ops.ema import flat_ema_update, flat_ema_update_reference
    torch.manual_seed(0)
    for n in (17, 4096, 1_000_003):
        mean = torch.randn(n, device="cuda")
        x = torch.randn(n, device="cuda")
        ref = mean.clone()
        flat_ema_update(mean, x, 0.996)
        flat_ema_update_reference(ref, x, 0.996)
        assert torch.allclose(mean, ref, atol=1e-6), \
            (mean - ref).abs().max().item()


def test_fused_byol_loss_forward_backward(ext):
    from byol_amd.objective import _FusedBYOLLoss, _loss_reference
    torch.manual_seed(1)
    B, D = 512, 256
    p1 = torch.randn(B, D, device="cuda", requires_grad=True)
    p2 = torch.randn(B, D, device="cuda", requires_grad=True)
    z1 = torch.randn(B, D, device="cuda")
    z2 = torch.randn(B, D, device="cuda")

    loss = _FusedBYOLLoss.apply(p1, p2, z1, z2)
    loss.backward()
    g1, g2 = p1.grad.clone(), p2.grad.clone()

    p1r = p1.detach().clone().requires_grad_(True)
    p2r = p2.detach().clone().requires_grad_(True)
    ref = _loss_reference(p1r, p2r, z1, z2)
    ref.backward()

    assert torch.allclose(loss, ref, rtol=1e-5, atol=1e-6), \
        (float(loss), float(ref))
    assert torch.allclose(g1, p1r.grad, rtol=1e-4, atol=1e-7)
    assert torch.allclose(g2, p2r.grad, rtol=1e-4, atol=1e-7)


def test_fused_lars_single_step_strict(ext):
    """One fused step on controlled synthetic flat tensors vs an fp64
    oracle of the reference LARS+momentum math."""
    from byol_amd.ops import require_extension
    C = require_extension("lars")
    torch.manual_seed(11)
    lens = [1000, 257, 64, 4096, 31]
    offs = [0]
    for n in lens[:-1]:
        offs.append(offs[-1] + n)
    total = sum(lens)
    p = torch.randn(total, device="cuda")
    g = torch.randn(total, device="cuda") * 0.1
    m = torch.zeros_like(p)
    wd = [1e-4, 0.0, 1e-4, 0.0, 1e-4]
    adapt = [1, 0, 1, 0, 1]
    trust, eps, lr, mu = 0.001, 0.0, 0.05, 0.9

    # fp64 oracle
    p64, g64 = p.double(), g.double()
    want = p64.clone()
    m64 = torch.zeros_like(p64)
    for s, (o, n) in enumerate(zip(offs, lens)):
        seg_p = want[o:o + n]
        geff = g64[o:o + n] + wd[s] * seg_p
        if adapt[s]:
            pn, gn = seg_p.norm(), geff.norm()
            if pn > 0 and gn > 0:
                geff = geff * (trust * pn / (gn + eps))
        m64[o:o + n] = geff  # first step
        seg_p -= lr * geff

    # fused
    dev = p.device
    seg_off = torch.tensor(offs, dtype=torch.int64, device=dev)
    seg_len = torch.tensor(lens, dtype=torch.int64, device=dev)
    seg_wd = torch.tensor(wd, dtype=torch.float32, device=dev)
    seg_adapt = torch.tensor(adapt, dtype=torch.int32, device=dev)
    chunk_seg, chunk_base = [], []
    for i, (o, n) in enumerate(zip(offs, lens)):
        for c in range(0, n, 65536):
            chunk_seg.append(i)
            chunk_base.append(c)
    chunk_seg = torch.tensor(chunk_seg, dtype=torch.int32, device=dev)
    chunk_base = torch.tensor(chunk_base, dtype=torch.int64, device=dev)
    norm_acc = torch.zeros(2 * len(lens), device=dev)
    alr = torch.ones(len(lens), device=dev)
    C.lars_momentum_step(p, g, m, norm_acc, alr, seg_off, seg_len, seg_wd,
                         seg_adapt, chunk_seg, chunk_base, trust, eps, lr,
                         mu, 0)
    assert torch.allclose(p.double(), want, rtol=1e-6, atol=1e-7), \
        (p.double() - want).abs().max().item()
    assert torch.allclose(m.double(), m64, rtol=1e-6, atol=1e-7)


def test_fused_lars_step_matches_eager(ext):
    from byol_amd import layers
    from byol_amd.models.byol import BYOL
    from byol_amd.optim.lars import LARS

    def build(seed):
        torch.manual_seed(seed)
        m = BYOL(arch="resnet18", base_network_output_size=512,
                 projection_output_size=8, classifier_output_size=3,
                 total_training_steps=10, head_latent_size=16).cuda()
        return m.finalize()

    m_fused = build(7)
    m_eager = build(7)
    assert torch.equal(m_fused.flat_space.flat_params,
                       m_eager.flat_space.flat_params)

    opt_f = LARS(torch.optim.SGD(layers.add_weight_decay(m_fused, 1e-4),
                                 lr=0.05, momentum=0.9), eps=0.0)
    opt_f.attach_flat_space(m_fused.flat_space)
    opt_e = LARS(torch.optim.SGD(layers.add_weight_decay(m_eager, 1e-4),
                                 lr=0.05, momentum=0.9), eps=0.0)
    # eager path: flat zero_grad but NO fused step (no attach)

    for step in range(3):
        torch.manual_seed(50 + step)
        x1 = torch.rand(4, 3, 32, 32, device="cuda")
        x2 = torch.rand(4, 3, 32, 32, device="cuda")
        for m, opt in ((m_fused, opt_f), (m_eager, opt_e)):
            m.train()
            out = m(x1, x2)
            loss = out["online_prediction1"].square().mean() + \
                out["linear_preds"].square().mean()
            m.flat_space.zero_grads()
            loss.backward()
            opt.step()
        # fp32 reduction-order noise compounds through BN batch stats over
        # steps; the strict check is the single-step fp64 test above
        diff = (m_fused.flat_space.flat_params
                - m_eager.flat_space.flat_params).abs().max()
        assert diff < 1e-4, f"step {step}: {diff}"


def test_byol_fused_loss_used_in_model_path(ext):
    """loss_function routes through the fused kernel on GPU fp32."""
    from byol_amd import objective
    torch.manual_seed(2)
    p = [torch.randn(16, 8, device="cuda", requires_grad=(i < 2))
         for i in range(4)]
    loss = objective.loss_function(*p)
    assert loss.requires_grad
    assert loss.grad_fn.__class__.__name__.startswith("_FusedBYOLLoss")


@pytest.mark.parametrize("shape,relu,residual", [
    ((16, 64, 14, 14), False, False),
    ((16, 64, 14, 14), True, False),
    ((8, 256, 7, 7), True, True),
    ((32, 2048, 7, 7), True, True),
])
def test_fused_bn2d_hip_vs_oracle(ext, shape, relu, residual):
    """HIP NHWC fused BN vs torch BatchNorm2d oracle (fwd+bwd+running)."""
    import torch.nn as nn
    from byol_amd.ops.bn import FusedBatchNorm
    torch.manual_seed(5)
    c = shape[1]
    fused = FusedBatchNorm(c, relu=relu).cuda()
    ref = nn.BatchNorm2d(c).cuda()
    x1 = torch.randn(shape, device="cuda").to(
        memory_format=torch.channels_last).requires_grad_(True)
    x2 = x1.detach().clone().requires_grad_(True)
    r1 = r2 = None
    if residual:
        r1 = torch.randn(shape, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        r2 = r1.detach().clone().requires_grad_(True)
    y1 = fused(x1, residual=r1)
    y2 = ref(x2)
    if residual:
        y2 = y2 + r2
    if relu:
        y2 = torch.relu(y2)
    assert torch.allclose(y1, y2, rtol=1e-4, atol=1e-5), \
        (y1 - y2).abs().max().item()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-4, atol=1e-4), \
        (x1.grad - x2.grad).abs().max().item()
    if residual:
        assert torch.allclose(r1.grad, r2.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(fused.weight.grad, ref.weight.grad, rtol=1e-3,
                          atol=1e-3)
    assert torch.allclose(fused.bias.grad, ref.bias.grad, rtol=1e-3,
                          atol=1e-3)
    assert torch.allclose(fused.running_mean, ref.running_mean, atol=1e-4)
    assert torch.allclose(fused.running_var, ref.running_var, rtol=1e-3,
                          atol=1e-4)


def test_fused_bn1d_hip_vs_oracle(ext):
    import torch.nn as nn
    from byol_amd.ops.bn import FusedBatchNorm
    torch.manual_seed(6)
    fused = FusedBatchNorm(4096, relu=True).cuda()
    ref = nn.BatchNorm1d(4096).cuda()
    x1 = torch.randn(512, 4096, device="cuda", requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = fused(x1)
    y2 = torch.relu(ref(x2))
    assert torch.allclose(y1, y2, rtol=1e-4, atol=1e-5)
    y1.sum().backward()
    y2.sum().backward()
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-4, atol=1e-4)


def test_fused_bn_eval_mode_hip(ext):
    import torch.nn as nn
    from byol_amd.ops.bn import FusedBatchNorm
    torch.manual_seed(7)
    fused = FusedBatchNorm(64).cuda()
    ref = nn.BatchNorm2d(64).cuda()
    x = torch.randn(8, 64, 7, 7, device="cuda").to(
        memory_format=torch.channels_last)
    fused(x)
    ref(x)
    fused.eval()
    ref.eval()
    with torch.no_grad():
        x2 = torch.randn(4, 64, 7, 7, device="cuda").to(
            memory_format=torch.channels_last)
        assert torch.allclose(fused(x2), ref(x2), rtol=1e-4, atol=1e-5)


def test_fused_ce_topk_vs_oracle(ext):
    import torch.nn.functional as F
    from byol_amd.engine import metrics
    from byol_amd.ops.classifier import _FusedCE
    torch.manual_seed(8)
    m, n = 1024, 1000
    logits = torch.randn(m, n, device="cuda", requires_grad=True)
    labels = torch.randint(n, (m,), device="cuda")
    loss, acc1, acc5 = _FusedCE.apply(logits, labels)
    ref_loss = F.cross_entropy(logits.detach(), labels)
    r1, r5 = metrics.topk(logits.detach(), labels, topk=(1, 5))
    assert torch.allclose(loss, ref_loss, rtol=1e-5, atol=1e-6)
    assert torch.allclose(acc1, r1, atol=1e-3), (float(acc1), float(r1))
    assert torch.allclose(acc5, r5, atol=1e-3), (float(acc5), float(r5))
    loss.backward()
    g_fused = logits.grad.clone()
    logits2 = logits.detach().clone().requires_grad_(True)
    F.cross_entropy(logits2, labels).backward()
    assert torch.allclose(g_fused, logits2.grad, rtol=1e-4, atol=1e-7), \
        (g_fused - logits2.grad).abs().max().item()


def test_aug_sample_matches_interpolate(ext):
    """Identity crop + no flip must equal F.interpolate bilinear resize."""
    import torch.nn.functional as F
    torch.manual_seed(9)
    b, hs, ws, s = 4, 64, 48, 32
    src = torch.rand(b, 3, hs, ws, device="cuda")
    src_nhwc = src.permute(0, 2, 3, 1).contiguous()
    crop = torch.tensor([[0.0, 0.0, hs, ws, 0.0]] * b,
                        device="cuda").reshape(-1)
    dst = torch.empty(b, s, s, 3, device="cuda")
    gray = torch.zeros(b, device="cuda")
    ext.aug_sample(src_nhwc.reshape(-1), dst.reshape(-1), gray, crop,
                   hs, ws, s, 0)
    want = F.interpolate(src, size=(s, s), mode="bilinear",
                         align_corners=False)
    got = dst.permute(0, 3, 1, 2)
    assert torch.allclose(got, want, atol=1e-5), \
        (got - want).abs().max().item()
    # gray accumulator = sum of gray over output pixels
    gwant = (0.299 * want[:, 0] + 0.587 * want[:, 1]
             + 0.114 * want[:, 2]).sum(dim=(1, 2))
    assert torch.allclose(gray, gwant, rtol=1e-3, atol=1e-2)


def test_aug_color_matches_reference(ext):
    from byol_amd.data.gpu_augment import apply_color_reference
    torch.manual_seed(10)
    b, s = 6, 16
    img = torch.rand(b, s, s, 3, device="cuda")
    cparam = torch.zeros(b, 10, device="cuda")
    for i in range(b):
        cparam[i] = torch.tensor(
            [1.0, 0.7 + 0.1 * i, 1.3 - 0.05 * i, 0.9, 0.05 * (i - 3),
             1.0 if i == 5 else 0.0,
             *torch.randperm(4, generator=torch.Generator().manual_seed(i)
                             ).tolist()])
    gray = (0.299 * img[..., 0] + 0.587 * img[..., 1]
            + 0.114 * img[..., 2]).sum(dim=(1, 2))
    gray_mean = gray / (s * s)
    work = img.clone()
    ext.aug_color(work.reshape(-1), gray, cparam.reshape(-1).contiguous(), s)
    want = apply_color_reference(img.permute(0, 3, 1, 2), cparam.cpu(),
                                 gray_mean)
    got = work.permute(0, 3, 1, 2)
    assert torch.allclose(got, want, atol=1e-4), \
        (got - want).abs().max().item()


def test_gpu_two_view_pipeline_end_to_end(ext):
    from byol_amd.data.gpu_augment import GPUTwoViewAugment
    torch.manual_seed(11)
    pipe = GPUTwoViewAugment(out_size=32, jitter_strength=1.0,
                             dali_mode=False, seed=3)
    batch = torch.rand(8, 3, 40, 40, device="cuda")
    a1, a2 = pipe(batch)
    for a in (a1, a2):
        assert a.shape == (8, 3, 32, 32)
        assert a.is_contiguous(memory_format=torch.channels_last)
        assert float(a.min()) >= 0.0 and float(a.max()) <= 1.0
    assert not torch.equal(a1, a2)


def test_mfma_conv1x1_vs_miopen(ext, monkeypatch):
    """Forced MFMA mode: fwd/dgrad/wgrad kernels vs F.conv2d oracle."""
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv1x1Fn
    monkeypatch.setenv("BYOL_MFMA_CONV1X1", "1")
    torch.manual_seed(12)
    for bsz, hgt, cin, cout in [(8, 56, 64, 256), (8, 28, 512, 128),
                                (4, 7, 2048, 512), (4, 14, 96, 224)]:
        x = torch.randn(bsz, cin, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(cout, cin, 1, 1, device="cuda") * 0.05
             ).requires_grad_(True)
        y = _Conv1x1Fn.apply(x, w)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        y2 = F.conv2d(x2, w2)
        y2.backward(g)
        assert torch.allclose(y, y2, rtol=1e-4, atol=1e-4), \
            (hgt, cin, cout, (y - y2).abs().max().item())
        assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-4)
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2), \
            (w.grad - w2.grad).abs().max().item()


def test_mfma_conv3x3_vs_miopen(ext, monkeypatch):
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv3x3Fn
    torch.manual_seed(13)
    for bsz, hgt, cin, cout, stride in [(4, 14, 64, 64, 1),
                                        (4, 15, 128, 128, 1),
                                        (4, 28, 128, 128, 2),
                                        (2, 9, 256, 256, 2)]:
        x = torch.randn(bsz, cin, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(cout, cin, 3, 3, device="cuda") * 0.05
             ).requires_grad_(True)
        y = _Conv3x3Fn.apply(x, w, stride)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        y2 = F.conv2d(x2, w2, stride=stride, padding=1)
        assert y.shape == y2.shape, (y.shape, y2.shape)
        assert torch.allclose(y, y2, rtol=1e-4, atol=1e-4), \
            (hgt, cin, cout, stride, (y - y2).abs().max().item())
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        y2.backward(g)
        assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-4)
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2)


def test_smoke_entrypoint():
    import __graft_entry__
    __graft_entry__.smoke()


def test_wgrad_v2_matches_oracle(ext, monkeypatch):
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv1x1Fn
    monkeypatch.setenv("BYOL_MFMA_CONV1X1", "1")
    monkeypatch.setenv("BYOL_WGRAD", "v2")
    torch.manual_seed(21)
    for bsz, hgt, cin, cout in [(8, 56, 64, 256), (4, 7, 2048, 512),
                                (4, 14, 96, 224)]:
        x = torch.randn(bsz, cin, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(cout, cin, 1, 1, device="cuda") * 0.05
             ).requires_grad_(True)
        y = _Conv1x1Fn.apply(x, w)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        F.conv2d(x2, w2).backward(g)
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2), \
            (hgt, cin, cout, (w.grad - w2.grad).abs().max().item())


def test_conv3x3_fast_path_matches_oracle(ext, monkeypatch):
    """glds fast path (padded input) — fast-eligible shapes (M%128==0)."""
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv3x3Fn
    monkeypatch.setenv("BYOL_CONV3X3_FAST", "1")
    torch.manual_seed(22)
    for bsz, hgt, c, stride in [(8, 16, 128, 1), (8, 16, 256, 2),
                                (2, 32, 512, 1), (8, 16, 64, 1)]:
        x = torch.randn(bsz, c, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last)
        w = torch.randn(c, c, 3, 3, device="cuda") * 0.05
        ho = (hgt + 2 - 3) // stride + 1
        m = bsz * ho * ho
        if m % 128 != 0:
            continue
        y = _Conv3x3Fn.apply(x, w, stride)
        y2 = F.conv2d(x, w, stride=stride, padding=1)
        assert torch.allclose(y, y2, rtol=1e-4, atol=1e-4), \
            (hgt, c, stride, (y - y2).abs().max().item())


def test_aug_sample_v2_matches_v1(ext):
    import torch.nn.functional as F
    torch.manual_seed(14)
    b, hs, ws, s = 6, 70, 50, 32
    src = torch.rand(b, hs, ws, 3, device="cuda").contiguous()
    crop = torch.tensor(
        [[3.0, 2.0, 40, 30, float(i % 2)] for i in range(b)],
        device="cuda").reshape(-1)
    out = []
    for v2 in (0, 1):
        dst = torch.empty(b, s, s, 3, device="cuda")
        gray = torch.zeros(b, device="cuda")
        ext.aug_sample(src.reshape(-1), dst.reshape(-1), gray, crop,
                       hs, ws, s, v2)
        out.append((dst, gray))
    assert torch.equal(out[0][0], out[1][0])
    assert torch.allclose(out[0][1], out[1][1], rtol=1e-4, atol=1e-3)


def test_bf16_fused_bn_vs_fp32_oracle(ext, monkeypatch):
    """bf16-I/O BN kernels vs an fp32 reference on the same (bf16-rounded)
    inputs; stats/params stay fp32 so tolerances are bf16-IO-class."""
    import torch.nn as nn
    from byol_amd.ops.bn import FusedBatchNorm
    monkeypatch.setenv("BYOL_BF16_BN", "1")
    torch.manual_seed(15)
    for shape, relu, residual in [((16, 64, 14, 14), True, False),
                                  ((8, 256, 7, 7), True, True)]:
        c = shape[1]
        fused = FusedBatchNorm(c, relu=relu).cuda()
        ref = nn.BatchNorm2d(c).cuda()
        xb = torch.randn(shape, device="cuda").to(torch.bfloat16)
        x1 = xb.to(memory_format=torch.channels_last).requires_grad_(True)
        x2 = xb.float().clone().requires_grad_(True)
        r1 = r2 = None
        if residual:
            rb = torch.randn(shape, device="cuda").to(torch.bfloat16)
            r1 = rb.to(memory_format=torch.channels_last).requires_grad_(True)
            r2 = rb.float().clone().requires_grad_(True)
        y1 = fused(x1, residual=r1)
        assert y1.dtype == torch.bfloat16
        y2 = ref(x2)
        if residual:
            y2 = y2 + r2
        if relu:
            y2 = torch.relu(y2)
        assert torch.allclose(y1.float(), y2, rtol=2e-2, atol=2e-2), \
            (y1.float() - y2).abs().max().item()
        g = torch.randn_like(y2).to(torch.bfloat16)
        y1.backward(g)
        y2.backward(g.float())
        assert torch.allclose(x1.grad.float(), x2.grad, rtol=5e-2,
                              atol=5e-2), \
            (x1.grad.float() - x2.grad).abs().max().item()
        assert torch.allclose(fused.running_mean, ref.running_mean,
                              atol=5e-3)


def _tiny_byol(seed=33):
    from byol_amd.models.byol import BYOL
    torch.manual_seed(seed)
    m = BYOL(arch="resnet18", base_network_output_size=512,
             projection_output_size=64, classifier_output_size=10,
             total_training_steps=100, head_latent_size=128).cuda()
    m.base_network.to(memory_format=torch.channels_last)
    m.finalize()
    m.train()
    return m


def _lars_for(model):
    from byol_amd import layers
    from byol_amd.optim.lars import LARS
    inner = torch.optim.SGD(layers.add_weight_decay(model, 1e-6), lr=0.05,
                            momentum=0.9)
    opt = LARS(inner, eps=0.0)
    opt.attach_flat_space(model.flat_space)
    return opt


def test_hipgraph_step_parity(ext):
    """Captured-graph replays produce the same parameters as eager steps:
    same tiny BYOL, same data sequence, 3 steps each way."""
    from byol_amd.engine.graph_step import GraphedTrainStep
    from byol_amd.objective import loss_function
    from byol_amd.ops.classifier import cross_entropy_topk

    torch.manual_seed(44)
    B, S = 16, 32
    batches = [(torch.rand(B, 3, S, S, device="cuda")
                .to(memory_format=torch.channels_last),
                torch.rand(B, 3, S, S, device="cuda")
                .to(memory_format=torch.channels_last),
                torch.randint(10, (B,), device="cuda"))
               for _ in range(5)]

    def make_step(model, opt):
        def step_body(a1, a2, lab):
            out = model(a1, a2)
            loss = loss_function(
                out["online_prediction1"].float(),
                out["online_prediction2"].float(),
                out["target_projection1"].float(),
                out["target_projection2"].float())
            ce, _x, _y = cross_entropy_topk(
                out["linear_preds"].float().contiguous(),
                torch.cat([lab, lab]))
            loss = loss + ce
            opt.zero_grad()
            loss.backward()
            opt.step()
            return loss
        return step_body

    # eager: warmup(2) on batches[0] then 3 steps on batches[1..3]
    m1 = _tiny_byol()
    o1 = _lars_for(m1)
    s1 = make_step(m1, o1)
    for _ in range(2):
        s1(*batches[0])
    for i in (1, 2, 3):
        s1(*batches[i])

    # graphed: identical sequence, steps via replay
    m2 = _tiny_byol()
    o2 = _lars_for(m2)
    s2 = make_step(m2, o2)
    static = tuple(t.clone() for t in batches[0])
    g = GraphedTrainStep(m2, o2, s2, static, warmup_steps=2)
    g.capture()
    for i in (1, 2, 3):
        g.replay(*batches[i])
    torch.cuda.synchronize()

    pa = m1.flat_space.flat_params
    pb = m2.flat_space.flat_params
    assert torch.allclose(pa, pb, rtol=1e-4, atol=1e-5), \
        (pa - pb).abs().max().item()
    ea = m1.target_network.mean
    eb = m2.target_network.mean
    assert m1.target_network.step == m2.target_network.step
    assert torch.allclose(ea, eb, rtol=1e-4, atol=1e-5), \
        (ea - eb).abs().max().item()


def test_rccl_single_rank_group(ext):
    """Exercise the real RCCL code path (init + all_reduce + broadcast)
    with a world-1 process group on the GPU — the multi-rank semantics are
    gloo-covered (tests/test_ddp_gloo.py); this pins the nccl backend
    itself working on the box."""
    import torch.distributed as dist
    if dist.is_initialized():
        pytest.skip("process group already initialized")
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        t = torch.arange(1024, device="cuda", dtype=torch.float32)
        dist.all_reduce(t)
        dist.broadcast(t, src=0)
        ref = torch.arange(1024, device="cuda", dtype=torch.float32)
        assert torch.equal(t, ref)
        # the packed SyncBN exchange path used by ops/bn.py
        acc = torch.randn(2 * 64, device="cuda")
        want = acc.clone()
        dist.all_reduce(acc)
        assert torch.allclose(acc, want)
    finally:
        dist.destroy_process_group()


def test_aug_blur_matches_composed(ext):
    """HIP separable blur vs the composed grouped-conv oracle."""
    from byol_amd.data.gpu_augment import _gaussian_blur_batched
    torch.manual_seed(16)
    b, s, k = 12, 64, 7
    img_nhwc = torch.rand(b, s, s, 3, device="cuda").contiguous()
    sigma = torch.tensor(
        [0.0, 0.5, 1.0, 1.7, 2.0, 0.0, 0.3, 1.2, 0.8, 1.5, 0.0, 2.0],
        device="cuda")
    tmp = torch.empty_like(img_nhwc)
    out = torch.empty_like(img_nhwc)
    wts = torch.empty(b, k, device="cuda")
    ext.aug_blur(img_nhwc.reshape(-1), tmp.reshape(-1), out.reshape(-1),
                 sigma, wts.reshape(-1), s, k)
    ref = _gaussian_blur_batched(
        img_nhwc.permute(0, 3, 1, 2).contiguous(), sigma, k)
    got = out.permute(0, 3, 1, 2)
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-5), \
        (got - ref).abs().max().item()


def test_auto_dispatch_backward_parity(ext, monkeypatch):
    """Auto mode (per-shape tables): fwd+bwd parity vs F.conv2d on a
    fwd-table shape, a dgrad-only shape, and an off-table shape."""
    import torch.nn.functional as F
    from byol_amd.ops.conv import _AUTO_DGRAD, _AUTO_SHAPES, _Conv1x1Fn
    monkeypatch.setenv("BYOL_MFMA_CONV1X1", "auto")
    torch.manual_seed(27)
    cases = [(8, 56, 64, 256),     # fwd + dgrad tables
             (32, 14, 1024, 256),  # dgrad-only table (MIOpen fwd)
             (32, 7, 512, 2048)]   # off both tables
    for bsz, hgt, cin, cout in cases:
        x = torch.randn(bsz, cin, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(cout, cin, 1, 1, device="cuda") * 0.05
             ).requires_grad_(True)
        y = _Conv1x1Fn.apply(x, w)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        F.conv2d(x2, w2).backward(g)
        assert torch.allclose(y, F.conv2d(x.detach(), w.detach()),
                              rtol=1e-4, atol=1e-4)
        assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-4), \
            (cin, cout, (x.grad - x2.grad).abs().max().item())
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2)
    assert (64, 64) in _AUTO_SHAPES and (1024, 256) in _AUTO_DGRAD


def test_fp16_scaler_training_step(ext):
    """--half-dtype fp16 parity mode: autocast fp16 + GradScaler over the
    fused-LARS step trains without inf/nan for a few steps."""
    from byol_amd.objective import loss_function
    from byol_amd.ops.classifier import cross_entropy_topk

    m = _tiny_byol(seed=55)
    opt = _lars_for(m)
    scaler = torch.amp.GradScaler("cuda")
    B, S = 16, 32
    for i in range(3):
        a1 = torch.rand(B, 3, S, S, device="cuda").to(
            memory_format=torch.channels_last)
        a2 = torch.rand(B, 3, S, S, device="cuda").to(
            memory_format=torch.channels_last)
        lab = torch.randint(10, (B,), device="cuda")
        with torch.autocast("cuda", dtype=torch.float16):
            out = m(a1, a2)
            loss = loss_function(
                out["online_prediction1"].float(),
                out["online_prediction2"].float(),
                out["target_projection1"].float(),
                out["target_projection2"].float())
            ce, _1, _5 = cross_entropy_topk(
                out["linear_preds"].float().contiguous(),
                torch.cat([lab, lab]))
            loss = loss + ce
        opt.zero_grad()
        scaler.scale(loss).backward()
        scaler.unscale_(opt)
        scaler.step(opt)
        scaler.update()
    assert torch.isfinite(m.flat_space.flat_params).all()
    assert torch.isfinite(loss.float())


def test_conv3x3_wgrad_matches_oracle(ext, monkeypatch):
    """9-tap wgrad kernel (BYOL_C3WGRAD) vs MIOpen weight grads, stride
    1 and 2."""
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv3x3Fn
    monkeypatch.setenv("BYOL_MFMA_CONV3X3", "1")
    monkeypatch.setenv("BYOL_C3WGRAD", "1")
    torch.manual_seed(31)
    for bsz, hgt, c, stride in [(8, 16, 128, 1), (8, 16, 256, 2),
                                (2, 32, 64, 1)]:
        x = torch.randn(bsz, c, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(c, c, 3, 3, device="cuda") * 0.05
             ).requires_grad_(True)
        ho = (hgt + 2 - 3) // stride + 1
        if (bsz * ho * ho) % 32 != 0:
            continue
        y = _Conv3x3Fn.apply(x, w, stride)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        F.conv2d(x2, w2, stride=stride, padding=1).backward(g)
        assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-4)
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2), \
            (hgt, c, stride, (w.grad - w2.grad).abs().max().item())


def test_conv3x3_dgrad_via_fwd_matches_oracle(ext, monkeypatch):
    """Stride-1 dgrad through the fwd fast kernel (rot180(W)^T on padded
    dy) vs MIOpen."""
    import torch.nn.functional as F
    from byol_amd.ops.conv import _Conv3x3Fn
    monkeypatch.setenv("BYOL_MFMA_CONV3X3", "1")
    monkeypatch.setenv("BYOL_C3DGRAD", "1")
    torch.manual_seed(35)
    for bsz, hgt, c in [(8, 16, 128), (2, 32, 64)]:
        if (bsz * hgt * hgt) % 128 != 0:
            continue
        x = torch.randn(bsz, c, hgt, hgt, device="cuda").to(
            memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(c, c, 3, 3, device="cuda") * 0.05
             ).requires_grad_(True)
        y = _Conv3x3Fn.apply(x, w, 1)
        g = torch.randn_like(y).to(memory_format=torch.channels_last)
        y.backward(g)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        F.conv2d(x2, w2, stride=1, padding=1).backward(g)
        assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-4), \
            (hgt, c, (x.grad - x2.grad).abs().max().item())
        assert torch.allclose(w.grad, w2.grad, rtol=1e-3, atol=1e-2)


def test_engine_train_epoch_on_gpu(ext, tmp_path):
    """The real trainer loop (setup -> lazy init -> train epoch -> test
    epoch) on GPU with the synthetic task: exercises the device
    prefetcher, fused BN/loss/CE/LARS through the engine, and the epoch
    bookkeeping."""
    import byol_amd.config as config
    from byol_amd.engine import trainer
    from byol_amd.optim import build_optimizer

    args = config.parse_args([
        "--task", "synthetic_multi_augment_image_folder",
        "--arch", "resnet18", "--representation-size", "512",
        "--batch-size", "16", "--epochs", "1", "--num-replicas", "1",
        "--image-size-override", "32", "--synthetic-classes", "8",
        "--synthetic-train-samples", "64", "--synthetic-test-samples", "32",
        "--workers-per-replica", "0", "--channels-last",
        "--model-dir", str(tmp_path / "m"),
        "--log-dir", str(tmp_path / "l"),
    ])
    args.cuda = True
    args.distributed_rank = 0
    loader, model, grapher = trainer.build_loader_model_grapher(args)
    opt, sched = build_optimizer(model, args)
    trainer.train(1, model, opt, loader.train_loader, grapher, args)
    loss = trainer.test(1, model, loader.test_loader, grapher, args)
    assert loss == loss  # finite, not NaN
    flat = (model.module if hasattr(model, "module") else model) \
        .flat_space.flat_params
    assert torch.isfinite(flat).all()


def test_engine_half_bf16_epoch_on_gpu(ext, tmp_path):
    """--half (bf16 autocast + bf16-I/O fused BN) through the real engine."""
    import byol_amd.config as config
    from byol_amd.engine import trainer
    from byol_amd.optim import build_optimizer

    args = config.parse_args([
        "--task", "synthetic_multi_augment_image_folder",
        "--arch", "resnet18", "--representation-size", "512",
        "--batch-size", "16", "--epochs", "1", "--num-replicas", "1",
        "--image-size-override", "32", "--synthetic-classes", "8",
        "--synthetic-train-samples", "48", "--synthetic-test-samples", "16",
        "--workers-per-replica", "0", "--channels-last", "--half",
        "--model-dir", str(tmp_path / "m"),
        "--log-dir", str(tmp_path / "l"),
    ])
    args.cuda = True
    args.distributed_rank = 0
    loader, model, grapher = trainer.build_loader_model_grapher(args)
    opt, sched = build_optimizer(model, args)
    trainer.train(1, model, opt, loader.train_loader, grapher, args)
    flat = (model.module if hasattr(model, "module") else model) \
        .flat_space.flat_params
    assert torch.isfinite(flat).all()
