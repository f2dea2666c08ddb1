"""Integration: the full BYOL objective + LARS pipeline actually optimizes
(deterministic overfit on one fixed batch, CPU)."""

import torch
import torch.nn.functional as F

from byol_amd import layers
from byol_amd.models.byol import BYOL
from byol_amd.objective import loss_function
from byol_amd.optim.lars import LARS


def test_loss_decreases_over_steps():
    torch.manual_seed(0)
    model = BYOL(arch="resnet18", base_network_output_size=512,
                 projection_output_size=16, classifier_output_size=4,
                 total_training_steps=40, head_latent_size=32).finalize()
    inner = torch.optim.SGD(layers.add_weight_decay(model, 1e-6), lr=0.5,
                            momentum=0.9)
    opt = LARS(inner, eps=0.0)
    opt.attach_flat_space(model.flat_space)
    model.train()

    a1 = torch.rand(8, 3, 32, 32)
    a2 = torch.rand(8, 3, 32, 32)
    labels = torch.randint(4, (8,))

    losses = []
    for _ in range(15):
        out = model(a1, a2)
        byol = loss_function(out["online_prediction1"],
                             out["online_prediction2"],
                             out["target_projection1"],
                             out["target_projection2"])
        ce = F.cross_entropy(out["linear_preds"],
                             torch.cat([labels, labels]))
        loss = byol + ce
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    # overfitting one batch must reduce the combined loss clearly
    assert min(losses[-3:]) < losses[0] - 0.3, losses
    assert all(torch.isfinite(torch.tensor(losses))), losses
