"""Metrics (the reference's ``helpers.metrics`` contract)."""

import torch

__all__ = ["topk"]


@torch.no_grad()
def topk(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    """Top-k accuracy in percent, one scalar per k (standard recipe;
    reference call site ``/root/reference/main.py:598``)."""
    num_classes = output.size(1)
    maxk = min(max(topk), num_classes)  # k clamped for tiny class counts
    batch_size = target.size(0)
    _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    for k in topk:
        k = min(k, num_classes)
        correct_k = correct[:k].reshape(-1).float().sum(0, keepdim=True)
        res.append(correct_k.mul_(100.0 / batch_size).squeeze(0))
    return res
