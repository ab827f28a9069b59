"""Evaluation metrics (reference: src/blades/utils.py:39-56)."""
from __future__ import annotations

import torch


def accuracy(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    """Precision@k for the specified values of k."""
    maxk = max(topk)
    batch_size = target.size(0)

    _, pred = output.topk(maxk, 1, True, True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))

    res = []
    for k in topk:
        correct_k = correct[:k].reshape(-1).float().sum(0)
        res.append(correct_k.mul_(100.0 / batch_size))
    return res


def top1_accuracy(output: torch.Tensor, target: torch.Tensor) -> float:
    return accuracy(output, target, topk=(1,))[0].item()
