"""Tensor utility shims (reference: aggregators/torch_utils.py).

Kept for API parity with code written against the reference (it imported
``torch._six``, which no longer exists — fixed here).  New code should use
blades_amd.ops directly.
"""
from __future__ import annotations

from math import inf
from typing import Iterable, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

_tensor_or_tensors = Union[torch.Tensor, Iterable[torch.Tensor]]


class HLoss(nn.Module):
    """Negative entropy of softmax(x)."""

    def forward(self, x):
        b = F.softmax(x, dim=1) * F.log_softmax(x, dim=1)
        return -1.0 * b.sum()


def _total_norm(parameters, norm_type: float) -> torch.Tensor:
    parameters = [p for p in parameters if p.dtype != torch.int64]
    if not parameters:
        return torch.tensor(0.0)
    device = parameters[0].device
    if norm_type == inf:
        norms = [p.detach().abs().max().to(device) for p in parameters]
        return norms[0] if len(norms) == 1 else torch.max(torch.stack(norms))
    return torch.norm(
        torch.stack([torch.norm(p.detach(), norm_type).to(device)
                     for p in parameters]), norm_type)


def clip_tensor_norm_(parameters: _tensor_or_tensors, max_norm: float,
                      norm_type: float = 2.0) -> torch.Tensor:
    """Scale tensor(s) so the joint norm is <= max_norm (in place); returns
    the (first) clipped tensor, matching the reference's return contract."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    parameters = list(parameters)
    total_norm = _total_norm(parameters, float(norm_type))
    clip_coef = torch.clamp(float(max_norm) / (total_norm + 1e-6), max=1.0)
    out = None
    for p in parameters:
        if p.dtype != torch.int64:
            p.detach().mul_(clip_coef.to(p.device))
            if out is None:
                out = p
    return out


def clip_para_norm_(parameters, max_norm: float,
                    norm_type: float = 2.0) -> torch.Tensor:
    """Clip a dict of parameters jointly (reference signature); returns the
    total norm."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    elif hasattr(parameters, "values"):
        parameters = list(parameters.values())
    else:
        parameters = list(parameters)
    total_norm = _total_norm(parameters, float(norm_type))
    clip_coef = torch.clamp(float(max_norm) / (total_norm + 1e-6), max=1.0)
    for p in parameters:
        if p.dtype != torch.int64:
            p.detach().mul_(clip_coef.to(p.device))
    return total_norm


def l2norm(model: dict) -> torch.Tensor:
    return torch.linalg.norm(torch.stack(
        [torch.linalg.norm(v) for v in model.values()
         if v.dtype != torch.int64]))


def l2dist(model1: dict, model2: dict) -> torch.Tensor:
    return torch.linalg.norm(torch.stack(
        [torch.linalg.norm(model1[k] - model2[k]) for k in model1
         if model1[k].dtype != torch.int64]))


def cos_sim(model1: dict, model2: dict) -> torch.Tensor:
    num = sum((model1[k] * model2[k]).sum() for k in model1)
    return num / torch.clamp(l2norm(model1) * l2norm(model2), min=1e-5)
