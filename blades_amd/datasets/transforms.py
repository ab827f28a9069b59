"""Tensor-native image transforms.

torchvision is not a dependency of this framework; the reference's CIFAR
train pipeline (RandomResizedCrop(0.75-1.0), HFlip(0.5), Normalize,
RandomErasing(0.25) — reference: datasets/cifar10.py:29-39) is reproduced
here as pure tensor ops so it runs on CPU or GPU and can be applied to
whole stacked batches.
"""
from __future__ import annotations

import random
from typing import Sequence

import torch
import torch.nn.functional as F


class Compose:
    def __init__(self, transforms: Sequence):
        self.transforms = list(transforms)

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, x):
        mean = self.mean.to(x.device, x.dtype)
        std = self.std.to(x.device, x.dtype)
        return (x - mean) / std


class RandomHorizontalFlip:
    def __init__(self, p: float = 0.5):
        self.p = p

    def __call__(self, x):
        if random.random() < self.p:
            return torch.flip(x, dims=[-1])
        return x


class RandomResizedCrop:
    """Crop a random area fraction in ``scale`` with aspect ratio in
    ``ratio``, resize back to ``size`` (bilinear)."""

    def __init__(self, size: int, scale=(0.75, 1.0), ratio=(1.0, 1.0)):
        self.size = size
        self.scale = scale
        self.ratio = ratio

    def __call__(self, x):
        squeeze = x.dim() == 3
        if squeeze:
            x = x.unsqueeze(0)
        _, _, H, W = x.shape
        area = H * W * random.uniform(*self.scale)
        aspect = random.uniform(*self.ratio)
        h = min(H, max(1, int(round((area * aspect) ** 0.5))))
        w = min(W, max(1, int(round((area / aspect) ** 0.5))))
        top = random.randint(0, H - h)
        left = random.randint(0, W - w)
        x = x[:, :, top:top + h, left:left + w]
        x = F.interpolate(x, size=(self.size, self.size), mode="bilinear",
                          align_corners=False)
        return x.squeeze(0) if squeeze else x


class RandomErasing:
    def __init__(self, p: float = 0.25, scale=(0.02, 0.33), ratio=(0.3, 3.3)):
        self.p = p
        self.scale = scale
        self.ratio = ratio

    def __call__(self, x):
        if random.random() >= self.p:
            return x
        H, W = x.shape[-2:]
        for _ in range(10):
            area = H * W * random.uniform(*self.scale)
            aspect = random.uniform(*self.ratio)
            h = int(round((area * aspect) ** 0.5))
            w = int(round((area / aspect) ** 0.5))
            if h < H and w < W and h > 0 and w > 0:
                top = random.randint(0, H - h)
                left = random.randint(0, W - w)
                x = x.clone()
                x[..., top:top + h, left:left + w] = torch.randn(
                    (*x.shape[:-2], h, w), device=x.device, dtype=x.dtype)
                return x
        return x
