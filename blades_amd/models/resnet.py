"""CIFAR-style ResNet-18 and WideResNet-28-10.

The reference does not ship these; the benchmark configs in
BASELINE.json require them (ResNet-18 d≈11.2M, WRN-28-10 d≈36.5M).  These are
standard architectures written for this framework with one extra knob:

``norm=``
    ``"batch"``       — nn.BatchNorm2d with running stats (eval parity with
                        torchvision-style training).
    ``"batch-local"`` — BatchNorm2d with ``track_running_stats=False``:
                        identical training-mode math, no in-place buffer
                        mutation, which is what the vmapped many-model client
                        engine requires (per-client running stats make no
                        sense in a federated population anyway).
    ``"group"``       — GroupNorm(min(32, C)) — the common FL substitution.

Use :func:`blades_amd.engine.make_vmap_safe` to convert an existing
``norm="batch"`` model in place.
"""
from __future__ import annotations

from typing import Callable

import torch.nn as nn
import torch.nn.functional as F


def _make_norm(norm: str) -> Callable[[int], nn.Module]:
    if norm == "batch":
        return lambda c: nn.BatchNorm2d(c)
    if norm == "batch-local":
        return lambda c: nn.BatchNorm2d(c, track_running_stats=False)
    if norm == "group":
        return lambda c: nn.GroupNorm(min(32, c), c)
    raise ValueError(f"unknown norm {norm!r}")


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes: int, planes: int, stride: int, norm_layer):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = norm_layer(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1, padding=1, bias=False)
        self.bn2 = norm_layer(planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_planes, planes, 1, stride=stride, bias=False),
                norm_layer(planes),
            )

    def forward(self, x):
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + self.shortcut(x)
        return F.relu(out)


class ResNet(nn.Module):
    """CIFAR ResNet (3x3 stem, no max-pool)."""

    def __init__(self, block, num_blocks, num_classes: int = 10, norm: str = "batch"):
        super().__init__()
        norm_layer = _make_norm(norm)
        self.in_planes = 64
        self.conv1 = nn.Conv2d(3, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = norm_layer(64)
        self.layer1 = self._make_layer(block, 64, num_blocks[0], 1, norm_layer)
        self.layer2 = self._make_layer(block, 128, num_blocks[1], 2, norm_layer)
        self.layer3 = self._make_layer(block, 256, num_blocks[2], 2, norm_layer)
        self.layer4 = self._make_layer(block, 512, num_blocks[3], 2, norm_layer)
        self.fc = nn.Linear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, n, stride, norm_layer):
        layers = []
        for s in [stride] + [1] * (n - 1):
            layers.append(block(self.in_planes, planes, s, norm_layer))
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def forward(self, x):
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def resnet18(num_classes: int = 10, norm: str = "batch") -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes=num_classes, norm=norm)


class WideBasic(nn.Module):
    def __init__(self, in_planes, planes, stride, norm_layer, dropout: float = 0.0):
        super().__init__()
        self.bn1 = norm_layer(in_planes)
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = norm_layer(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1, padding=1, bias=False)
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
        self.shortcut = None
        if stride != 1 or in_planes != planes:
            self.shortcut = nn.Conv2d(in_planes, planes, 1, stride=stride, bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x))
        sc = self.shortcut(out) if self.shortcut is not None else x
        out = self.conv1(out)
        out = self.conv2(self.dropout(F.relu(self.bn2(out))))
        return out + sc


class WideResNet(nn.Module):
    """WRN-depth-width (pre-activation) for 32x32 inputs."""

    def __init__(self, depth: int = 28, widen: int = 10, num_classes: int = 10,
                 norm: str = "batch", dropout: float = 0.0):
        super().__init__()
        assert (depth - 4) % 6 == 0, "depth must be 6n+4"
        n = (depth - 4) // 6
        norm_layer = _make_norm(norm)
        widths = [16, 16 * widen, 32 * widen, 64 * widen]
        self.conv1 = nn.Conv2d(3, widths[0], 3, stride=1, padding=1, bias=False)
        in_planes = widths[0]
        blocks = []
        for stage, w in enumerate(widths[1:]):
            for i in range(n):
                stride = (2 if stage > 0 else 1) if i == 0 else 1
                blocks.append(WideBasic(in_planes, w, stride, norm_layer, dropout))
                in_planes = w
        self.blocks = nn.Sequential(*blocks)
        self.bn = norm_layer(in_planes)
        self.fc = nn.Linear(in_planes, num_classes)

    def forward(self, x):
        out = self.conv1(x)
        out = self.blocks(out)
        out = F.relu(self.bn(out))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def wide_resnet28_10(num_classes: int = 10, norm: str = "batch") -> WideResNet:
    return WideResNet(28, 10, num_classes=num_classes, norm=norm)
