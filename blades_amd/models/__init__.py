"""Model zoo.

Reference zoo: MLP (MNIST), CCT (CIFAR-10).  BASELINE.json
additionally requires ResNet-18 and WideResNet-28-10 — provided here.

``get_model(name, **kw)`` is the string registry used by the CLI and bench.
"""
from __future__ import annotations

from torch import nn

from .mlp import MLP
from .resnet import ResNet, WideResNet, resnet18, wide_resnet28_10
from .cct import CCT, CCTNet, cct_2_3x2_32, cvt_2_4_32, vit_lite_2_4_32
from .text_cct import (TextCCT, text_cct_2, text_cct_4, text_cct_6,
                       text_transformer_2, text_transformer_4,
                       text_transformer_6)

_REGISTRY = {
    "mlp": lambda **kw: MLP(**kw),
    "resnet18": lambda **kw: resnet18(**kw),
    "wrn28_10": lambda **kw: wide_resnet28_10(**kw),
    "wideresnet28_10": lambda **kw: wide_resnet28_10(**kw),
    "cct": lambda **kw: cct_2_3x2_32(**kw),
    "cct_2_3x2_32": lambda **kw: cct_2_3x2_32(**kw),
    "text_cct_2": lambda **kw: text_cct_2(**kw),
    "text_cct_4": lambda **kw: text_cct_4(**kw),
    "text_cct_6": lambda **kw: text_cct_6(**kw),
    "text_transformer_2": lambda **kw: text_transformer_2(**kw),
    "cvt": lambda **kw: cvt_2_4_32(**kw),
    "vit_lite": lambda **kw: vit_lite_2_4_32(**kw),
}


def get_model(name: str, **kw) -> nn.Module:
    try:
        factory = _REGISTRY[name.lower()]
    except KeyError:
        raise KeyError(f"unknown model {name!r}; available: {sorted(_REGISTRY)}")
    return factory(**kw)


def register_model(name: str):
    def deco(fn):
        _REGISTRY[name.lower()] = fn
        return fn
    return deco


def num_params(model: nn.Module) -> int:
    return sum(p.numel() for p in model.parameters() if p.requires_grad)


__all__ = [
    "MLP", "ResNet", "WideResNet", "CCT", "CCTNet",
    "resnet18", "wide_resnet28_10", "cct_2_3x2_32", "cvt_2_4_32",
    "vit_lite_2_4_32",
    "get_model", "register_model", "num_params",
]
