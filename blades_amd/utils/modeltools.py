"""Model parameter helpers (reference: src/blades/utils.py:98-114)."""
from __future__ import annotations

import torch
from torch import nn


def reset_model_weights(model: nn.Module) -> None:
    """Re-initialize every submodule that defines ``reset_parameters``."""

    @torch.no_grad()
    def weight_reset(m: nn.Module):
        reset_parameters = getattr(m, "reset_parameters", None)
        if callable(reset_parameters):
            reset_parameters()

    model.apply(fn=weight_reset)
