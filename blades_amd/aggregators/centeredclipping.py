"""Centered clipping (reference: aggregators/centeredclipping.py:13-49).

Karimireddy et al., "Learning from History for Byzantine Robust
Optimization" (ICML 2021).  Momentum-centered clipping iterated n_iter
times: v ← v + mean_k(clip(u_k − v, τ)).  STATEFUL: the momentum vector
persists across rounds (reference: centeredclipping.py:27,37-38) and is
checkpointed via state_dict.  HIP kernel K7 fuses the row-norm + scaled
centered mean into one slab pass per iteration.
"""
from __future__ import annotations

from typing import Optional

import torch

from blades_amd import ops
from .base import _BaseAggregator


class Centeredclipping(_BaseAggregator):
    def __init__(self, tau: float = 10.0, n_iter: int = 5):
        super().__init__()
        self.tau = tau
        self.n_iter = n_iter
        self.momentum: Optional[torch.Tensor] = None

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        if self.momentum is None:
            self.momentum = torch.zeros(U.shape[1], device=U.device, dtype=U.dtype)
        else:
            self.momentum = self.momentum.to(device=U.device, dtype=U.dtype)
        for _ in range(self.n_iter):
            self.momentum = ops.centered_clip_iter(U, self.momentum, self.tau)
        return self.momentum.clone().detach()

    def state_dict(self) -> dict:
        return {"momentum": None if self.momentum is None else self.momentum.cpu()}

    def load_state_dict(self, state: dict) -> None:
        m = state.get("momentum")
        self.momentum = None if m is None else m.clone()

    def __str__(self):
        return f"Clipping (tau={self.tau}, n_iter={self.n_iter})"
