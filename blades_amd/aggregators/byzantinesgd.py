"""ByzantineSGD filtering (Alistarh et al. 2018, "Byzantine Stochastic
Gradient Descent").

The reference shipped this as dead code (class name ``ByzantineSGD`` never
matched its string registry — aggregators/byzantinesgd.py:16-80, SURVEY.md
§2.2); here it is a working, registered aggregator (name
``byzantinesgd``).  Per-worker running statistics:

  A_i += <g_i, θ − θ₀>   (cumulative correlation with the model path)
  B_i += g_i             (cumulative gradient)

Each round, workers whose A / B / current gradient deviate from the
respective medians beyond (th_A, th_B, 4·th_V) are permanently removed;
the mean of the surviving set is returned.  The vector medians use the
pairwise-distance machinery (MFMA Gram on GPU) instead of the reference's
O(K²) python norm loops.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from blades_amd import ops
from .base import _BaseAggregator


class Byzantinesgd(_BaseAggregator):
    def __init__(self, num_clients: int = 20, th_A: float = 10.0,
                 th_B: float = 10.0, th_V: float = 10.0,
                 model: Optional[torch.nn.Module] = None):
        super().__init__()
        self.m = num_clients
        self.th_A = th_A
        self.th_B = th_B
        self.th_V = th_V
        self._model = model
        self._theta0: Optional[torch.Tensor] = None
        self.A: Optional[torch.Tensor] = None
        self.B: Optional[torch.Tensor] = None
        self.good: List[int] = list(range(num_clients))

    def _flat_model(self) -> Optional[torch.Tensor]:
        if self._model is None:
            return None
        from blades_amd.engine.flat import ParamSpec

        return ParamSpec.from_module(self._model).flatten(self._model)

    @staticmethod
    def _vector_median(V: torch.Tensor, threshold: float) -> int:
        """Index of the first row within ``threshold`` of > m/2 rows."""
        D = ops.pairwise_sq_dists(V).sqrt()
        counts = (D <= threshold).sum(dim=1)  # includes self
        ok = (counts > V.shape[0] / 2).nonzero(as_tuple=False)
        if ok.numel() == 0:
            raise RuntimeError("No vector median found; loosen the threshold")
        return int(ok[0])

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        m = U.shape[0]
        if self.A is None:
            self.A = torch.zeros(m, device=U.device)
            self.B = torch.zeros_like(U)
            self.good = list(range(m))
        theta = self._flat_model()
        if theta is not None and self._theta0 is None:
            self._theta0 = theta.clone()
        model_diff = (theta - self._theta0).to(U.device) if theta is not None \
            else torch.zeros(U.shape[1], device=U.device)

        self.A += ops.row_dots(U, model_diff)
        self.B += U

        A_med = self.A.median()
        b_idx = self._vector_median(self.B, self.th_B)
        g_idx = self._vector_median(U, 2 * self.th_V)

        a_dev = (self.A - A_med).abs()
        b_dev = ops.row_diff_norms(self.B, self.B[b_idx])
        g_dev = ops.row_diff_norms(U, U[g_idx])
        keep = ((a_dev <= self.th_A) & (b_dev <= self.th_B)
                & (g_dev <= 4 * self.th_V))
        self.good = [i for i in self.good if bool(keep[i])]
        if not self.good:
            raise RuntimeError("ByzantineSGD filtered out every worker")
        sel = torch.tensor(self.good, device=U.device)
        return U[sel].mean(dim=0)

    def state_dict(self) -> dict:
        return {
            "A": None if self.A is None else self.A.cpu(),
            "B": None if self.B is None else self.B.cpu(),
            "good": list(self.good),
            "theta0": None if self._theta0 is None else self._theta0.cpu(),
        }

    def load_state_dict(self, state: dict) -> None:
        self.A = state.get("A")
        self.B = state.get("B")
        self.good = list(state.get("good", []))
        self._theta0 = state.get("theta0")

    def __str__(self):
        return "ByzantineSGD"
