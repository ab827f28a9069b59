"""Asynchronous / decentralized aggregation bases.

Working equivalents of the reference's async machinery (reference:
aggregators/mean.py:42-116, centeredclipping.py:52-137 — most of it dead
code there; functional and unit-tested here).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from blades_amd import ops


class _BaseAsyncAggregator:
    """Aggregates a partially-arrived round: ``inputs`` may contain None for
    clients whose update has not arrived."""

    def __call__(self, inputs: List[Optional[torch.Tensor]]) -> torch.Tensor:
        raise NotImplementedError


class _AsyncMean(_BaseAsyncAggregator):
    """Mean over arrived updates, normalized by the FULL population size
    (missing clients contribute zero — reference semantics,
    aggregators/mean.py:79-86)."""

    def __call__(self, inputs):
        filtered = [x for x in inputs if x is not None]
        if not filtered:
            raise ValueError("no updates arrived")
        return torch.stack(filtered, dim=0).sum(dim=0) / len(inputs)


class _AsyncCenteredClipping(_BaseAsyncAggregator):
    """Centered clipping over the arrived subset (momentum persists)."""

    def __init__(self, tau: float = 10.0, n_iter: int = 5):
        self.tau = tau
        self.n_iter = n_iter
        self.momentum: Optional[torch.Tensor] = None

    def __call__(self, inputs):
        arrived = [x for x in inputs if x is not None]
        if not arrived:
            raise ValueError("no updates arrived")
        U = torch.stack(arrived)
        if self.momentum is None:
            self.momentum = torch.zeros(U.shape[1], device=U.device,
                                        dtype=U.dtype)
        for _ in range(self.n_iter):
            self.momentum = ops.centered_clip_iter(U, self.momentum, self.tau)
        return self.momentum.clone()


class _DecentralizedAggregator:
    """Gossip step: s = Σ_j w_j · x_j over a node's neighborhood, driven by
    one row of a mixing matrix (reference: aggregators/mean.py:89-116)."""

    def __init__(self, node_index: int, neighbor_indices: List[int],
                 weights: torch.Tensor):
        assert weights.dim() == 1
        self.node_index = node_index
        self.neighbor_indices = list(neighbor_indices)
        self.weights = weights

    def __call__(self, inputs: List[torch.Tensor]) -> torch.Tensor:
        assert len(inputs) == 1 + len(self.neighbor_indices), \
            "inputs = [own update, *neighbor updates]"
        s = self.weights[self.node_index] * inputs[0]
        for idx, inp in zip(self.neighbor_indices, inputs[1:]):
            s = s + self.weights[idx] * inp
        return s
