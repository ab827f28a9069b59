"""AutoGM — geometric median with auto-tuned weights
(reference: aggregators/autogm.py:15-65).

Outer loop: given current median z, compute per-client distances, find the
water-filling threshold η* over the sorted distances
(η = (Σ_{i≤p} d_i + λ)/(p+1), largest η with η ≥ d_p), set
α_k = max(η* − d_k, 0)/λ, re-run weighted GeoMed; stop on relative ftol.
"""
from __future__ import annotations

from typing import Optional

import torch

from blades_amd import ops
from .base import _BaseAggregator
from .geomed import Geomed


class Autogm(_BaseAggregator):
    supports_shard = True

    def __init__(self, lamb: Optional[float] = None, maxiter: int = 100,
                 eps: float = 1e-6, ftol: float = 1e-10):
        super().__init__()
        self.lamb = lamb
        self.maxiter = maxiter
        self.eps = eps
        self.ftol = ftol
        self.gm_agg = Geomed(maxiter=maxiter, eps=eps, ftol=ftol)

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        K = U.shape[0]
        lamb = float(K) if self.lamb is None else self.lamb
        alpha = torch.full((K,), 1.0 / K, device=U.device, dtype=U.dtype)

        median = self.gm_agg(U, alpha)
        dists = ops.row_diff_norms(U, median)
        obj = (alpha * dists).sum()
        global_obj = obj + lamb * (alpha * alpha).sum() / 2

        for _ in range(self.maxiter):
            prev_global_obj = global_obj
            dists = ops.row_diff_norms(U, median)
            d_sorted, _ = torch.sort(dists)
            # water-filling threshold (reference: autogm.py:52-58)
            csum = torch.cumsum(d_sorted, dim=0)
            p = torch.arange(1, K + 1, device=U.device, dtype=U.dtype)
            etas = (csum + lamb) / p
            valid = etas - d_sorted >= 0
            eta_optimal = etas[valid][-1] if valid.any() else etas[0]
            alpha = torch.clamp(eta_optimal - dists, min=0) / lamb

            median = self.gm_agg(U, alpha)
            dists = ops.row_diff_norms(U, median)
            gm_sum = (alpha * dists).sum()
            global_obj = gm_sum + lamb * (alpha * alpha).sum() / 2
            if torch.abs(prev_global_obj - global_obj) < self.ftol * torch.abs(global_obj):
                break
        return median

    def aggregate_shard(self, U_shard, runtime):
        """Distributed AutoGM: the inner GeoMed runs its shard-aware form;
        per-client distances come from partial norms + all-reduce, so the
        water-filling weight update is replicated exactly on every rank."""
        K = U_shard.shape[0]
        lamb = float(K) if self.lamb is None else self.lamb
        alpha = torch.full((K,), 1.0 / K, device=U_shard.device,
                           dtype=U_shard.dtype)

        def dists_for(z_shard):
            part = ops.row_diff_norms(U_shard, z_shard) ** 2
            runtime.all_reduce_(part)
            return part.sqrt()

        median = self.gm_agg.aggregate_shard(U_shard, runtime, alpha)
        dists = dists_for(median)
        obj = (alpha * dists).sum()
        global_obj = obj + lamb * (alpha * alpha).sum() / 2
        for _ in range(self.maxiter):
            prev_global_obj = global_obj
            dists = dists_for(median)
            d_sorted, _ = torch.sort(dists)
            csum = torch.cumsum(d_sorted, dim=0)
            p = torch.arange(1, K + 1, device=U_shard.device,
                             dtype=U_shard.dtype)
            etas = (csum + lamb) / p
            valid = etas - d_sorted >= 0
            eta_optimal = etas[valid][-1] if valid.any() else etas[0]
            alpha = torch.clamp(eta_optimal - dists, min=0) / lamb
            median = self.gm_agg.aggregate_shard(U_shard, runtime, alpha)
            dists = dists_for(median)
            gm_sum = (alpha * dists).sum()
            global_obj = gm_sum + lamb * (alpha * alpha).sum() / 2
            if torch.abs(prev_global_obj - global_obj) < self.ftol * torch.abs(global_obj):
                break
        return median

    def __str__(self):
        return "AutoGM"
