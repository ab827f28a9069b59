"""Krum / Multi-Krum (reference: aggregators/krum.py:93-125).

Blanchard et al., "Machine Learning with Adversaries: Byzantine Tolerant
Gradient Descent" (NeurIPS 2017).

score(i) = Σ of the n−f−2 smallest ‖u_i − u_j‖² (j ≠ i); the m lowest-score
updates are summed.  The reference computes K(K−1)/2 separate norm calls in
a Python double loop (krum.py:73-90); here the distance matrix comes from
one Gram matrix ‖u_i‖² + ‖u_j‖² − 2·U·Uᵀ — HIP kernels K4 (MFMA f32 Gram)
+ K5 (row-select scores).
"""
from __future__ import annotations

from blades_amd import ops
from .base import _BaseAggregator


class Krum(_BaseAggregator):
    supports_shard = True

    def __init__(self, num_clients: int = 20, num_byzantine: int = 5, m: int = 1):
        super().__init__()
        self.n = num_clients
        self.f = num_byzantine
        self.m = m

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        # scores always use the actual population size, not the configured
        # one (the reference would silently mis-score on a mismatch)
        n = U.shape[0]
        if 2 * self.f + 2 > n:
            raise ValueError(f"Too many Byzantine workers: 2*{self.f}+2 > {n}")
        D = ops.pairwise_sq_dists(U)
        scores = ops.krum_scores(D, self.f)
        top_m = scores.argsort()[: self.m]
        return U[top_m].sum(dim=0)

    def aggregate_shard(self, U_shard, runtime):
        """Distributed Krum (SURVEY.md §5.7/§7 hard-part 3): each rank
        computes the partial Gram of its coordinate shard (MFMA kernel K4),
        one all-reduce of the K×K matrix assembles the exact pairwise
        distances, selection is replicated, and the winning rows are summed
        shard-locally."""
        import torch as _t

        from blades_amd import ops

        n = U_shard.shape[0]
        if 2 * self.f + 2 > n:
            raise ValueError(f"Too many Byzantine workers: 2*{self.f}+2 > {n}")
        G = ops.gram(U_shard)
        runtime.all_reduce_(G)
        sq = G.diagonal()
        D = (sq.unsqueeze(0) + sq.unsqueeze(1) - 2 * G).clamp_min_(0)
        D.fill_diagonal_(0)
        scores = ops.krum_scores(D, self.f)
        top_m = scores.argsort()[: self.m]
        return U_shard[top_m].sum(dim=0)

    def __str__(self):
        return f"Krum (m={self.m})"


class Multikrum(Krum):
    """Multi-Krum with m>1 selected updates (string-registry name
    ``multikrum``)."""

    def __init__(self, num_clients: int = 20, num_byzantine: int = 5, m: int = 5):
        super().__init__(num_clients, num_byzantine, m)

    def __str__(self):
        return f"Multi-Krum (m={self.m})"
