"""Clipped clustering (reference: aggregators/clippedclustering.py:20-66).

Li et al., "An Experimental Study of Byzantine-Robust Aggregation Schemes
in Federated Learning".  Each update is clipped to the median of the
HISTORICAL L2 norms (state that grows across rounds and is checkpointed),
then the cosine/agglomerative scheme of :mod:`clustering` selects the larger
cluster to average.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from blades_amd import ops
from .base import _BaseAggregator
from .clustering import complete_linkage_two_clusters


class Clippedclustering(_BaseAggregator):
    def __init__(self, tau: Optional[float] = None):
        super().__init__()
        self.tau = tau
        self.l2norm_his: List[float] = []

    def __call__(self, inputs):
        U = self._get_updates(inputs).clone()
        K = U.shape[0]
        norms = ops.row_norms(U)
        self.l2norm_his.extend(norms.cpu().tolist())
        threshold = self.tau if self.tau else float(np.median(self.l2norm_his))
        thr = torch.full_like(norms, threshold)
        ops.row_clip_to_norm_(U, thr)

        cos = ops.cos_sim_gram(U)
        sim = cos.cpu().numpy()
        np.fill_diagonal(sim, 1.0)
        sim = np.nan_to_num(sim, nan=-1.0, posinf=1.0, neginf=-1.0)
        labels = complete_linkage_two_clusters(1.0 - sim)
        flag = 1 if labels.sum() > K // 2 else 0
        sel = torch.from_numpy(labels == flag).to(U.device)
        return ops.col_mean(U[sel])

    supports_shard = True

    def aggregate_shard(self, U_shard, runtime):
        """Distributed form: global row norms from partial sq-norms +
        all-reduce (the norm history stays replicated — every rank sees the
        same norms), shard rows clipped by the global threshold, then the
        shard-aware clustering scheme."""
        from .clustering import Clustering

        U = U_shard.clone()
        K = U.shape[0]
        sqn = ops.row_sq_norms(U)
        runtime.all_reduce_(sqn)
        norms = sqn.sqrt()
        self.l2norm_his.extend(norms.cpu().tolist())
        threshold = self.tau if self.tau else float(np.median(self.l2norm_his))
        scale = torch.clamp(threshold / norms.clamp_min(1e-12), max=1.0)
        U.mul_(scale.unsqueeze(1))
        G = ops.gram(U)
        runtime.all_reduce_(G)
        sel = Clustering._select_larger_cluster(G)
        return ops.col_mean(U[sel])

    def state_dict(self) -> dict:
        return {"l2norm_his": list(self.l2norm_his)}

    def load_state_dict(self, state: dict) -> None:
        self.l2norm_his = list(state.get("l2norm_his", []))

    def __str__(self):
        return "ClippedClustering"
