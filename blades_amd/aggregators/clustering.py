"""Agglomerative-clustering defense (reference: aggregators/clustering.py:13-44).

Sattler et al., "On the byzantine robustness of clustered federated
learning".  Build the K×K cosine-similarity matrix (the reference uses a
scipy double loop; here one row-normalized Gram — HIP kernel K8 on MFMA),
split the population into 2 clusters by complete-linkage agglomerative
clustering on distance = 1 − cos, and average the LARGER cluster.

The K×K clustering itself stays host-side (K ≤ 1e4 ⇒ tiny vs the slab math),
implemented here directly (single-linkage-free Lance-Williams update) so the
framework does not depend on sklearn at runtime; sklearn, when present, is
used in tests as the cross-check.
"""
from __future__ import annotations

import numpy as np
import torch

from blades_amd import ops
from .base import _BaseAggregator


def complete_linkage_two_clusters(dist: np.ndarray) -> np.ndarray:
    """Agglomerative clustering, complete linkage, until 2 clusters remain.

    Returns a 0/1 label per row.  Equivalent to sklearn's
    AgglomerativeClustering(metric='precomputed', linkage='complete',
    n_clusters=2) (verified in tests/test_aggregators.py).
    """
    n = dist.shape[0]
    D = dist.astype(np.float64).copy()
    np.fill_diagonal(D, np.inf)
    clusters = {i: [i] for i in range(n)}
    while len(clusters) > 2:
        keys = list(clusters)
        sub = D[np.ix_(keys, keys)]
        i_idx, j_idx = divmod(np.argmin(sub), sub.shape[1])
        a, b = keys[i_idx], keys[j_idx]
        if a > b:
            a, b = b, a
        # complete linkage: new dist = max of the two
        for k in keys:
            if k in (a, b):
                continue
            D[a, k] = D[k, a] = max(D[a, k], D[b, k])
        D[b, :] = np.inf
        D[:, b] = np.inf
        clusters[a].extend(clusters[b])
        del clusters[b]
    labels = np.zeros(n, dtype=np.int64)
    for lbl, members in enumerate(clusters.values()):
        labels[members] = lbl
    return labels


class Clustering(_BaseAggregator):
    supports_shard = True

    @staticmethod
    def _select_larger_cluster(G: torch.Tensor) -> torch.Tensor:
        """2-cluster complete linkage on 1−cos from a full Gram; returns the
        bool row mask of the larger cluster."""
        K = G.shape[0]
        norms = G.diagonal().sqrt().clamp_min(1e-8)
        cos = (G / norms.unsqueeze(0) / norms.unsqueeze(1)).clamp(-1.0, 1.0)
        sim = cos.cpu().numpy()
        np.fill_diagonal(sim, 1.0)
        sim = np.nan_to_num(sim, nan=-1.0, posinf=1.0, neginf=-1.0)
        labels = complete_linkage_two_clusters(1.0 - sim)
        flag = 1 if labels.sum() > K // 2 else 0
        return torch.from_numpy(labels == flag).to(G.device)

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        sel = self._select_larger_cluster(ops.gram(U))
        return ops.col_mean(U[sel])

    def aggregate_shard(self, U_shard, runtime):
        """Partial Gram + all-reduce; clustering (tiny, host-side) is
        replicated on every rank, the cluster mean stays shard-local."""
        G = ops.gram(U_shard)
        runtime.all_reduce_(G)
        sel = self._select_larger_cluster(G)
        return ops.col_mean(U_shard[sel])

    def __str__(self):
        return "Clustering"
