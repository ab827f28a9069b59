"""Population partitioners: iid split and Dirichlet(α) label-skew.

One shared, tested implementation of the non-iid loop every reference
dataset class duplicated (reference: datasets/cifar10.py:70-101,
datasets/mnist.py analogous).
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np


def iid_partition(n_samples: int, num_clients: int) -> List[np.ndarray]:
    return np.array_split(np.arange(n_samples), num_clients)


def dirichlet_partition(labels: np.ndarray, num_clients: int, alpha: float,
                        num_classes: int, min_size_floor: int = 10,
                        rng: np.random.RandomState = None) -> List[np.ndarray]:
    """Label-skew partition: per class, proportions ~ Dir(α) across clients,
    rejecting draws until every client has ≥ ``min_size_floor`` samples
    (reference: datasets/cifar10.py:76-101)."""
    if rng is None:
        rng = np.random
    N = labels.shape[0]
    min_size = 0
    while min_size < min_size_floor:
        idx_batch: List[List[int]] = [[] for _ in range(num_clients)]
        for k in range(num_classes):
            idx_k = np.where(labels == k)[0]
            rng.shuffle(idx_k)
            proportions = rng.dirichlet(np.repeat(alpha, num_clients))
            # cap clients that already exceed their fair share
            proportions = np.array([
                p * (len(idx_j) < N / num_clients)
                for p, idx_j in zip(proportions, idx_batch)
            ])
            proportions = proportions / proportions.sum()
            cuts = (np.cumsum(proportions) * len(idx_k)).astype(int)[:-1]
            idx_batch = [idx_j + idx.tolist()
                         for idx_j, idx in zip(idx_batch, np.split(idx_k, cuts))]
        min_size = min(len(idx_j) for idx_j in idx_batch)
    out = []
    for j in range(num_clients):
        rng.shuffle(idx_batch[j])
        out.append(np.asarray(idx_batch[j]))
    return out


def build_client_dicts(x: np.ndarray, y: np.ndarray,
                       splits: List[np.ndarray]) -> Tuple[List[str], Dict]:
    ids = [str(i) for i in range(len(splits))]
    data = {cid: {"x": x[idx], "y": y[idx].flatten()} for cid, idx in zip(ids, splits)}
    return ids, data
