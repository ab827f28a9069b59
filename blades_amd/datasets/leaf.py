"""LEAF-style federated dataset utilities.

Functional equivalents of the reference's vendored LEAF CLI helpers
(reference: models/utils/sample.py, split_data.py, util.py — standalone
scripts there; importable functions here).  Used for building custom
federated splits from flat (x, y) arrays.
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import numpy as np


def iid_divide(items: Sequence, g: int) -> List[list]:
    """Divide ``items`` into g groups as evenly as possible
    (reference: models/utils/util.py iid_divide)."""
    items = list(items)
    num_elems = len(items)
    group_size = num_elems // g
    num_big = num_elems - group_size * g
    glist = []
    bi = 0
    for i in range(g):
        size = group_size + (1 if i < num_big else 0)
        glist.append(items[bi:bi + size])
        bi += size
    return glist


def sample_iid(y: np.ndarray, num_users: int,
               rng: np.random.RandomState = None) -> Dict[int, np.ndarray]:
    """Uniform sample split across users."""
    rng = rng or np.random
    idx = rng.permutation(len(y))
    return {u: np.asarray(part) for u, part in enumerate(iid_divide(idx, num_users))}


def sample_noniid(y: np.ndarray, num_users: int, shards_per_user: int = 2,
                  rng: np.random.RandomState = None) -> Dict[int, np.ndarray]:
    """Shard-based label-skew sampling (LEAF / McMahan-style): sort by
    label, cut into ``num_users*shards_per_user`` shards, deal
    ``shards_per_user`` random shards to each user."""
    rng = rng or np.random
    order = np.argsort(y, kind="stable")
    shards = iid_divide(order, num_users * shards_per_user)
    shard_ids = rng.permutation(len(shards))
    out: Dict[int, np.ndarray] = {}
    for u in range(num_users):
        take = shard_ids[u * shards_per_user:(u + 1) * shards_per_user]
        out[u] = np.concatenate([shards[s] for s in take])
    return out


def train_test_split(indices: np.ndarray, frac: float = 0.9,
                     rng: np.random.RandomState = None
                     ) -> Tuple[np.ndarray, np.ndarray]:
    """Per-user train/test split (reference: models/utils/split_data.py)."""
    rng = rng or np.random
    idx = rng.permutation(indices)
    cut = int(len(idx) * frac)
    return idx[:cut], idx[cut:]


def remove_small_users(user_idx: Dict[int, np.ndarray],
                       min_samples: int) -> Dict[int, np.ndarray]:
    """Drop users with fewer than ``min_samples`` samples
    (reference: models/utils/remove_users.py)."""
    return {u: v for u, v in user_idx.items() if len(v) >= min_samples}
