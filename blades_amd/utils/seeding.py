"""Determinism helpers.

``set_random_seed`` keeps the reference surface (reference:
src/blades/utils.py:116-124).  ``client_philox_seed`` is new: the MI355X
runtime gives every (client, round) pair its own counter-based stream so
results are invariant to how clients are sharded across ranks (reference
relied on driver-side RNG cache/restore instead -- src/blades/simulator.py:153-165).
"""
from __future__ import annotations

import os
import random

import numpy as np
import torch


def set_random_seed(seed_value: int = 0, use_cuda: bool = False) -> None:
    if seed_value is None:
        seed_value = 0
    np.random.seed(seed_value)
    random.seed(seed_value)
    torch.manual_seed(seed_value)
    os.environ["PYTHONHASHSEED"] = str(seed_value)
    if use_cuda and torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed_value)
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False


_PHILOX_MIX = 0x9E3779B97F4A7C15


def client_philox_seed(base_seed: int, client_id: int, round_idx: int, tag: int = 0) -> int:
    """Deterministic 63-bit stream key for (client, round, tag).

    A splitmix-style hash: layout-invariant (does not depend on which rank
    hosts the client) and collision-resistant across the population sizes
    this framework targets (K <= 1e4, rounds <= 1e6).
    """
    x = (base_seed & 0xFFFFFFFFFFFFFFFF) ^ (client_id * 0xBF58476D1CE4E5B9) ^ (
        round_idx * 0x94D049BB133111EB
    ) ^ (tag * _PHILOX_MIX)
    x &= 0xFFFFFFFFFFFFFFFF
    x ^= x >> 30
    x = (x * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    x ^= x >> 27
    x = (x * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    x ^= x >> 31
    return x & 0x7FFFFFFFFFFFFFFF
