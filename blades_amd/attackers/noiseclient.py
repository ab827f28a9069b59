"""Noise attack (reference: attackers/noiseclient.py:8-25).

Uploads N(mean, std) noise of the update's shape.  K12: drawn from a
counter-based per-(client, round) Philox stream so the result is identical
regardless of rank sharding (the reference drew from the global CPU RNG).
"""
from __future__ import annotations

from typing import Optional

import torch

from blades_amd.client import ByzantineClient
from blades_amd.ops import philox_normal
from blades_amd.utils import client_philox_seed


class NoiseClient(ByzantineClient):
    def __init__(self, mean: Optional[float] = 0.1, std: Optional[float] = 0.1,
                 *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._noise_mean = mean
        self._noise_std = std

    def omniscient_callback(self, simulator) -> None:
        cur = self.get_update()
        ctx = getattr(simulator, "_attack_ctx", None)
        if ctx is not None:
            seed = client_philox_seed(ctx.base_seed, ctx.row_of(self), ctx.round, tag=12)
            self._state["saved_update"] = philox_normal(
                cur.shape, self._noise_mean, self._noise_std, seed,
                device=cur.device, dtype=cur.dtype)
        else:
            self._state["saved_update"] = torch.normal(
                self._noise_mean, self._noise_std, size=cur.shape).to(cur.device)
