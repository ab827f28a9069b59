"""Centered clipping (reference: aggregators/centeredclipping.py:13-49).

Karimireddy et al., "Learning from History for Byzantine Robust
Optimization" (ICML 2021).  Momentum-centered clipping iterated n_iter
times: v ← v + mean_k(clip(u_k − v, τ)).  STATEFUL: the momentum vector
persists across rounds (reference: centeredclipping.py:27,37-38) and is
checkpointed via state_dict.  HIP kernel K7 fuses the row-norm + scaled
centered mean into one slab pass per iteration.
"""
from __future__ import annotations

from typing import Optional

import torch

from blades_amd import ops
from .base import _BaseAggregator


class Centeredclipping(_BaseAggregator):
    def __init__(self, tau: float = 10.0, n_iter: int = 5):
        super().__init__()
        self.tau = tau
        self.n_iter = n_iter
        self.momentum: Optional[torch.Tensor] = None

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        if self.momentum is None:
            self.momentum = torch.zeros(U.shape[1], device=U.device, dtype=U.dtype)
        else:
            if self.momentum.numel() > U.shape[1]:
                # momentum from a coordinate-sharded run carries zero pad
                # columns at the tail; trim to the true d
                self.momentum = self.momentum[:U.shape[1]]
            self.momentum = self.momentum.to(device=U.device, dtype=U.dtype)
        for _ in range(self.n_iter):
            self.momentum = ops.centered_clip_iter(U, self.momentum, self.tau)
        return self.momentum.clone().detach()

    supports_shard = True

    def aggregate_shard(self, U_shard: torch.Tensor, runtime) -> torch.Tensor:
        """Distributed centered clipping: per-iteration row norms from
        partial sq-norms + all-reduce; the momentum vector is kept FULL on
        every rank (d floats — tiny next to the slab) with only this rank's
        slice updated per iteration and re-assembled by one all-gather at
        the end, so checkpointing stays rank-independent."""
        from blades_amd import ops

        K, dshard = U_shard.shape
        full_len = dshard * runtime.world_size
        if self.momentum is None:
            self.momentum = torch.zeros(full_len, device=U_shard.device,
                                        dtype=U_shard.dtype)
        else:
            self.momentum = self.momentum.to(device=U_shard.device,
                                             dtype=U_shard.dtype)
            if self.momentum.numel() != full_len:
                # momentum carried over from a different world size / gather
                # layout (e.g. full-gather ↔ coordinate-shard, or resume at
                # another ws): the first true-d coordinates are the state,
                # everything past them is zero pad — re-layout, don't drop.
                m = self.momentum
                if m.numel() < full_len:
                    m = torch.cat([m, torch.zeros(full_len - m.numel(),
                                                  device=m.device,
                                                  dtype=m.dtype)])
                else:
                    m = m[:full_len].clone()
                self.momentum = m
        lo = runtime.rank * dshard
        v_shard = self.momentum[lo:lo + dshard].clone()
        for _ in range(self.n_iter):
            part = ops.row_diff_norms(U_shard, v_shard) ** 2
            runtime.all_reduce_(part)
            norms = part.sqrt().clamp_min(1e-12)
            scale = torch.clamp(self.tau / norms, max=1.0) / K
            v_shard = v_shard * (1.0 - scale.sum()) \
                + ops.weighted_col_sum(U_shard, scale)
        self.momentum = runtime.all_gather_flat(v_shard)
        return v_shard

    def state_dict(self) -> dict:
        return {"momentum": None if self.momentum is None else self.momentum.cpu()}

    def load_state_dict(self, state: dict) -> None:
        m = state.get("momentum")
        self.momentum = None if m is None else m.clone()

    def __str__(self):
        return f"Clipping (tau={self.tau}, n_iter={self.n_iter})"
