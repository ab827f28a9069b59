"""Aggregator base class.

Input contract matches the reference (aggregators/mean.py:21-39): a callable
over ``List[BladesClient] | List[Tensor] | Tensor`` returning the aggregated
d-vector.  ``_get_updates`` stacks client updates into U ∈ R^{K×d}.

Two framework-internal attributes drive the MI355X runtime:

* ``coordinate_shardable`` — True when agg(U) restricted to a coordinate
  slice equals the slice of agg(U) (Mean/Median/TrimmedMean...).  The 8-rank
  runtime then aggregates rank-local [K, d/8] shards with no full-U gather
  (SURVEY.md §5.7).
* ``state_dict()/load_state_dict()`` — stateful aggregators (momentum, norm
  history) expose their state for checkpointing (SURVEY.md §7 hard-part 6).
"""
from __future__ import annotations

from typing import List, Union

import torch

Tensor = torch.Tensor


class _BaseAggregator:
    coordinate_shardable: bool = False

    def __init__(self, *args, **kwargs):
        pass

    def _get_updates(self, inputs) -> Tensor:
        # late import to avoid a cycle (client.py imports aggregators for docs)
        from blades_amd.client import BladesClient

        if isinstance(inputs, torch.Tensor):
            return inputs
        if isinstance(inputs, (list, tuple)):
            if all(isinstance(e, BladesClient) for e in inputs):
                return torch.stack([w.get_update() for w in inputs])
            if all(isinstance(e, torch.Tensor) for e in inputs):
                return torch.stack(list(inputs), dim=0)
        raise TypeError(
            "aggregator input must be a Tensor, a list of Tensors, or a list "
            f"of BladesClient; got {type(inputs)}"
        )

    def __call__(self, inputs) -> Tensor:
        raise NotImplementedError

    # ---- coordinate-sharded execution
    # Row-wise aggregators (GeoMed/Krum/...) override aggregate_shard to run
    # on a [K, d/ws] coordinate shard: shard-local column math + all-reduce
    # of the small K-length / K×K row statistics.  Coordinate-wise
    # aggregators don't need it (their __call__ on the shard IS the shard
    # of the answer).
    supports_shard: bool = False

    def aggregate_shard(self, U_shard: Tensor, runtime) -> Tensor:
        """Return this rank's d/ws slice of the aggregate."""
        raise NotImplementedError(
            f"{type(self).__name__} has no shard-aware path")

    # ---- checkpointing hooks (stateless default)
    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, state: dict) -> None:
        pass
