"""Mean aggregation (reference: aggregators/mean.py:62-76)."""
from __future__ import annotations

from blades_amd import ops
from .base import _BaseAggregator


class Mean(_BaseAggregator):
    r"""Sample mean over the updates of all given clients (HIP kernel K1)."""

    coordinate_shardable = True

    def __call__(self, inputs):
        return ops.col_mean(self._get_updates(inputs))

    def __str__(self):
        return "Mean"
