"""Coordinate-wise median (reference: aggregators/median.py:9-25).

Yin et al., "Byzantine-robust distributed learning: Towards optimal
statistical rates" (PMLR v80).  For even K the two middle order statistics
are averaged, matching the reference's ``(median(U) - median(-U))/2``.
HIP kernel K2: in-LDS selection per coordinate tile.
"""
from __future__ import annotations

from blades_amd import ops
from .base import _BaseAggregator


class Median(_BaseAggregator):
    coordinate_shardable = True

    def __call__(self, inputs):
        return ops.col_median(self._get_updates(inputs))

    def __str__(self):
        return "Coordinate-wise Median"
