"""Coordinate-wise trimmed mean (reference: aggregators/trimmedmean.py:9-45).

Per coordinate: drop the ``b`` largest and ``b`` smallest entries, average
the remaining K−2b.  HIP kernel K3 does the select+accumulate in one pass
over U (the reference materializes topk(U), topk(−U) and a 3K×d concat).

The reference silently shrinks ``b`` when K − 2b ≤ 0
(trimmedmean.py:29-36, flagged as a bug in SURVEY.md §2.1) — we keep that
behavior for API parity but emit a warning.
"""
from __future__ import annotations

import warnings

from blades_amd import ops
from .base import _BaseAggregator


class Trimmedmean(_BaseAggregator):
    coordinate_shardable = True

    def __init__(self, nb: int = 5):
        super().__init__()
        self.b = nb

    def __call__(self, inputs):
        U = self._get_updates(inputs)
        b = self.b
        if U.shape[0] - 2 * b <= 0:
            while b > 0 and U.shape[0] - 2 * b <= 0:
                b -= 1
            if b < 0 or U.shape[0] - 2 * b <= 0:
                raise RuntimeError(f"K={U.shape[0]} too small for any trim")
            warnings.warn(
                f"Trimmedmean: K={U.shape[0]} <= 2*nb={2 * self.b}; "
                f"shrinking b to {b} (reference-compatible behavior)"
            )
        return ops.trimmed_mean(U, b)

    def __str__(self):
        return f"Trimmed Mean (b={self.b})"
