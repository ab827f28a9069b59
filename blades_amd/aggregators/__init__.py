"""Robust-aggregator registry.

Same importable names as the reference registry
(reference: aggregators/__init__.py:10-18, which exports 8 schemes;
``fltrust`` was string-loadable but unexported there — exported here), plus
``multikrum``.  The string-resolution rule the reference's Simulator uses
(module ``blades.aggregators.<name>``, class ``<Name.capitalize()>``,
simulator.py:112-114) works against this package unchanged.
"""
from __future__ import annotations

from .base import _BaseAggregator
from .mean import Mean
from .median import Median
from .trimmedmean import Trimmedmean
from .krum import Krum, Multikrum
from .geomed import Geomed
from .autogm import Autogm
from .centeredclipping import Centeredclipping
from .clustering import Clustering
from .clippedclustering import Clippedclustering
from .fltrust import Fltrust
from .byzantinesgd import Byzantinesgd
from .async_ import (_AsyncCenteredClipping, _AsyncMean,
                     _BaseAsyncAggregator, _DecentralizedAggregator)

_REGISTRY = {
    "mean": Mean,
    "median": Median,
    "trimmedmean": Trimmedmean,
    "krum": Krum,
    "multikrum": Multikrum,
    "geomed": Geomed,
    "autogm": Autogm,
    "centeredclipping": Centeredclipping,
    "clustering": Clustering,
    "clippedclustering": Clippedclustering,
    "fltrust": Fltrust,
    "byzantinesgd": Byzantinesgd,
}


def get_aggregator(name: str, **kwargs) -> _BaseAggregator:
    try:
        cls = _REGISTRY[name.lower()]
    except KeyError:
        raise KeyError(f"unknown aggregator {name!r}; available: {sorted(_REGISTRY)}")
    return cls(**kwargs)


def register_aggregator(name: str):
    def deco(cls):
        _REGISTRY[name.lower()] = cls
        return cls
    return deco


__all__ = [
    "Mean", "Median", "Trimmedmean", "Krum", "Multikrum", "Geomed", "Autogm",
    "Centeredclipping", "Clustering", "Clippedclustering", "Fltrust",
    "Byzantinesgd",
    "get_aggregator", "register_aggregator", "_BaseAggregator",
    "_BaseAsyncAggregator", "_AsyncMean", "_AsyncCenteredClipping",
    "_DecentralizedAggregator",
]
