"""Client execution engines.

* :class:`FusedEngine` — batched many-model training over a [C, d] HBM slab
  (the MI355X replacement for the reference's sequential Ray-actor loop).
* :class:`LoopEngine`  — reference-exact per-client loop, used for custom
  client subclasses and as the semantics oracle.

``split_fusable`` partitions a client population between them.
"""
from __future__ import annotations

from typing import List, Tuple

from blades_amd.client import BladesClient, uses_default_training
from .flat import ParamSpec
from .fused import FusedEngine, make_vmap_safe
from .loop import LoopEngine


def split_fusable(clients: List[BladesClient]) -> Tuple[List[BladesClient], List[BladesClient]]:
    fusable, custom = [], []
    for c in clients:
        (fusable if uses_default_training(c) else custom).append(c)
    return fusable, custom


__all__ = ["ParamSpec", "FusedEngine", "LoopEngine", "make_vmap_safe",
           "split_fusable"]
