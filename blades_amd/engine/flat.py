"""Flat-parameter machinery.

The whole framework treats a model's trainable parameters as one flat fp32
vector in ``named_parameters`` order, restricted to ``requires_grad`` params
— the exact flattening the reference uses for updates (reference:
client.py:216-228) and the server uses for applying them (server.py:66-74).

``ParamSpec`` precomputes that layout once.  The client engine keeps the
whole population as a single [C, d] HBM slab; ``batched_views`` exposes each
parameter as a strided [C, *shape] view into it with zero copies, which is
what ``torch.func.functional_call``+``vmap`` consume directly.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Iterator, List, Optional, Tuple

import torch
import torch.nn as nn

Tensor = torch.Tensor


@dataclass
class ParamSpec:
    names: List[str]
    shapes: List[torch.Size]
    numels: List[int]
    offsets: List[int]  # start offset of each param in the flat vector
    d: int

    @classmethod
    def from_module(cls, model: nn.Module) -> "ParamSpec":
        names, shapes, numels, offsets = [], [], [], []
        off = 0
        for name, p in model.named_parameters():
            if not p.requires_grad:
                continue
            names.append(name)
            shapes.append(p.shape)
            numels.append(p.numel())
            offsets.append(off)
            off += p.numel()
        return cls(names=names, shapes=shapes, numels=numels, offsets=offsets, d=off)

    # ------------------------------------------------------------- flatten
    def iter_params(self, model: nn.Module) -> Iterator[nn.Parameter]:
        params = dict(model.named_parameters())
        for name in self.names:
            yield params[name]

    def flatten(self, model: nn.Module, device=None, out: Optional[Tensor] = None) -> Tensor:
        first = next(iter(self.iter_params(model)))
        dev = device if device is not None else first.device
        if out is None:
            out = torch.empty(self.d, device=dev, dtype=torch.float32)
        for p, off, n in zip(self.iter_params(model), self.offsets, self.numels):
            out[off:off + n].copy_(p.data.view(-1))
        return out

    @torch.no_grad()
    def load(self, model: nn.Module, vec: Tensor) -> None:
        for p, off, n in zip(self.iter_params(model), self.offsets, self.numels):
            p.data.copy_(vec[off:off + n].view_as(p.data))

    def slices(self, vec: Tensor) -> Iterator[Tensor]:
        for off, n in zip(self.offsets, self.numels):
            yield vec[off:off + n]

    def named_slices(self, vec: Tensor) -> Iterator[Tuple[str, Tensor]]:
        for name, off, n, shape in zip(self.names, self.offsets, self.numels, self.shapes):
            yield name, vec[off:off + n].view(shape)

    # ------------------------------------------------------- batched slab
    def batched_views(self, slab: Tensor) -> Dict[str, Tensor]:
        """Per-parameter [C, *shape] views into a [C, d] slab.

        The slab may have row stride > d (padded slabs for 16B-aligned HIP
        kernel access) as long as rows are innermost-contiguous."""
        assert slab.dim() == 2 and slab.shape[1] == self.d and slab.stride(1) == 1
        C = slab.shape[0]
        row_stride = slab.stride(0)
        out: Dict[str, Tensor] = {}
        for name, off, n, shape in zip(self.names, self.offsets, self.numels, self.shapes):
            stride = (row_stride,) + tuple(torch.empty(shape).stride())
            out[name] = slab.as_strided(size=(C, *shape), stride=stride,
                                        storage_offset=slab.storage_offset() + off)
        return out

    def views(self, vec: Tensor) -> Dict[str, Tensor]:
        """Per-parameter views into a flat [d] vector."""
        return {name: t for name, t in self.named_slices(vec)}
