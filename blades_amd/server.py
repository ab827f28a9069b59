"""Server layer (reference: src/blades/server.py:6-75).

Holds the global model + optimizer + aggregator.  ``apply_update`` keeps the
reference's pseudo-gradient semantics — p.grad = −update_slice, then
optimizer.step() — with a fused flat fast path: when the optimizer is plain
SGD (no momentum / weight-decay / dampening), θ ← θ + lr·Δ is applied
directly on the flat parameter vector (HIP op K16; on GPU this is one fused
axpy instead of a per-parameter scatter + optimizer pass).
"""
from __future__ import annotations

from typing import Callable

import torch

from blades_amd.engine.flat import ParamSpec


class BladesServer:
    def __init__(self, optimizer: torch.optim.Optimizer, model: torch.nn.Module,
                 aggregator: Callable[[list], torch.Tensor], *args, **kwargs):
        self.optimizer = optimizer
        self.model = model
        self.aggregator = aggregator
        self._spec = ParamSpec.from_module(model)

    def get_opt(self) -> torch.optim.Optimizer:
        return self.optimizer

    def get_model(self) -> torch.nn.Module:
        return self.model

    def zero_grad(self, set_to_none: bool = False) -> None:
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def _plain_sgd(self) -> bool:
        if type(self.optimizer) is not torch.optim.SGD:
            return False
        for g in self.optimizer.param_groups:
            if g.get("momentum", 0) or g.get("weight_decay", 0) or g.get("dampening", 0) or g.get("nesterov", False):
                return False
        return True

    @torch.no_grad()
    def apply_update(self, update: torch.Tensor) -> None:
        """Apply the aggregated flat update as a pseudo-gradient."""
        update = update.detach()
        if self._plain_sgd() and len(self.optimizer.param_groups) == 1:
            lr = self.optimizer.param_groups[0]["lr"]
            # fused path: θ += lr·Δ written back through per-param views
            for p, sl in zip(self._spec.iter_params(self.model),
                             self._spec.slices(update)):
                p.data.add_(sl.view_as(p).to(p.device), alpha=lr)
            return
        # general path — reference semantics (server.py:54-75)
        self.zero_grad()
        beg = 0
        for group in self.optimizer.param_groups:
            for p in group["params"]:
                if not p.requires_grad:
                    continue
                end = beg + p.data.numel()
                x = update[beg:end].reshape_as(p.data)
                p.grad = -x.clone().detach().to(p.device)
                beg = end
        self.optimizer.step()

    # ------- flat-vector access used by the distributed runtime/checkpoint
    def flat_parameters(self, device=None, out=None) -> torch.Tensor:
        return self._spec.flatten(self.model, device=device, out=out)

    def load_flat_parameters(self, vec: torch.Tensor) -> None:
        self._spec.load(self.model, vec)
