"""IPM — inner-product manipulation attack
(reference: attackers/ipmclient.py:4-16).

Xie et al. 2020.  Crafted update = −ε · mean(honest updates); HIP kernel K11
(masked column mean) on the slab path.
"""
from __future__ import annotations

from blades_amd import ops
from blades_amd.client import ByzantineClient


class IpmClient(ByzantineClient):
    def __init__(self, epsilon: float = 0.5, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.epsilon = epsilon

    def omniscient_callback(self, simulator) -> None:
        ctx = getattr(simulator, "_attack_ctx", None)
        if ctx is not None:
            key = "honest_mean"
            if key not in ctx.cache:
                ctx.cache[key] = ops.masked_col_mean(ctx.U, ctx.honest_mask)
            self._state["saved_update"] = -self.epsilon * ctx.cache[key]
            return
        updates = [c.get_update() for c in simulator.get_clients()
                   if not c.is_byzantine()]
        self._state["saved_update"] = -self.epsilon * sum(updates) / len(updates)
