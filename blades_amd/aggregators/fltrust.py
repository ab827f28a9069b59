"""FLTrust (reference: aggregators/fltrust.py:8-38).

Cao et al., "FLTrust: Byzantine-robust Federated Learning via Trust
Bootstrapping".  Requires exactly one trusted client (set via
``Simulator.set_trusted_clients``).  Trust score = ReLU(cos(u_trusted, u_k));
each untrusted update is renormed to the trusted norm; output is the
trust-weighted mean.  HIP kernel K9: one fused row-dot/row-norm pass + one
weighted column sum.
"""
from __future__ import annotations

import torch

from blades_amd import ops
from .base import _BaseAggregator


class Fltrust(_BaseAggregator):
    def __call__(self, clients):
        from blades_amd.client import BladesClient

        if not (isinstance(clients, (list, tuple))
                and all(isinstance(c, BladesClient) for c in clients)):
            raise TypeError("Fltrust requires a list of BladesClient "
                            "(it needs trust flags)")
        trusted = [c for c in clients if c.is_trusted()]
        assert len(trusted) == 1, "Fltrust needs exactly one trusted client"
        untrusted = [c for c in clients if not c.is_trusted()]

        t = trusted[0].get_update()
        U = torch.stack([c.get_update() for c in untrusted])
        t_norm = t.norm()
        dots = ops.row_dots(U, t)
        norms = ops.row_norms(U).clamp_min(1e-6)
        cos = dots / (norms * t_norm.clamp_min(1e-6))
        ts = torch.relu(cos)
        # renorm each row to the trusted norm, weight by trust score
        w = ts * (t_norm / norms)
        return ops.weighted_col_sum(U, w) / ts.sum().clamp_min(1e-12)

    supports_shard = True

    def aggregate_shard(self, U_shard: torch.Tensor, runtime, clients=None):
        """Distributed FLTrust: per-row dots with the trusted row and row
        norms come from shard partials + one all-reduce of the 2K+1 small
        statistics; the trust-weighted mean stays shard-local."""
        from blades_amd import ops

        assert clients is not None, "FLTrust shard form needs the client list"
        trusted_rows = [i for i, c in enumerate(clients) if c.is_trusted()]
        assert len(trusted_rows) == 1, "Fltrust needs exactly one trusted client"
        t_row = trusted_rows[0]
        untrusted = [i for i in range(len(clients)) if i != t_row]

        t = U_shard[t_row]
        idx = torch.tensor(untrusted, device=U_shard.device)
        Uu = U_shard[idx]
        # pack [dots | row sq-norms | trusted sq-norm] into one all-reduce
        stats = torch.cat([
            ops.row_dots(Uu, t),
            ops.row_sq_norms(Uu),
            (t * t).sum().reshape(1),
        ])
        runtime.all_reduce_(stats)
        n = len(untrusted)
        dots = stats[:n]
        norms = stats[n:2 * n].sqrt().clamp_min(1e-6)
        t_norm = stats[-1].sqrt()
        cos = dots / (norms * t_norm.clamp_min(1e-6))
        ts = torch.relu(cos)
        w = ts * (t_norm / norms)
        return ops.weighted_col_sum(Uu, w) / ts.sum().clamp_min(1e-12)

    def __str__(self):
        return "FLTrust"
