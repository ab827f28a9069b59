"""hipGraph-captured global round.

The fused round is launch-bound on MI355X: the vmapped per-client
forward/backward decomposes into ~1e5 small kernel dispatches per round
(measured: ~2 µs/dispatch average, profiles/r01_vmap_resnet18_kernel_stats.csv),
so launch gaps, not kernels, set the round time.  The whole round body —
local training on static batch buffers, update-slab write, built-in
omniscient attacks, robust aggregation, and the flat server step — is
GPU-static for the benchmark configs, so it is captured ONCE into a
hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm) and replayed each
round; per round the host only refills the static data buffers and replays.

Capture conditions (checked by :meth:`CapturedRound.supported`):
  * single rank (multi-GPU rounds keep the eager path in round 1),
  * every client fusable, every Byzantine client an exact built-in type
    whose attack semantics this module replicates in-graph
    (ALIE / IPM / label-flip / sign-flip),
  * a capturable aggregator (Mean / Median / Trimmedmean / Krum — fixed
    iteration structure, no data-dependent host branching),
  * plain-SGD server (flat θ ← θ + lr·Δ apply).

Anything else silently keeps the eager path — same numerics, fewer
assumptions.  Learning rates live in device scalars so schedulers work
without re-capture.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from blades_amd import ops
from blades_amd.aggregators import Krum, Mean, Median, Trimmedmean
from blades_amd.attackers import (AlieClient, IpmClient, LabelflippingClient,
                                  SignflippingClient)
from blades_amd.client import BladesClient

_CAPTURABLE_AGGS = (Mean, Median, Trimmedmean, Krum)
_CAPTURABLE_BYZ = (AlieClient, IpmClient, LabelflippingClient,
                   SignflippingClient)


class CapturedRound:
    WARMUP_ROUNDS = 2  # eager body runs (MIOpen find, allocator settling)

    # ------------------------------------------------------------ gating
    @staticmethod
    def supported(sim, clients: List[BladesClient], local_steps: int) -> Optional[str]:
        """Return a reason string when capture is NOT possible, else None."""
        if sim.device.type != "cuda":
            return "not on GPU"
        if sim.runtime.distributed:
            return "multi-rank round (eager path in this version)"
        from blades_amd.client import uses_default_training

        for c in clients:
            if c.is_byzantine() and type(c) not in _CAPTURABLE_BYZ:
                return f"non-built-in byzantine client {type(c).__name__}"
            if not uses_default_training(c):
                return f"custom client {type(c).__name__}"
        if not isinstance(sim.aggregator, _CAPTURABLE_AGGS):
            return f"aggregator {type(sim.aggregator).__name__} not capturable"
        if isinstance(sim.aggregator, Trimmedmean):
            K = len(clients)
            b = sim.aggregator.b
            if K - 2 * b <= 0 or 2 * b * 4 * 64 > 64 * 1024:
                return "trimmedmean b out of kernel range"
        if not sim.server._plain_sgd():
            return "server optimizer is not plain SGD"
        return None

    # ------------------------------------------------------------- build
    def __init__(self, sim, clients: List[BladesClient], local_steps: int):
        self.sim = sim
        self.clients = clients
        self.local_steps = local_steps
        self.device = sim.device
        self.spec = sim._spec
        self.engine = sim._fused

        C = len(clients)
        d = self.spec.d
        d_pad = -(-d // 4) * 4
        self._buf = torch.zeros(C, d_pad, device=self.device)
        self.U = self._buf[:, :d]

        # static data buffers (shapes from one probe fetch)
        probe = sim.dataset.get_stacked_train_data(
            [c.id() for c in clients], local_steps, device=self.device)
        self.Xs = [x.clone() for x, _ in probe]
        self.Ys = [y.clone() for _, y in probe]
        self._probe = probe  # first round's data, already fetched

        # device-scalar learning rates (scheduler-safe)
        self.client_lr = torch.zeros((), device=self.device)
        self.server_lr = torch.zeros((), device=self.device)

        # static attack plan
        self.honest_mask = torch.tensor(
            [not c.is_byzantine() for c in clients], device=self.device)
        self.n_honest = int(self.honest_mask.sum().item())
        alie_groups = {}
        ipm_groups = {}
        for i, c in enumerate(clients):
            if type(c) is AlieClient:
                alie_groups.setdefault(float(c.z_max), []).append(i)
            elif type(c) is IpmClient:
                ipm_groups.setdefault(float(c.epsilon), []).append(i)
        self.alie_groups = [
            (z, torch.tensor(rows, device=self.device))
            for z, rows in alie_groups.items()]
        self.ipm_groups = [
            (eps, torch.tensor(rows, device=self.device))
            for eps, rows in ipm_groups.items()]

        # wire client API views once: get_update() returns slab rows forever
        for i, c in enumerate(clients):
            c.save_update_view(self.U[i])

        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self._eager_runs = 0

    # --------------------------------------------------------------- body
    def _body(self) -> None:
        sim = self.sim
        self.engine.run_round(sim._theta, self.clients, None,
                              self.local_steps, self.client_lr,
                              out=self.U, data=list(zip(self.Xs, self.Ys)))
        torch.nan_to_num_(self.U)
        for z, rows in self.alie_groups:
            mu, std = ops.masked_col_mean_std(self.U, self.honest_mask,
                                              unbiased=True,
                                              count=self.n_honest)
            self.U.index_copy_(0, rows,
                               (mu - std * z).unsqueeze(0).expand(len(rows), -1))
        for eps, rows in self.ipm_groups:
            hm = ops.masked_col_mean(self.U, self.honest_mask,
                                     count=self.n_honest)
            self.U.index_copy_(0, rows,
                               (-eps * hm).unsqueeze(0).expand(len(rows), -1))
        delta = sim.aggregator(self.U)
        sim._theta.add_(delta * self.server_lr)

    def _fill(self) -> None:
        if self._probe is not None:
            data = self._probe
            self._probe = None
        else:
            data = self.sim.dataset.get_stacked_train_data(
                [c.id() for c in self.clients], self.local_steps,
                device=self.device)
        for s, (x, y) in enumerate(data):
            self.Xs[s].copy_(x)
            self.Ys[s].copy_(y)

    # ---------------------------------------------------------------- run
    def run(self, client_lr: float, server_lr: float) -> None:
        self.client_lr.fill_(client_lr)
        self.server_lr.fill_(server_lr)
        self._fill()
        if self.graph is not None:
            self.graph.replay()
            return
        if self._eager_runs < self.WARMUP_ROUNDS:
            self._body()
            self._eager_runs += 1
            return
        # capture on a side stream (torch.cuda.graph manages pool + stream)
        torch.cuda.synchronize(self.device)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._body()
        self.graph = g
        # the capture pass itself did not execute; replay for this round
        g.replay()
