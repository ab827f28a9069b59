"""hipGraph-captured global round — single-rank AND multi-rank.

The fused round is launch-bound on MI355X: the vmapped per-client
forward/backward decomposes into ~1e5 small kernel dispatches per round
(measured: ~2 µs/dispatch average, profiles/r01_vmap_resnet18_kernel_stats.csv),
so launch gaps, not kernels, set the round time.  The whole round body —
local training on static batch buffers, update-slab write, the RCCL
gather collective, built-in omniscient attacks, robust aggregation, and
the flat server step — is GPU-static for the benchmark configs, so it is
captured ONCE into a hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm)
and replayed each round; per round the host only refills the static data
buffers and replays.

Multi-rank capture (round-2): RCCL supports stream
capture, and torch.distributed's NCCL backend issues collectives on the
capture stream, so the all-gather (full mode) or all-to-all + Δ-shard
all-gather (coordinate mode) are captured INSIDE the graph.  The warmup
eager rounds establish the communicator before capture.  Collective
buffers are static; uneven client shards go through a captured
index_select with a precomputed row map.

Capture conditions (checked by :meth:`CapturedRound.supported`):
  * every client fusable, every Byzantine client an exact built-in type
    whose attack semantics this module replicates in-graph
    (ALIE / IPM / label-flip / sign-flip),
  * a capturable aggregator — fixed iteration structure, no
    data-dependent host branching: Mean / Median / Trimmedmean / Krum in
    full-gather mode; their coordinate/shard forms in coordinate mode,
  * plain-SGD server (flat θ ← θ + lr·Δ apply),
  * no streamed coordinate rounds (streaming exists to avoid exactly the
    static slab a graph would pin).

Anything else silently keeps the eager path — same numerics, fewer
assumptions.  Learning rates live in device scalars so schedulers work
without re-capture.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from blades_amd import ops
from blades_amd.aggregators import Krum, Mean, Median, Trimmedmean
from blades_amd.attackers import (AlieClient, IpmClient, LabelflippingClient,
                                  SignflippingClient)
from blades_amd.client import BladesClient

_CAPTURABLE_AGGS = (Mean, Median, Trimmedmean, Krum)
_CAPTURABLE_BYZ = (AlieClient, IpmClient, LabelflippingClient,
                   SignflippingClient)


class CaptureFailed(RuntimeError):
    """Raised when hipGraph capture itself fails (e.g. a collective that
    does not support stream capture); the caller falls back to eager."""


class CapturedRound:
    WARMUP_ROUNDS = 2  # eager body runs (MIOpen find, allocator settling)

    # ------------------------------------------------------------ gating
    @staticmethod
    def supported(sim, clients: List[BladesClient], local_steps: int) -> Optional[str]:
        """Return a reason string when capture is NOT possible, else None."""
        if sim.device.type != "cuda":
            return "not on GPU"
        from blades_amd.client import uses_default_training

        for c in clients:
            if c.is_byzantine() and type(c) not in _CAPTURABLE_BYZ:
                return f"non-built-in byzantine client {type(c).__name__}"
            if not uses_default_training(c):
                return f"custom client {type(c).__name__}"
        if not isinstance(sim.aggregator, _CAPTURABLE_AGGS):
            return f"aggregator {type(sim.aggregator).__name__} not capturable"
        if isinstance(sim.aggregator, Trimmedmean):
            if len(clients) - 2 * sim.aggregator.b <= 0:
                return "trimmedmean b out of range"
        if not sim.server._plain_sgd():
            return "server optimizer is not plain SGD"
        if sim._stream_clients:
            return ("streamed rounds are not captured (a whole-shard "
                    "capture would defeat the memory point of streaming)")
        if sim.runtime.distributed:
            if dist.get_backend() != "nccl":
                return "multi-rank capture needs the RCCL backend"
            if sim._use_coordinate() and not getattr(
                    sim.aggregator, "coordinate_shardable", False) \
                    and not isinstance(sim.aggregator, Krum):
                return (f"aggregator {type(sim.aggregator).__name__} has no "
                        "capturable shard form")
        return None

    # ------------------------------------------------------------- build
    def __init__(self, sim, clients: List[BladesClient], local_steps: int):
        self.sim = sim
        self.clients = clients
        self.local_steps = local_steps
        self.device = sim.device
        self.spec = sim._spec
        self.engine = sim._fused
        rt = sim.runtime

        K = len(clients)
        d = self.spec.d
        self.mode = "single"
        if rt.distributed:
            self.mode = "coordinate" if sim._use_coordinate() else "full"

        self.shard = rt.my_shard(clients) if rt.distributed else clients
        Cl = len(self.shard)

        if self.mode == "coordinate":
            per_rank = -(-d // rt.world_size)
            self.dshard = -(-per_rank // 4) * 4
            d_pad = self.dshard * rt.world_size
        else:
            self.dshard = 0
            d_pad = -(-d // 4) * 4
        self.d_pad = d_pad

        # local slab this rank trains into (rows zero-padded to kmax in the
        # distributed modes so the collectives see equal splits)
        sizes = ([len(s) for s in rt.shard_indices(K)]
                 if rt.distributed else [K])
        kmax = max(sizes)
        rows_alloc = kmax if rt.distributed else K
        self._buf = torch.zeros(rows_alloc, d_pad, device=self.device)
        self.U_local = self._buf[:Cl, :d]

        if self.mode == "single":
            self.U = self._buf[:, :d]
        elif self.mode == "full":
            self._gath = torch.zeros(rt.world_size * kmax, d_pad,
                                     device=self.device)
            if K == rt.world_size * kmax:
                self._gidx = None
                self.U = self._gath[:, :d]
            else:  # uneven shards: captured index_select back to [K, d]
                gidx = [r * kmax + j for r, n in enumerate(sizes)
                        for j in range(n)]
                self._gidx = torch.tensor(gidx, device=self.device)
                self._Ufull = torch.zeros(K, d_pad, device=self.device)
                self.U = self._Ufull[:, :d]
        else:  # coordinate
            # all_to_all send view is built in-body from _buf; recv is static
            self._a2a_recv = torch.zeros(rt.world_size, kmax, self.dshard,
                                         device=self.device)
            if K == rt.world_size * kmax:
                self._gidx = None
                self.Ucoord = self._a2a_recv.view(rt.world_size * kmax,
                                                  self.dshard)
            else:
                gidx = [r * kmax + j for r, n in enumerate(sizes)
                        for j in range(n)]
                self._gidx = torch.tensor(gidx, device=self.device)
                self._Uc = torch.zeros(K, self.dshard, device=self.device)
                self.Ucoord = self._Uc
            self._delta_full = torch.zeros(d_pad, device=self.device)
        self._kmax = kmax
        self._sizes = sizes

        # static data buffers (shapes from one probe fetch) — SHARD clients
        probe = sim.dataset.get_stacked_train_data(
            [c.id() for c in self.shard], local_steps, device=self.device)
        self.Xs = [x.clone() for x, _ in probe]
        self.Ys = [y.clone() for _, y in probe]
        self._probe = probe  # first round's data, already fetched

        # device-scalar learning rates (scheduler-safe)
        self.client_lr = torch.zeros((), device=self.device)
        self.server_lr = torch.zeros((), device=self.device)

        # static attack plan over the GLOBAL population
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
        if self.mode == "coordinate":
            for i, c in enumerate(self.shard):
                c.save_update_view(self.U_local[i])
        else:
            U = self.U
            for i, c in enumerate(clients):
                c.save_update_view(U[i])

        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self._eager_runs = 0

    # --------------------------------------------------------------- body
    def _attack_rows(self, U: torch.Tensor) -> None:
        """ALIE/IPM omniscient rewrites, identical on every rank (K10/K11).
        ``U`` is the full [K, d] slab or the [K, dshard] coordinate shard —
        the attacks are coordinate-wise, so the same code serves both."""
        for z, rows in self.alie_groups:
            mu, std = ops.masked_col_mean_std(U, self.honest_mask,
                                              unbiased=True,
                                              count=self.n_honest)
            U.index_copy_(0, rows,
                          (mu - std * z).unsqueeze(0).expand(len(rows), -1))
        for eps, rows in self.ipm_groups:
            hm = ops.masked_col_mean(U, self.honest_mask,
                                     count=self.n_honest)
            U.index_copy_(0, rows,
                          (-eps * hm).unsqueeze(0).expand(len(rows), -1))

    def _body(self) -> None:
        sim = self.sim
        rt = sim.runtime
        self.engine.run_round(sim._theta, self.shard, None,
                              self.local_steps, self.client_lr,
                              out=self.U_local,
                              data=list(zip(self.Xs, self.Ys)))
        torch.nan_to_num_(self.U_local)

        if self.mode == "single":
            self._attack_rows(self.U)
            delta = sim.aggregator(self.U)
            sim._theta.add_(delta * self.server_lr)
            return

        if self.mode == "full":
            dist.all_gather_into_tensor(self._gath, self._buf)
            if self._gidx is not None:
                torch.index_select(self._gath, 0, self._gidx,
                                   out=self._Ufull)
            self._attack_rows(self.U)
            delta = sim.aggregator(self.U)
            sim._theta.add_(delta * self.server_lr)
            return

        # coordinate: [kmax, ws, dshard] -> all_to_all -> [ws, kmax, dshard]
        send = self._buf.view(self._kmax, rt.world_size, self.dshard) \
            .transpose(0, 1).contiguous()
        dist.all_to_all_single(self._a2a_recv, send)
        if self._gidx is not None:
            torch.index_select(
                self._a2a_recv.view(-1, self.dshard), 0, self._gidx,
                out=self._Uc)
        self._attack_rows(self.Ucoord)
        agg = sim.aggregator
        if getattr(agg, "coordinate_shardable", False):
            delta_shard = agg(self.Ucoord)
        else:
            delta_shard = agg.aggregate_shard(self.Ucoord, rt)
        dist.all_gather_into_tensor(self._delta_full,
                                    delta_shard.contiguous())
        sim._theta.add_(self._delta_full[:self.spec.d] * self.server_lr)

    def _fill(self) -> None:
        if self._probe is not None:
            data = self._probe
            self._probe = None
        else:
            data = self.sim.dataset.get_stacked_train_data(
                [c.id() for c in self.shard], self.local_steps,
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
        # capture on a side stream (torch.cuda.graph manages pool + stream);
        # ranks must enter capture together — the barrier keeps RCCL's
        # capture windows aligned across the communicator
        if self.sim.runtime.distributed:
            dist.barrier()
        torch.cuda.synchronize(self.device)
        g = torch.cuda.CUDAGraph()
        try:
            with torch.cuda.graph(g):
                self._body()
        except Exception as e:  # capture-time failure -> eager fallback
            raise CaptureFailed(str(e)) from e
        self.graph = g
        # the capture pass itself did not execute; replay for this round
        g.replay()
