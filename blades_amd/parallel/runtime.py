"""Distributed runtime: one process per MI355X, RCCL over xGMI.

Replaces the reference's Ray actor pool + object store (reference:
simulator.py:90-98, actor.py:6-48).  Mapping:

* model broadcast (pickled nn.Module per round)  -> one ncclBroadcast of the
  flat θ vector at run start; afterwards every rank applies the same
  deterministic aggregate, so θ stays replicated with ZERO per-round
  broadcast traffic;
* update gather (Ray futures of CPU tensors)     -> ncclAllGather of the
  rank-local [K/ws, d] update slab (device-resident, no host copies).
  All-gather (not gather-to-root) so omniscient attacks and aggregation run
  rank-local with full knowledge — no second broadcast;
* eval metric gather                             -> all_gather_object of the
  small per-client metric dicts.

Initialization follows torchrun env (RANK/WORLD_SIZE/LOCAL_RANK,
MASTER_ADDR/PORT); backend "nccl" IS RCCL on ROCm, "gloo" drives the same
code paths on CPU CI.  world_size == 1 short-circuits every collective.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional, Sequence

import numpy as np
import torch
import torch.distributed as dist


class DistributedRuntime:
    def __init__(self, device: Optional[torch.device] = None,
                 timeout_s: float = 600.0):
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))

        if self.world_size > 1 and not dist.is_initialized():
            # BLADES_AMD_BACKEND=gloo forces gloo with GPU compute — the
            # hardware rehearsal mode for multi-process rounds on a 1-GPU
            # lease (collectives staged through host memory)
            backend = os.environ.get(
                "BLADES_AMD_BACKEND",
                "nccl" if torch.cuda.is_available() else "gloo")
            dist.init_process_group(
                backend=backend,
                timeout=datetime.timedelta(seconds=timeout_s),
            )

        if device is not None:
            self.device = torch.device(device)
        elif torch.cuda.is_available():
            self.device = torch.device(f"cuda:{self.local_rank}")
            torch.cuda.set_device(self.device)
        else:
            self.device = torch.device("cpu")

    # ------------------------------------------------------------- helpers
    @property
    def distributed(self) -> bool:
        return self.world_size > 1

    def _stage_cpu(self, t: torch.Tensor) -> bool:
        """True when a collective on ``t`` must be staged through host
        memory: the gloo backend cannot move CUDA tensors.  This makes
        multi-process rounds with GPU compute runnable under gloo — the
        hardware rehearsal mode for the RCCL path on a 1-GPU lease (RCCL
        refuses two ranks on one device: 'Duplicate GPU detected')."""
        return (self.distributed and t.is_cuda
                and dist.get_backend() == "gloo")

    def is_main(self) -> bool:
        return self.rank == 0

    def barrier(self) -> None:
        if self.distributed:
            dist.barrier()

    def shard_indices(self, n: int) -> List[np.ndarray]:
        """Contiguous split of range(n) across ranks (np.array_split
        semantics, matching the reference's client→actor split,
        simulator.py:223)."""
        return np.array_split(np.arange(n), self.world_size)

    def my_shard(self, items: Sequence) -> List:
        idx = self.shard_indices(len(items))[self.rank]
        return [items[i] for i in idx]

    # --------------------------------------------- staged primitive wrappers
    def _all_gather_into(self, out: torch.Tensor, send: torch.Tensor) -> None:
        if self._stage_cpu(send):
            host_out = torch.empty(out.shape, dtype=out.dtype)
            dist.all_gather_into_tensor(host_out, send.cpu())
            out.copy_(host_out)
        else:
            dist.all_gather_into_tensor(out, send)

    def _all_to_all(self, recv: torch.Tensor, send: torch.Tensor) -> None:
        if self._stage_cpu(send):
            host_recv = torch.empty(recv.shape, dtype=recv.dtype)
            dist.all_to_all_single(host_recv, send.cpu())
            recv.copy_(host_recv)
        else:
            dist.all_to_all_single(recv, send)

    # ---------------------------------------------------------- collectives
    def broadcast_flat(self, vec: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.distributed:
            if self._stage_cpu(vec):
                host = vec.cpu()
                dist.broadcast(host, src=src)
                vec.copy_(host)
            else:
                dist.broadcast(vec, src=src)
        return vec

    def all_gather_rows(self, local: torch.Tensor, total_rows: int) -> torch.Tensor:
        """Gather per-rank row blocks into the full [total_rows, d] matrix.

        Rank r contributes the rows of shard r (contiguous).  Shards may be
        uneven; blocks are padded to the max shard size for the collective
        (NCCL all_gather needs equal shapes) and sliced after.
        """
        if not self.distributed:
            return local
        d = local.shape[1]
        sizes = [len(s) for s in self.shard_indices(total_rows)]
        kmax = max(sizes)
        if local.shape[0] < kmax:
            pad = torch.zeros(kmax - local.shape[0], d, device=local.device,
                              dtype=local.dtype)
            send = torch.cat([local, pad], dim=0).contiguous()
        else:
            send = local.contiguous()
        out = torch.empty(self.world_size * kmax, d, device=local.device,
                          dtype=local.dtype)
        self._all_gather_into(out, send)
        blocks = [out[r * kmax: r * kmax + sizes[r]] for r in range(self.world_size)]
        return torch.cat(blocks, dim=0)

    def all_to_all_row_block(self, local: torch.Tensor, counts,
                             dshard: int) -> torch.Tensor:
        """Like :meth:`all_to_all_coordinate_shard` but for an explicit
        per-rank row-count list (streamed client chunks): rank r
        contributes ``counts[r]`` rows; returns [sum(counts), dshard] in
        rank order."""
        if not self.distributed:
            return local
        ws = self.world_size
        kmax = max(counts)
        if local.shape[0] < kmax:
            pad = torch.zeros(kmax - local.shape[0], local.shape[1],
                              device=local.device, dtype=local.dtype)
            send = torch.cat([local, pad], dim=0)
        else:
            send = local
        send = send.view(kmax, ws, dshard).transpose(0, 1).contiguous()
        recv = torch.empty_like(send)
        self._all_to_all(recv, send)
        blocks = [recv[r, :counts[r]] for r in range(ws)]
        return torch.cat(blocks, dim=0)

    def all_to_all_coordinate_shard(self, local: torch.Tensor,
                                    total_rows: int,
                                    dshard: int) -> torch.Tensor:
        """Client-sharded [K/ws, ws*dshard] -> coordinate-sharded
        [K, dshard] (this rank's coordinate slice of EVERY client's update).

        The SP-style re-shard of SURVEY.md §5.7: one all-to-all moves
        (K/ws)·d·(ws-1)/ws bytes per rank — ws× less traffic than a full
        all-gather — and aggregation work splits d/ws per rank.  Rows are
        padded to the max shard size (all_to_all_single needs equal splits)
        and re-ordered back to global client order.
        """
        if not self.distributed:
            return local
        sizes = [len(s) for s in self.shard_indices(total_rows)]
        kmax = max(sizes)
        ws = self.world_size
        if local.shape[0] < kmax:
            pad = torch.zeros(kmax - local.shape[0], local.shape[1],
                              device=local.device, dtype=local.dtype)
            send = torch.cat([local, pad], dim=0)
        else:
            send = local
        # [kmax, ws, dshard] -> [ws, kmax, dshard] so split s goes to rank s
        send = send.view(kmax, ws, dshard).transpose(0, 1).contiguous()
        recv = torch.empty_like(send)
        self._all_to_all(recv, send)
        # recv[r] = rank r's rows (padded), MY coordinate slice
        blocks = [recv[r, :sizes[r]] for r in range(ws)]
        return torch.cat(blocks, dim=0)  # [K, dshard], global client order

    def all_gather_flat(self, shard: torch.Tensor) -> torch.Tensor:
        """Gather equal-size 1-D shards into one vector (rank order)."""
        if not self.distributed:
            return shard
        out = torch.empty(self.world_size * shard.numel(),
                          device=shard.device, dtype=shard.dtype)
        self._all_gather_into(out, shard.contiguous())
        return out

    def all_reduce_(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if self.distributed:
            red = dist.ReduceOp.SUM if op == "sum" else dist.ReduceOp.MAX
            if self._stage_cpu(t):
                host = t.cpu()
                dist.all_reduce(host, op=red)
                t.copy_(host)
            else:
                dist.all_reduce(t, op=red)
        return t

    def all_gather_object(self, obj) -> List:
        if not self.distributed:
            return [obj]
        out = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def max_over_ranks(self, value: float) -> float:
        if not self.distributed:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        # gloo handles CPU tensors; nccl needs device tensors
        if dist.get_backend() == "nccl":
            t = t.to(self.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())
