"""Simulator — public entry point and global-round orchestration.

API parity with the reference Simulator (reference: src/blades/
simulator.py:21-457): same constructor kwargs, same ``run()`` kwargs, same
string registries (aggregator name -> ``blades_amd.aggregators.<name>``
class ``<Name>``; attack name -> ``blades_amd.attackers.<name>client`` class
``<Name>Client``), same return value (list of per-round wall-clock seconds),
``register_attackers`` / ``set_trusted_clients`` / ``get_clients`` /
RNG cache-restore helpers.

MI355X-native execution (replaces Ray actors/trainers — SURVEY.md §5.8):

  per round:  every rank trains its client shard as ONE fused batched
  computation (engine.FusedEngine) writing a rank-local [K/ws, d] update
  slab in HBM -> RCCL all-gather of slabs over xGMI -> omniscient attacks
  computed rank-locally on the full slab (deterministic, identical on all
  ranks) -> robust aggregation (HIP kernels) -> fused flat server step.
  θ stays replicated; no per-round broadcast.

``mode``/``num_actors``/``num_trainers``/``gpu_per_actor`` are accepted for
compatibility and recorded, but the rank runtime subsumes them (the
reference's half-built "trainer" mode — simulator.py:249-280 — is covered
by the same engine).
"""
from __future__ import annotations

import importlib
import logging
import time
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import torch

from blades_amd.aggregators import _BaseAggregator, get_aggregator
from blades_amd.aggregators.fltrust import Fltrust
from blades_amd.attackers import get_attacker_cls
from blades_amd.client import BladesClient, ByzantineClient
from blades_amd.engine import FusedEngine, LoopEngine, ParamSpec, split_fusable
from blades_amd.parallel import DistributedRuntime
from blades_amd.server import BladesServer
from blades_amd.utils import (JsonStatsLogger, initialize_logger,
                              reset_model_weights, set_random_seed,
                              top1_accuracy, trace_range)


class _AttackContext:
    """Round-local view handed to omniscient callbacks (fast path).

    ``U`` is the FULL gathered update slab [K, d]; ``honest_mask`` marks
    non-Byzantine rows; ``cache`` shares per-round computations between
    attackers (e.g. all ALIE clients reuse one mean/std pass).
    """

    def __init__(self, U: torch.Tensor, honest_mask: torch.Tensor,
                 round_idx: int, base_seed: int, rows: Dict[Any, int]):
        self.U = U
        self.honest_mask = honest_mask
        self.round = round_idx
        self.base_seed = base_seed
        self._rows = rows
        self.cache: Dict[str, Any] = {}

    def row_of(self, client: BladesClient) -> int:
        return self._rows[client.id()]


class Simulator:
    """Synchronous parallel Byzantine-robust FL simulation on MI355X."""

    def __init__(
        self,
        dataset,
        num_byzantine: Optional[int] = 0,
        attack: Optional[str] = None,
        attack_kws: Optional[Dict[str, Any]] = None,
        aggregator: Union[Callable[[list], torch.Tensor], str] = "mean",
        aggregator_kws: Optional[Dict[str, Any]] = None,
        num_actors: Optional[int] = 1,
        num_trainers: Optional[int] = 1,
        gpu_per_actor: Optional[float] = 0,
        mode: Optional[str] = "actor",
        log_path: str = "./outputs",
        metrics: Optional[dict] = None,
        use_cuda: Optional[bool] = False,
        seed: Optional[int] = None,
        engine: str = "auto",
        client_chunk: Optional[int] = None,
        device: Optional[str] = None,
        hip_graphs: bool = True,
        gather: str = "auto",
        stream_clients: Optional[int] = None,
        **kwargs,
    ):
        self.use_actor = mode == "actor"
        self._engine_choice = engine
        self._client_chunk = client_chunk

        want_cuda = bool(use_cuda or (gpu_per_actor or 0) > 0
                         or (device is not None and str(device).startswith("cuda")))
        if want_cuda and not torch.cuda.is_available():
            raise RuntimeError("use_cuda requested but no GPU is visible")
        self.runtime = DistributedRuntime(
            device=device if device is not None else
            (None if want_cuda else torch.device("cpu")))
        self.device = self.runtime.device

        if aggregator_kws is None:
            aggregator_kws = {}
        self._init_aggregator(aggregator=aggregator, aggregator_kws=aggregator_kws)

        initialize_logger(log_path, rank=self.runtime.rank)
        self.metrics = {"top1": top1_accuracy} if metrics is None else metrics
        self.json_logger = JsonStatsLogger()
        self.debug_logger = logging.getLogger("debug")
        self.debug_logger.info(str(self))
        self.debug_logger.info(
            f"runtime: rank {self.runtime.rank}/{self.runtime.world_size} on "
            f"{self.device}; compat kwargs num_actors={num_actors} "
            f"num_trainers={num_trainers} gpu_per_actor={gpu_per_actor} "
            f"mode={mode} are subsumed by the rank runtime")

        self.random_states: Dict[str, Any] = {}
        self.omniscient_callbacks: List[Callable] = []
        self._seed = seed if seed is not None else 0
        self._attack_ctx: Optional[_AttackContext] = None

        if kwargs:
            unknown = ", ".join(kwargs)
            raise RuntimeError(f"Unknown keyword argument(s): {unknown}")

        # dataset handling (reference: simulator.py:100-102, fixing the
        # "FLDataset passed directly" crash noted in SURVEY.md §2.1)
        from blades_amd.datasets import FLDataset

        if isinstance(dataset, FLDataset) or hasattr(dataset, "get_train_data"):
            self.dataset = dataset
        else:
            traindls, testdls = dataset.get_dls()
            self.dataset = FLDataset(traindls, testdls)

        if attack_kws is None:
            attack_kws = {}
        self._setup_clients(attack, num_byzantine=num_byzantine,
                            attack_kws=attack_kws)

        set_random_seed(self._seed, use_cuda=self.device.type == "cuda")

        # populated by run()
        self.server: Optional[BladesServer] = None
        self._spec: Optional[ParamSpec] = None
        self._fused: Optional[FusedEngine] = None
        self._loop: Optional[LoopEngine] = None
        self.global_model: Optional[torch.nn.Module] = None
        # hipGraph round capture (engine/graphs.py); auto-disabled when the
        # population/aggregator/optimizer is not capturable
        import os as _os

        self._use_graphs = hip_graphs and _os.environ.get(
            "BLADES_AMD_NO_GRAPHS", "0") != "1"
        self._graph_round = None
        self._theta: Optional[torch.Tensor] = None
        self._model_stale = False  # True while θ (graph mode) is ahead of model
        # update-gather strategy: "full" all-gathers the whole U on every
        # rank; "coordinate" re-shards by coordinate (SP-style all-to-all,
        # SURVEY.md §5.7) — mandatory at 1e4-client × WRN scale where U is
        # 1.46 TB; "auto" picks coordinate when the aggregator/attacks
        # support it
        if gather not in ("auto", "full", "coordinate"):
            raise ValueError(f"gather must be auto|full|coordinate, got {gather}")
        self._gather = gather
        # streamed coordinate rounds: train + reshard the shard in chunks of
        # this many clients, so the rank-local client slab never
        # materializes (1e4-client × WRN scale: slab and coordinate shard
        # are each 182 GB/rank — together they exceed 288 GB; streamed, the
        # transient is stream_clients × d floats)
        self._stream_clients = stream_clients

    # ------------------------------------------------------------ builders
    def _init_aggregator(self, aggregator, aggregator_kws) -> None:
        if isinstance(aggregator, str):
            try:
                self.aggregator = get_aggregator(aggregator, **aggregator_kws)
            except KeyError:
                # reference-style importlib fallback for user modules
                agg_path = importlib.import_module(
                    f"blades_amd.aggregators.{aggregator}")
                agg_scheme = getattr(agg_path, aggregator.capitalize())
                self.aggregator = agg_scheme(**aggregator_kws)
        else:
            self.aggregator = aggregator

    def _setup_clients(self, attack: Optional[str], num_byzantine, attack_kws):
        if attack is None:
            num_byzantine = 0
        users = self.dataset.get_clients()
        if num_byzantine > len(users):
            raise ValueError(
                f"num_byzantine={num_byzantine} exceeds population {len(users)}")
        self._clients: Dict[Any, BladesClient] = {}
        for i, u in enumerate(users):
            if i < num_byzantine:
                try:
                    attack_scheme = get_attacker_cls(attack)
                except KeyError:
                    module_path = importlib.import_module(
                        f"blades_amd.attackers.{attack}client")
                    attack_scheme = getattr(module_path,
                                            f"{attack.capitalize()}Client")
                client = attack_scheme(id=u, device=self.device, **attack_kws)
                self._register_omniscient_callback(client.omniscient_callback)
            else:
                client = BladesClient(id=u, device=self.device)
            self._clients[u] = client

    def _register_omniscient_callback(self, callback) -> None:
        self.omniscient_callbacks.append(callback)

    # ---------------------------------------------------------- public API
    def get_clients(self) -> List[BladesClient]:
        return list(self._clients.values())

    def set_trusted_clients(self, ids: List) -> None:
        for id in ids:
            self._clients[id].trust()

    def register_attackers(self, clients: List[ByzantineClient],
                           replace_indices: Optional[List[int]] = None) -> None:
        """Replace clients with custom attackers (reference:
        simulator.py:167-187; the off-by-one asserts there are fixed)."""
        if replace_indices is not None:
            assert len(clients) == len(replace_indices)
        else:
            replace_indices = list(range(len(clients)))
        assert len(clients) <= len(self._clients)

        client_li = self.get_clients()
        for attacker, i in zip(clients, replace_indices):
            id = client_li[i].id()
            attacker.set_id(id)
            attacker.device = self.device
            self._clients[id] = attacker
            self._register_omniscient_callback(attacker.omniscient_callback)

    # --------------------------------------------------- RNG cache helpers
    def cache_random_state(self) -> None:
        if self.device.type == "cuda":
            self.random_states["torch_cuda"] = torch.cuda.get_rng_state()
        self.random_states["torch"] = torch.get_rng_state()
        self.random_states["numpy"] = np.random.get_state()

    def restore_random_state(self) -> None:
        if self.device.type == "cuda":
            torch.cuda.set_rng_state(self.random_states["torch_cuda"])
        torch.set_rng_state(self.random_states["torch"])
        np.random.set_state(self.random_states["numpy"])

    def parallel_call(self, clients, f: Callable[[BladesClient], None]) -> None:
        self.cache_random_state()
        for worker in clients:
            f(worker)
        self.restore_random_state()

    def parallel_get(self, clients, f: Callable[[BladesClient], Any]) -> list:
        results = []
        for w in clients:
            self.cache_random_state()
            results.append(f(w))
            self.restore_random_state()
        return results

    # -------------------------------------------------------- round engine
    def _ensure_client_model(self, client: BladesClient, lr: float) -> None:
        if client.model is None:
            client.device = self.device
            client.set_model(self.global_model, torch.optim.SGD, lr)

    def get_phase_times(self) -> Dict[str, float]:
        """Cumulative host-side seconds per round phase (local_train /
        gather / reshard / attack / aggregate / apply / eval / graph_round)
        — the rocprof-free first look at where a run spends time."""
        from blades_amd.utils.tracing import phase_seconds

        return dict(phase_seconds)

    def sync_model(self) -> None:
        """Write θ (graph-mode source of truth) back into the module params."""
        if self._model_stale:
            self.server.load_flat_parameters(self._theta)
            self._model_stale = False

    def _maybe_graph_round(self, global_round: int, local_steps: int,
                           lr: float) -> bool:
        if not self._use_graphs or self.device.type != "cuda":
            return False
        from blades_amd.engine.graphs import CapturedRound

        if self._graph_round is None:
            reason = CapturedRound.supported(self, self.get_clients(),
                                             local_steps)
            if reason is not None:
                self.debug_logger.info(f"hipGraph round capture off: {reason}")
                self._use_graphs = False
                return False
            self.server.flat_parameters(device=self.device, out=self._theta)
            self._graph_round = CapturedRound(self, self.get_clients(),
                                              local_steps)
        if self._graph_round.local_steps != local_steps:
            self.debug_logger.info("local_steps changed; dropping hipGraph")
            self._graph_round = None
            self._use_graphs = False
            return False
        server_lr = self.server_opt.param_groups[0]["lr"]
        from blades_amd.engine.graphs import CaptureFailed

        try:
            with trace_range("blades/graph_round"):
                self._graph_round.run(lr, server_lr)
        except CaptureFailed as e:
            # capture failed (e.g. a backend without stream-capture
            # support): symmetric across ranks for backend capability
            # errors — fall back to the eager round
            self.debug_logger.warning(
                f"hipGraph capture failed ({e}); falling back to eager")
            self._graph_round = None
            self._use_graphs = False
            torch.cuda.synchronize(self.device)
            return False
        self._model_stale = True
        return True

    def _coordinate_supported(self) -> Optional[str]:
        """Reason the coordinate-sharded gather can NOT be used, else None."""
        if not self.runtime.distributed:
            return "single rank (nothing to re-shard)"
        if not (getattr(self.aggregator, "coordinate_shardable", False)
                or getattr(self.aggregator, "supports_shard", False)):
            return (f"aggregator {type(self.aggregator).__name__} has no "
                    "coordinate-sharded form")
        from blades_amd.attackers import (AlieClient, IpmClient,
                                          LabelflippingClient, NoiseClient,
                                          SignflippingClient)
        from blades_amd.client import uses_default_training

        ok_types = (AlieClient, IpmClient, NoiseClient, LabelflippingClient,
                    SignflippingClient)
        for c in self.get_clients():
            if c.is_byzantine() and type(c) not in ok_types:
                return f"custom byzantine client {type(c).__name__}"
            if not uses_default_training(c):
                return f"custom client {type(c).__name__}"
        return None

    def _use_coordinate(self) -> bool:
        if self._gather == "full":
            return False
        reason = self._coordinate_supported()
        if reason is None:
            return True
        if self._gather == "coordinate":
            raise RuntimeError(f"gather='coordinate' not possible: {reason}")
        return False

    def train_round(self, global_round: int, local_steps: int,
                    clients: List[BladesClient], lr: float) -> None:
        """One global round (reference: train_actor, simulator.py:203-247)."""
        if self._maybe_graph_round(global_round, local_steps, lr):
            return
        self.sync_model()
        rt = self.runtime
        all_clients = self.get_clients()
        rows = {c.id(): i for i, c in enumerate(all_clients)}
        shard = rt.my_shard(all_clients)
        coordinate = self._use_coordinate()

        # slab rows padded to a float4 multiple: the HIP kernels take the
        # 16B-vectorized path on every row while views stay zero-copy.
        # Coordinate mode additionally pads to a world-size multiple so the
        # all-to-all splits evenly.
        d = self._spec.d
        if coordinate:
            per_rank = -(-d // rt.world_size)          # ceil(d / ws)
            dshard = -(-per_rank // 4) * 4             # round up to float4
            d_pad = dshard * rt.world_size
        else:
            dshard = 0
            d_pad = -(-d // 4) * 4

        if (coordinate and self._stream_clients is None
                and self._engine_choice != "loop"
                and self.device.type == "cuda"):
            # memory-aware default: stream when slab + coordinate shard
            # would not both fit comfortably (the reshard transiently holds
            # the client slab [K/ws, d] AND the coordinate shard [K, d/ws]
            # — each K·d/ws·4 bytes)
            try:
                free_b, _ = torch.cuda.mem_get_info(self.device)
                both = 2 * len(shard) * d_pad * 4
                if both > 0.5 * free_b:
                    auto = max(1, int((0.2 * free_b) / (d_pad * 4)))
                    self._stream_clients = auto
                    self.debug_logger.info(
                        f"auto stream_clients={auto} (slab+shard {both/1e9:.1f} "
                        f"GB vs {free_b/1e9:.1f} GB free)")
            except Exception:
                pass

        if (coordinate and self._stream_clients
                and self._engine_choice != "loop"):
            # streamed rounds drive the fused engine directly (the loop
            # engine would defeat the memory point of streaming)
            self._train_round_streamed(global_round, local_steps, lr,
                                       all_clients, shard, rows, dshard, d,
                                       d_pad)
            return

        with trace_range("blades/local_train"):
            theta = self.server.flat_parameters(device=self.device,
                                                out=self._theta)
            fusable, custom = split_fusable(shard)
            if self._engine_choice == "loop":
                fusable, custom = [], shard
            # single-rank / full-gather streaming: train the shard in
            # client chunks written straight into the slab, so transient
            # vmap gradient outputs are chunk-sized, not shard-sized
            # (config 5 at ws=1: the [1250, 36.5M] slab alone is 182 GB —
            # whole-shard grads would double it past 288 GB)
            stream = self._stream_clients
            # buffer reused across eager rounds (pads were zeroed once and
            # are never written; the engine overwrites [:, :d] fully)
            cache_key = (len(shard), d_pad)
            if getattr(self, "_buf_cache_key", None) != cache_key:
                self._buf_cache = torch.zeros(len(shard), d_pad,
                                              device=self.device)
                self._buf_cache_key = cache_key
            buf_local = self._buf_cache
            U_local = buf_local[:, :d]
            local_pos = {c.id(): i for i, c in enumerate(shard)}
            if len(fusable) == len(shard):
                if stream and stream < len(shard):
                    for lo in range(0, len(shard), stream):
                        part = shard[lo:lo + stream]
                        self._fused.run_round(theta, part, self.dataset,
                                              local_steps, lr,
                                              out=U_local[lo:lo + len(part)])
                else:
                    self._fused.run_round(theta, shard, self.dataset,
                                          local_steps, lr, out=U_local)
            else:
                if fusable:
                    Uf = self._fused.run_round(theta, fusable, self.dataset,
                                               local_steps, lr)
                    idx = torch.tensor([local_pos[c.id()] for c in fusable],
                                       device=self.device)
                    U_local.index_copy_(0, idx, Uf)
                if custom:
                    updates = self._loop.run_round(self.global_model, custom,
                                                   self.dataset, local_steps, lr)
                    for c in custom:
                        U_local[local_pos[c.id()]].copy_(
                            updates[c.id()].to(self.device))

        if coordinate:
            self._aggregate_coordinate(global_round, buf_local, U_local,
                                       all_clients, shard, rows, dshard, d)
            return

        with trace_range("blades/gather"):
            buf = rt.all_gather_rows(buf_local, total_rows=len(all_clients))
            U = buf[:, :d]
            torch.nan_to_num_(U)  # K18 sanitize (reference: client.py:198)

        # hand every client its row view (zero-copy)
        for c in all_clients:
            c.save_update_view(U[rows[c.id()]])

        with trace_range("blades/attack"):
            honest_mask = torch.tensor([not c.is_byzantine() for c in all_clients],
                                       device=self.device)
            self._attack_ctx = _AttackContext(U, honest_mask, global_round,
                                              self._seed, rows)
            for cb in self.omniscient_callbacks:
                cb(self)
            # write back rows replaced by attackers
            for c in all_clients:
                saved = c._get_saved_update()
                row = U[rows[c.id()]]
                if saved is not row:
                    row.copy_(saved.to(self.device))
                    c.save_update_view(row)
            self._attack_ctx = None
            # K18 again: the reference sanitizes at READ time, i.e. AFTER
            # omniscient attacks (client.py:198 via _get_updates) — crafted
            # NaN/Inf rows must not reach the aggregator
            torch.nan_to_num_(U)

        with trace_range("blades/aggregate"):
            agg = self.server.aggregator
            if isinstance(agg, Fltrust):
                aggregated = agg(all_clients)
            elif isinstance(agg, _BaseAggregator):
                aggregated = agg(U)
            else:  # user callable: reference passes the client list
                aggregated = agg(all_clients)

        with trace_range("blades/apply"):
            self.server.apply_update(aggregated.to(self.device))

    def _aggregate_coordinate(self, global_round: int, buf_local, U_local,
                              all_clients, shard, rows, dshard: int,
                              d: int) -> None:
        """SP-style aggregation: all-to-all to coordinate shards [K, d/ws],
        attacks + aggregation on the shard, all-gather of the Δ shards.

        The only full-row state that ever exists is the rank-local client
        shard (clients in my shard keep their update views; cross-rank rows
        are never materialized — this is what makes the 1e4-client × WRN
        configs fit in 288 GB/GPU, SURVEY.md §7 hard-part 2).
        """
        from blades_amd.attackers import AlieClient, IpmClient, NoiseClient
        from blades_amd.ops import philox_normal
        from blades_amd.utils import client_philox_seed

        rt = self.runtime
        local_pos = {c.id(): i for i, c in enumerate(shard)}

        # noise attackers need their FULL row -> craft pre-reshard on the
        # owner rank (deterministic per (client, round): layout-invariant)
        for c in shard:
            if type(c) is NoiseClient:
                seed = client_philox_seed(self._seed, rows[c.id()],
                                          global_round, tag=12)
                U_local[local_pos[c.id()]].copy_(philox_normal(
                    (d,), c._noise_mean, c._noise_std, seed,
                    device=self.device))
            c.save_update_view(U_local[local_pos[c.id()]])

        with trace_range("blades/reshard"):
            Ucoord = rt.all_to_all_coordinate_shard(
                buf_local, total_rows=len(all_clients), dshard=dshard)
            torch.nan_to_num_(Ucoord)
        self._finish_coordinate(global_round, Ucoord, all_clients, dshard, d)

    def _train_round_streamed(self, global_round: int, local_steps: int, lr,
                              all_clients, shard, rows, dshard: int, d: int,
                              d_pad: int) -> None:
        """Coordinate round in client chunks: train `stream_clients` of the
        shard, all-to-all that row block, scatter into the coordinate
        shard, repeat — the rank-local [K/ws, d] slab never materializes
        (SURVEY.md §7 hard-part 2; docs/roadmap.md item 3)."""
        from blades_amd.attackers import NoiseClient
        from blades_amd.ops import philox_normal
        from blades_amd.utils import client_philox_seed

        rt = self.runtime
        chunk = self._stream_clients
        sizes = [len(s) for s in rt.shard_indices(len(all_clients))]
        offsets = np.cumsum([0] + sizes[:-1])
        n_chunks = max(-(-s // chunk) for s in sizes)
        theta = self.server.flat_parameters(device=self.device,
                                            out=self._theta)

        Ucoord = torch.zeros(len(all_clients), dshard, device=self.device)
        for i in range(n_chunks):
            lo = i * chunk
            my_rows = shard[lo:lo + chunk]
            counts = [max(0, min(chunk, sizes[r] - lo))
                      for r in range(rt.world_size)]
            buf = torch.zeros(len(my_rows), d_pad, device=self.device)
            with trace_range("blades/local_train"):
                if my_rows:
                    self._fused.run_round(theta, my_rows, self.dataset,
                                          local_steps, lr,
                                          out=buf[:, :d])
                for j, c in enumerate(my_rows):
                    if type(c) is NoiseClient:
                        seed = client_philox_seed(self._seed, rows[c.id()],
                                                  global_round, tag=12)
                        buf[j, :d].copy_(philox_normal(
                            (d,), c._noise_mean, c._noise_std, seed,
                            device=self.device))
                    c.save_update_view(buf[j, :d])
            with trace_range("blades/reshard"):
                part = rt.all_to_all_row_block(buf, counts, dshard)
                # scatter the rank-ordered rows to their global positions
                gidx = torch.tensor(
                    [int(offsets[r]) + lo + j
                     for r in range(rt.world_size)
                     for j in range(counts[r])], device=self.device)
                Ucoord.index_copy_(0, gidx, part)
        torch.nan_to_num_(Ucoord)
        self._finish_coordinate(global_round, Ucoord, all_clients, dshard, d)

    def _finish_coordinate(self, global_round: int, Ucoord, all_clients,
                           dshard: int, d: int) -> None:
        from blades_amd.attackers import AlieClient, IpmClient

        rt = self.runtime
        with trace_range("blades/attack"):
            honest = torch.tensor([not c.is_byzantine() for c in all_clients],
                                  device=self.device)
            n_honest = int(honest.sum())
            from blades_amd import ops as _ops

            alie_groups: Dict[float, List[int]] = {}
            ipm_groups: Dict[float, List[int]] = {}
            for i, c in enumerate(all_clients):
                if type(c) is AlieClient:
                    alie_groups.setdefault(float(c.z_max), []).append(i)
                elif type(c) is IpmClient:
                    ipm_groups.setdefault(float(c.epsilon), []).append(i)
            for z, rws in alie_groups.items():
                mu, std = _ops.masked_col_mean_std(Ucoord, honest,
                                                   unbiased=True,
                                                   count=n_honest)
                idx = torch.tensor(rws, device=self.device)
                Ucoord.index_copy_(0, idx, (mu - std * z).unsqueeze(0)
                                   .expand(len(rws), -1))
            for eps, rws in ipm_groups.items():
                hm = _ops.masked_col_mean(Ucoord, honest, count=n_honest)
                idx = torch.tensor(rws, device=self.device)
                Ucoord.index_copy_(0, idx, (-eps * hm).unsqueeze(0)
                                   .expand(len(rws), -1))

        with trace_range("blades/aggregate"):
            if getattr(self.aggregator, "coordinate_shardable", False):
                delta_shard = self.aggregator(Ucoord)
            else:  # row-wise aggregator with a shard-aware form
                import inspect

                sig = inspect.signature(self.aggregator.aggregate_shard)
                if "clients" in sig.parameters:
                    delta_shard = self.aggregator.aggregate_shard(
                        Ucoord, rt, clients=all_clients)
                else:
                    delta_shard = self.aggregator.aggregate_shard(Ucoord, rt)

        with trace_range("blades/apply"):
            delta = rt.all_gather_flat(delta_shard)[:d]
            self.server.apply_update(delta)

    # reference-name aliases.  The reference's "trainer" mode
    # (simulator.py:249-280, half-built ray.train/DDP pool) is subsumed by
    # the same fused rank runtime — both names run the same round.
    def train_actor(self, global_round: int, num_rounds: int,
                    clients: List[BladesClient], lr: float) -> None:
        self.train_round(global_round, num_rounds, clients, lr)

    def train_trainer(self, epoch: int, num_rounds: int, clients) -> None:
        client_list = (list(clients.values()) if isinstance(clients, dict)
                       else list(clients))
        lr = getattr(self, "_last_client_lr", 0.1)
        self.train_round(epoch, num_rounds, client_list, lr)

    # --------------------------------------------------------------- eval
    def test_actor(self, global_round: int, batch_size: int):
        rt = self.runtime
        all_clients = self.get_clients()
        shard = rt.my_shard(all_clients)
        with trace_range("blades/eval"):
            if self._fused is not None and self._engine_choice != "loop":
                if self._model_stale:
                    theta = self._theta  # graph mode: θ is current
                else:
                    theta = self.server.flat_parameters(device=self.device,
                                                        out=self._theta)
                local_metrics = self._fused.evaluate(
                    theta, shard, self.dataset, global_round, batch_size,
                    self.metrics)
            else:
                self.sync_model()
                for c in shard:
                    self._ensure_client_model(c, 0.0)
                local_metrics = self._loop.evaluate(
                    self.global_model, shard, self.dataset, global_round,
                    batch_size, self.metrics)
            gathered = rt.all_gather_object(local_metrics)
            metrics = [m for part in gathered for m in part]
        # per-client validation records (reference: client.py:170 logged one
        # per client from inside evaluate; here rank 0 logs the gathered set)
        if rt.is_main():
            for m in metrics:
                self.json_logger.write(m)
        loss, top1 = self.log_validate(metrics)
        self.debug_logger.info(
            f"Test global round {global_round}, loss: {loss}, top1: {top1}")
        return loss, top1

    # ------------------------------------------------------------- logging
    def log_variance(self, cur_round: int, update) -> None:
        """Per-round update-variance diagnostics (reference:
        simulator.py:309-322)."""
        stacked = torch.vstack(list(update)) if not isinstance(update, torch.Tensor) else update
        var = torch.var(stacked, dim=0, unbiased=False)
        r = {
            "_meta": {"type": "variance"},
            "Round": cur_round,
            "avg": torch.mean(var).item(),
            "norm": torch.norm(var).item(),
            "avg_norm": torch.mean(var / torch.mean(stacked ** 2, dim=0)).item(),
        }
        self.json_logger.write(r)

    def log_train(self, progress, batch_idx, epoch, results) -> None:
        """Weighted training-metric record (reference: simulator.py:337-362
        — a dead path there due to two bugs; functional here).

        ``results``: list of {"length": n, "loss": l, "metrics": {...}}.
        """
        length = sum(res["length"] for res in results)
        r = {
            "_meta": {"type": "train"},
            "Round": epoch,
            "B": batch_idx,
            "Length": length,
            "Loss": sum(res["loss"] * res["length"] for res in results) / length,
        }
        for metric_name in self.metrics:
            r[metric_name] = (sum(res["metrics"][metric_name] * res["length"]
                                  for res in results) / length)
        self.debug_logger.info(
            f"[Round {epoch} B{batch_idx}] Loss: {r['Loss']:.4f} "
            + " ".join(f"{name}={r[name]:>8.4f}" for name in self.metrics))
        self.json_logger.write(r)

    def log_validate(self, metrics):
        top1 = np.average([m["top1"] for m in metrics],
                          weights=[m["Length"] for m in metrics])
        loss = np.average([m["Loss"] for m in metrics],
                          weights=[m["Length"] for m in metrics])
        r = {
            "_meta": {"type": "test"},
            "Round": metrics[0]["E"],
            "top1": float(top1),
            "Length": int(np.sum([m["Length"] for m in metrics])),
            "Loss": float(loss),
        }
        self.json_logger.write(r)
        return loss, top1

    # ----------------------------------------------------------------- run
    def run(
        self,
        model: torch.nn.Module,
        server_optimizer: Union[torch.optim.Optimizer, str] = "SGD",
        client_optimizer: Union[torch.optim.Optimizer, str] = "SGD",
        loss: Optional[str] = "crossentropy",
        global_rounds: Optional[int] = 1,
        local_steps: Optional[int] = 1,
        validate_interval: Optional[int] = 1,
        test_batch_size: Optional[int] = 64,
        server_lr: Optional[float] = 0.1,
        client_lr: Optional[float] = 0.1,
        server_lr_scheduler=None,
        client_lr_scheduler=None,
    ) -> List[float]:
        """Run the adversarial training; returns per-round wall-clock seconds
        (the reference's measurement hook, simulator.py:453-457)."""
        reset_model_weights(model)
        model = model.to(self.device)
        self.global_model = model
        self._spec = ParamSpec.from_module(model)
        self._theta = torch.empty(self._spec.d, device=self.device)
        self._graph_round = None
        self._model_stale = False

        if server_optimizer == "SGD":
            self.server_opt = torch.optim.SGD(model.parameters(), lr=server_lr)
        else:
            self.server_opt = server_optimizer
        self.client_opt = client_optimizer
        self.server = BladesServer(optimizer=self.server_opt, model=model,
                                   aggregator=self.aggregator)

        # replicate initial weights across ranks (the ONLY broadcast needed;
        # afterwards every rank applies the same aggregate deterministically)
        if self.runtime.distributed:
            theta0 = self.server.flat_parameters(device=self.device)
            self.runtime.broadcast_flat(theta0)
            self.server.load_flat_parameters(theta0)

        # engines
        self._loop = LoopEngine(device=self.device)
        self._fused = FusedEngine(model, self._spec, self.device,
                                  client_chunk=self._client_chunk)

        clients = self.get_clients()
        self.parallel_call(clients, lambda c: c.set_loss(loss))
        # custom clients (overridden hooks) need a materialized model;
        # fusable ones train out of the slab (no per-client deep copy)
        _, custom = split_fusable(clients)
        if self._engine_choice == "loop":
            custom = clients
        for c in custom:
            self._ensure_client_model(c, client_lr)

        global_start = time.time()
        ret: List[float] = []
        cur_lr = client_lr
        self._last_client_lr = client_lr
        for r in range(1, global_rounds + 1):
            round_start = time.time()
            self.train_round(r, local_steps, clients, cur_lr)
            if validate_interval and r % validate_interval == 0:
                l, t1 = self.test_actor(global_round=r,
                                        batch_size=test_batch_size)
            if server_lr_scheduler:
                server_lr_scheduler.step()
            if client_lr_scheduler:
                client_lr_scheduler.step()
                cur_lr = client_lr_scheduler.get_last_lr()[0]
                self._last_client_lr = cur_lr
            ret.append(time.time() - round_start)
            self.debug_logger.info(
                f"E={r}; Client learning rate = {cur_lr}; "
                f"Time cost = {time.time() - global_start}")
        self.sync_model()
        return ret

    def __str__(self) -> str:
        return (f"Simulator(aggregator={self.aggregator}, "
                f"clients={len(getattr(self, '_clients', {}))})")
