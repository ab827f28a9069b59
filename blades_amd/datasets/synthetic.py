"""Synthetic device-resident federated dataset.

Benchmark configs run on synthetic data + random-init weights
(BASELINE.json; no network for real datasets).  MI355X-first design: the
whole per-client sample pool is generated once, stays resident in HBM, and
the hot loop only takes slab views — zero host↔device traffic and zero RNG
inside the timed region.

Per-client pools are keyed by (seed, client_id) counter-based streams so
the data a client sees is identical regardless of rank sharding.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch

from blades_amd.utils import client_philox_seed
from .customdataset import CustomTensorDataset


class SyntheticFLDataset:
    def __init__(self, num_clients: int = 100, samples_per_client: int = 64,
                 batch_size: int = 32, shape: Sequence[int] = (3, 32, 32),
                 num_classes: int = 10, seed: int = 0, device: str = "cpu",
                 test_samples_per_client: int = 32, learnable: bool = False):
        self.num_clients = num_clients
        self.samples_per_client = samples_per_client
        self.batch_size = batch_size
        self.shape = tuple(shape)
        self.num_classes = num_classes
        self.seed = seed
        self.device = torch.device(device)
        self.test_samples_per_client = test_samples_per_client
        # learnable=True labels samples with a fixed random linear teacher
        # (same teacher for train and test), so eval loss can actually drop;
        # learnable="templates" draws X = template[y] + 0.5*noise from
        # num_classes fixed random templates — quickly learnable at conv
        # scale (the robustness-curve demonstration)
        self.learnable = learnable
        self._teacher = None
        self._templates = None
        self._clients = list(range(num_clients))
        self._cursor = [0] * num_clients
        # lazy pools: client -> (X [S,*shape], y [S])
        self._pool: dict = {}
        self._test: dict = {}

    # ------------------------------------------------------------ plumbing
    def get_clients(self) -> List[int]:
        return self._clients

    def _gen_pool(self, u_id: int, n: int, tag: int):
        # ALWAYS generate on CPU: CUDA and CPU philox streams differ, and a
        # client's data must be identical regardless of the device (or rank)
        # that hosts it.  One-time cost outside any timed region; pools then
        # live on self.device.
        g = torch.Generator()
        g.manual_seed(client_philox_seed(self.seed, int(u_id), 0, tag=tag))
        X = torch.randn((n, *self.shape), generator=g)
        if self.learnable == "templates":
            if self._templates is None:
                gt = torch.Generator()
                gt.manual_seed(client_philox_seed(self.seed, 0, 0, tag=98))
                self._templates = torch.randn((self.num_classes, *self.shape),
                                              generator=gt)
            y = torch.randint(0, self.num_classes, (n,), generator=g)
            X = self._templates[y] + 0.5 * X
        elif self.learnable:
            if self._teacher is None:
                gt = torch.Generator()
                gt.manual_seed(client_philox_seed(self.seed, 0, 0, tag=99))
                dim = int(torch.tensor(self.shape).prod())
                self._teacher = torch.randn((dim, self.num_classes),
                                            generator=gt)
            y = (X.flatten(1) @ self._teacher).argmax(dim=1)
        else:
            y = torch.randint(0, self.num_classes, (n,), generator=g)
        return X.to(self.device), y.to(self.device)

    def _train_pool(self, u_id: int):
        if u_id not in self._pool:
            self._pool[u_id] = self._gen_pool(u_id, self.samples_per_client, tag=1)
        return self._pool[u_id]

    def materialize(self, client_ids: Optional[Sequence[int]] = None) -> None:
        """Pre-generate pools for the given clients (all by default) so no
        allocation happens inside a timed region."""
        for u in (client_ids if client_ids is not None else self._clients):
            self._train_pool(u)

    # ------------------------------------------------------------ train API
    def _next_batch_idx(self, u_id: int) -> torch.Tensor:
        S = self.samples_per_client
        B = self.batch_size
        start = self._cursor[u_id]
        self._cursor[u_id] = (start + B) % S
        idx = torch.arange(start, start + B, device=self.device) % S
        return idx

    def get_train_data(self, u_id, num_batches: int):
        X, y = self._train_pool(u_id)
        out = []
        for _ in range(num_batches):
            idx = self._next_batch_idx(u_id)
            out.append((X[idx], y[idx]))
        return out

    def get_stacked_train_data(self, ids: Sequence[int], steps: int,
                               device=None) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        """[C, B, ...] stacked batches per local step, device-resident."""
        dev = torch.device(device) if device is not None else self.device
        out = []
        for _ in range(steps):
            xs, ys = [], []
            for u in ids:
                X, y = self._train_pool(u)
                idx = self._next_batch_idx(u)
                xs.append(X[idx])
                ys.append(y[idx])
            out.append((torch.stack(xs).to(dev), torch.stack(ys).to(dev)))
        return out

    # ------------------------------------------------------------- test API
    def get_all_test_data(self, u_id):
        if u_id not in self._test:
            X, y = self._gen_pool(u_id, self.test_samples_per_client, tag=2)
            self._test[u_id] = CustomTensorDataset(X.cpu(), y.cpu())
        return self._test[u_id]

    def get_stacked_test_data(self, ids: Sequence[int], device=None):
        """[C, n_test, ...] stacked test tensors (batched-eval fast path);
        identical data to get_all_test_data (same CPU-generated pools)."""
        dev = torch.device(device) if device is not None else self.device
        xs, ys = [], []
        for u in ids:
            X, y = self._gen_pool(u, self.test_samples_per_client, tag=2)
            xs.append(X)
            ys.append(y)
        return torch.stack(xs).to(dev), torch.stack(ys).to(dev)
