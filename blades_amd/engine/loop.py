"""Per-client loop engine.

Reference-exact execution (the Ray actor's client loop, reference:
actor.py:23-48, minus Ray): set_para → set_lr → on_train_round_begin →
local_training → on_train_round_end → get_update, sequentially per client.
Used for (a) user clients that override training hooks, (b) CPU runs, and
(c) as the semantics oracle the fused engine is tested against.
"""
from __future__ import annotations

from typing import Dict, List

import torch

from blades_amd.client import BladesClient


class LoopEngine:
    def __init__(self, device: str = "cpu"):
        self.device = device

    def run_round(self, global_model: torch.nn.Module, clients: List[BladesClient],
                  dataset, local_steps: int, lr: float) -> Dict[str, torch.Tensor]:
        """Train every client locally; returns {client_id: flat update}."""
        updates: Dict[str, torch.Tensor] = {}
        for client in clients:
            client.set_para(global_model)
            client.set_lr(lr)
            client.on_train_round_begin()
            data = dataset.get_train_data(client.id(), local_steps)
            client.local_training(data)
            client.on_train_round_end()
            updates[client.id()] = client.get_update()
        return updates

    def evaluate(self, global_model: torch.nn.Module, clients: List[BladesClient],
                 dataset, round_number: int, batch_size: int, metrics) -> List[dict]:
        results = []
        for client in clients:
            client.set_para(global_model)
            r = client.evaluate(
                round_number=round_number,
                test_set=dataset.get_all_test_data(client.id()),
                batch_size=batch_size,
                metrics=metrics,
            )
            r["Client"] = client.id()
            results.append(r)
        return results
