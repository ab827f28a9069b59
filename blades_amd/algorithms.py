"""FL algorithm presets.

The reference shipped empty stubs (algorithms/fedavg.py,
algorithms/bladestrainer.py — 1 line each) and realized the
algorithms implicitly through ``local_steps``: FedSGD = 1 local step with
server_lr applying the pseudo-gradient, FedAvg = multiple local steps with
the parameter delta averaged.  These helpers make that explicit.
"""
from __future__ import annotations

from typing import Any, Dict


def fedsgd(client_lr: float = 0.1, server_lr: float = 1.0) -> Dict[str, Any]:
    """run() kwargs for FedSGD: one local step; the aggregated delta is the
    (negated, lr-scaled) gradient applied at server_lr."""
    return {"local_steps": 1, "client_lr": client_lr, "server_lr": server_lr}


def fedavg(local_steps: int = 5, client_lr: float = 0.1,
           server_lr: float = 1.0) -> Dict[str, Any]:
    """run() kwargs for FedAvg-style multi-step local training; server_lr=1
    with Mean aggregation reproduces plain federated averaging."""
    return {"local_steps": local_steps, "client_lr": client_lr,
            "server_lr": server_lr}
