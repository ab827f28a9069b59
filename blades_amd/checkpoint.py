"""Checkpoint / resume.

New capability (the reference never persists training state — SURVEY.md
§5.4): a checkpoint is the flat fp32 parameter vector (in the same
named-parameters flattening order updates use), server optimizer state,
round counter, RNG states, and any stateful aggregator's state
(Centeredclipping momentum, Clippedclustering norm history).

Layout is a single ``torch.save`` file; the dataset pickle cache
(datasets/basedataset.py) keeps its reference-compatible format separately.
"""
from __future__ import annotations

import random
from typing import Any, Dict, Optional

import numpy as np
import torch

from blades_amd.engine.flat import ParamSpec

FORMAT_VERSION = 1


def capture_rng_states(device: Optional[torch.device] = None) -> Dict[str, Any]:
    states = {
        "torch": torch.get_rng_state(),
        "numpy": np.random.get_state(),
        "python": random.getstate(),
    }
    if device is not None and device.type == "cuda":
        states["torch_cuda"] = torch.cuda.get_rng_state(device)
    return states


def restore_rng_states(states: Dict[str, Any],
                       device: Optional[torch.device] = None) -> None:
    torch.set_rng_state(states["torch"])
    np.random.set_state(states["numpy"])
    random.setstate(states["python"])
    if "torch_cuda" in states and device is not None and device.type == "cuda":
        torch.cuda.set_rng_state(states["torch_cuda"], device)


def save_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    round_idx: int = 0,
                    aggregator=None,
                    extra: Optional[Dict[str, Any]] = None,
                    device: Optional[torch.device] = None) -> None:
    spec = ParamSpec.from_module(model)
    ckpt = {
        "format_version": FORMAT_VERSION,
        "round": round_idx,
        "flat_params": spec.flatten(model).cpu(),
        "param_names": spec.names,
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "aggregator_state": aggregator.state_dict()
        if aggregator is not None and hasattr(aggregator, "state_dict") else {},
        "rng": capture_rng_states(device),
        "extra": extra or {},
    }
    torch.save(ckpt, path)


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    aggregator=None,
                    restore_rng: bool = True,
                    device: Optional[torch.device] = None) -> Dict[str, Any]:
    # weights_only load: checkpoints contain only tensors, primitives and
    # the numpy/python RNG-state tuples — no arbitrary pickle execution.
    with torch.serialization.safe_globals(
            [np.ndarray, np.dtype, np.dtypes.UInt32DType,
             np._core.multiarray._reconstruct]):
        ckpt = torch.load(path, map_location="cpu", weights_only=True)
    if ckpt.get("format_version") != FORMAT_VERSION:
        raise ValueError(f"unsupported checkpoint version {ckpt.get('format_version')}")
    spec = ParamSpec.from_module(model)
    if ckpt["param_names"] != spec.names:
        raise ValueError("checkpoint parameter layout does not match model")
    spec.load(model, ckpt["flat_params"].to(next(model.parameters()).device))
    if optimizer is not None and ckpt["optimizer"] is not None:
        optimizer.load_state_dict(ckpt["optimizer"])
    if aggregator is not None and hasattr(aggregator, "load_state_dict"):
        aggregator.load_state_dict(ckpt.get("aggregator_state", {}))
    if restore_rng and "rng" in ckpt:
        restore_rng_states(ckpt["rng"], device)
    return ckpt
