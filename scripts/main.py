#!/usr/bin/env python3
"""Experiment driver (reference: scripts/main.py / scripts/cifar10.py).

Single GPU / CPU:
    python scripts/main.py --dataset synthetic --model mlp --agg median ...
8x MI355X (one rank per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 scripts/main.py --use-cuda ...
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from args import parse_arguments  # noqa: E402


def build_dataset(options):
    from blades_amd.datasets import (CIFAR10, CIFAR100, MNIST,
                                     SyntheticFLDataset)

    if options.dataset == "synthetic":
        shape = (1, 28, 28) if options.model == "mlp" else (3, 32, 32)
        return SyntheticFLDataset(
            num_clients=options.num_clients, batch_size=options.batch_size,
            shape=shape, num_classes=100 if options.dataset == "cifar100" else 10,
            seed=options.seed)
    cls = {"cifar10": CIFAR10, "cifar100": CIFAR100, "mnist": MNIST}[options.dataset]
    return cls(data_root=options.data_root, train_bs=options.batch_size,
               iid=options.iid, alpha=options.alpha,
               num_clients=options.num_clients, seed=options.seed)


def main():
    options = parse_arguments()
    os.makedirs(options.log_dir, exist_ok=True)

    from blades_amd import Simulator
    from blades_amd.models import get_model

    dataset = build_dataset(options)
    sim = Simulator(
        dataset=dataset,
        aggregator=options.agg,
        aggregator_kws=options.agg_args.get(options.agg, {}),
        num_byzantine=options.num_byzantine,
        attack=options.attack,
        attack_kws=options.attack_args.get(options.attack, {}),
        use_cuda=options.use_cuda,
        log_path=options.log_dir,
        seed=options.seed,
        engine=options.engine,
        hip_graphs=not options.no_hip_graphs,
    )
    model_kw = {}
    if options.model in ("resnet18", "wrn28_10"):
        model_kw["norm"] = "batch-local"
    if options.dataset == "cifar100":
        model_kw["num_classes"] = 100
    model = get_model(options.model, **model_kw)

    times = sim.run(
        model,
        global_rounds=options.global_round,
        local_steps=options.local_round,
        client_lr=options.lr,
        server_lr=options.server_lr,
        validate_interval=options.validate_interval,
        test_batch_size=options.test_batch_size,
    )
    if sim.runtime.is_main():
        import numpy as np
        print(f"finished {len(times)} rounds; "
              f"mean round time {np.mean(times):.4f}s; logs in {options.log_dir}")


if __name__ == "__main__":
    main()
