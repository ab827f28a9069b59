#!/usr/bin/env python3
"""Flagship benchmark — BASELINE.json headline metric.

Metric: global rounds/sec on CIFAR10-shaped synthetic data, ResNet-18
FedSGD, 100 clients per GPU (weak scaling) with 20% ALIE attackers and
TrimmedMean aggregation (BASELINE.json config 2 at N=1; the same per-GPU
work at N=2/4/8).

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``;
for N>1 launched under torch.distributed.run with one rank per GPU.
W untimed warmup rounds, then EXACTLY K timed rounds bracketed by
barrier + torch.cuda.synchronize on both sides; MAX time over ranks;
rank 0 prints one JSON line.

Compute dtype is fp32 — the reference trains in plain fp32 PyTorch, and
gfx950 has no TF32; using bf16 would be below-reference precision and
invalid per the bench rules.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch


#: BASELINE.json named configs (per-GPU client counts; weak scaling)
CONFIG_PRESETS = {
    # 1: MNIST MLP FedSGD, 10 clients / 2 noise, Mean (CPU plumbing config)
    "1": dict(model="mlp", clients_per_gpu=10, byz_frac=0.2, attack="noise",
              aggregator="mean", local_steps=1, num_classes=10),
    # 2: CIFAR10 ResNet-18 FedSGD, 100 clients / 20 ALIE, TrimmedMean (headline)
    "2": dict(model="resnet18", clients_per_gpu=100, byz_frac=0.2,
              attack="alie", aggregator="trimmedmean", local_steps=1,
              num_classes=10),
    # 3: CIFAR10 ResNet-18 FedAvg (5 local steps), 1000 clients, Median
    "3": dict(model="resnet18", clients_per_gpu=125, byz_frac=0.0,
              attack=None, aggregator="median", local_steps=5,
              num_classes=10),
    # 4: CIFAR10 ResNet-18, 100 clients / 20 IPM, Krum
    "4": dict(model="resnet18", clients_per_gpu=100, byz_frac=0.2,
              attack="ipm", aggregator="krum", local_steps=1,
              num_classes=10),
    # 5: CIFAR100 WRN-28-10 FedSGD, labelflipping, GeoMed (sharded scale)
    "5": dict(model="wrn28_10", clients_per_gpu=32, byz_frac=0.1,
              attack="labelflipping", aggregator="geomed", local_steps=1,
              num_classes=100),
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--config", type=str, default=None,
                   help="BASELINE.json config preset 1-5 (overridable)")
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--clients-per-gpu", type=int, default=100)
    p.add_argument("--byz-frac", type=float, default=0.2)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--local-steps", type=int, default=1)
    p.add_argument("--model", type=str, default="resnet18")
    p.add_argument("--aggregator", type=str, default="trimmedmean")
    p.add_argument("--attack", type=str, default="alie")
    p.add_argument("--client-chunk", type=int, default=None)
    p.add_argument("--stream-clients", type=int, default=None,
                   help="streamed coordinate rounds: clients per reshard chunk")
    p.add_argument("--num-classes", type=int, default=10)
    p.add_argument("--samples-per-client", type=int, default=64)
    args, _ = p.parse_known_args()
    if args.config is not None:
        preset = CONFIG_PRESETS[args.config]
        defaults = {a.dest: a.default for a in p._actions}
        for k, v in preset.items():
            # presets fill only values the user left at their defaults
            if getattr(args, k) == defaults.get(k):
                setattr(args, k, v)
    return args


def main():
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world_size)
    rank = int(os.environ.get("RANK", "0"))

    use_cuda = torch.cuda.is_available()
    if not use_cuda:
        # CPU smoke mode (scaled down so it finishes in seconds)
        args.clients_per_gpu = min(args.clients_per_gpu, 8)
        args.model = "mlp"

    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import get_model

    total_clients = args.clients_per_gpu * n_gpus
    num_byz = int(round(args.byz_frac * total_clients))
    if args.attack in (None, "none", "None"):
        args.attack = None
        num_byz = 0

    shape = (1, 28, 28) if args.model == "mlp" else (3, 32, 32)
    model_kw = {"num_classes": args.num_classes}
    if args.model in ("resnet18", "wrn28_10"):
        model_kw["norm"] = "batch-local"

    agg_kws = {}
    if args.aggregator == "trimmedmean":
        agg_kws = {"nb": max(1, num_byz)}
    elif args.aggregator in ("krum", "multikrum"):
        agg_kws = {"num_clients": total_clients, "num_byzantine": num_byz}
    attack_kws = {}
    if args.attack == "alie":
        attack_kws = {"num_clients": total_clients, "num_byzantine": num_byz}
    elif args.attack == "labelflipping":
        attack_kws = {"num_classes": args.num_classes}

    # 1-GPU rehearsal of multi-rank launches: force every rank onto one
    # device (e.g. BLADES_BENCH_DEVICE=cuda:0 with BLADES_AMD_BACKEND=gloo)
    device_override = os.environ.get("BLADES_BENCH_DEVICE")

    sim = Simulator(
        device=device_override,
        dataset=SyntheticFLDataset(
            num_clients=total_clients,
            samples_per_client=args.samples_per_client,
            batch_size=args.batch, shape=shape,
            num_classes=args.num_classes, seed=0, device="cpu"),
        num_byzantine=num_byz,
        attack=args.attack,
        attack_kws=attack_kws,
        aggregator=args.aggregator,
        aggregator_kws=agg_kws,
        use_cuda=use_cuda,
        log_path=os.environ.get("BLADES_BENCH_LOG", "/tmp/blades_bench_logs"),
        seed=1234,
        client_chunk=args.client_chunk,
        stream_clients=args.stream_clients,
    )
    device = sim.device
    # move the synthetic pools to the device and pre-materialize the shard
    sim.dataset.device = device
    shard_ids = [c.id() for c in sim.runtime.my_shard(sim.get_clients())]
    sim.dataset.materialize(shard_ids)

    model = get_model(args.model, **model_kw)
    # build engines/server without running any round
    sim.run(model, global_rounds=0, local_steps=args.local_steps,
            validate_interval=0, client_lr=0.1, server_lr=1.0)

    clients = sim.get_clients()

    def one_round(r):
        sim.train_round(r, args.local_steps, clients, 0.1)

    rt = sim.runtime
    for r in range(args.warmup):
        one_round(r + 1)

    rt.barrier()
    if use_cuda:
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    for r in range(args.steps):
        one_round(args.warmup + r + 1)
    rt.barrier()
    if use_cuda:
        torch.cuda.synchronize(device)
    elapsed = time.perf_counter() - t0

    elapsed = rt.max_over_ranks(elapsed)
    rounds_per_sec = args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    from blades_amd.utils.tracing import phase_seconds
    phases = {k.replace("blades/", ""): round(v, 3)
              for k, v in sorted(phase_seconds.items())}

    mem = {}
    if use_cuda:
        free_b, total_b = torch.cuda.mem_get_info(device)
        mem = {"max_allocated_gb": round(
                   torch.cuda.max_memory_allocated(device) / 1e9, 2),
               "reserved_gb": round(
                   torch.cuda.memory_reserved(device) / 1e9, 2),
               "device_total_gb": round(total_b / 1e9, 2),
               "device_used_gb": round((total_b - free_b) / 1e9, 2)}

    if rank == 0:
        print(json.dumps({
            "metric": f"global rounds/sec ({args.model} FedSGD, "
                      f"{args.clients_per_gpu} clients/GPU, "
                      f"{args.attack}+{args.aggregator})",
            "value": rounds_per_sec,
            "unit": "rounds/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * args.clients_per_gpu * n_gpus,
                "clients": args.clients_per_gpu * n_gpus,
                "byzantine": num_byz,
                "attack": args.attack,
                "aggregator": args.aggregator,
                "local_steps": args.local_steps,
                "seq_len": None,
                "parallelism": f"client-sharded dp{n_gpus}",
            },
            "phase_seconds_total": phases,
            "memory": mem,
        }))


if __name__ == "__main__":
    main()
