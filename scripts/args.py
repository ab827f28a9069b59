"""Experiment CLI — flag surface and log-dir naming compatible with the
reference (reference: scripts/args.py:7-63), extended with MI355X runtime
knobs (engine, hip-graphs, distributed handled by torchrun env).

The reference's ``gpu_per_actor = (num_gpus - 0.05)/num_actors`` fractional
packing is replaced by one-process-per-GPU client sharding; the flags are
still accepted so existing sweep scripts parse unchanged.
"""
from __future__ import annotations

import argparse
import os

import torch


def parse_arguments(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--use-cuda", action="store_true", default=False)
    parser.add_argument("--use_actor", action="store_true", default=False)
    parser.add_argument("--seed", type=int, default=1)
    parser.add_argument("--global_round", type=int, default=400)
    parser.add_argument("--local_round", type=int, default=50)
    parser.add_argument("--batch_size", type=int, default=32)
    parser.add_argument("--test_batch_size", type=int, default=128)
    parser.add_argument("--log_interval", type=int, default=10)
    parser.add_argument("--metrics_name", type=str, default="none")
    parser.add_argument("--attack", type=str, default="signflipping")
    parser.add_argument("--dataset", type=str, default="cifar10",
                        help="cifar10 | cifar100 | mnist | synthetic")
    parser.add_argument("--agg", type=str, default="clippedclustering")
    parser.add_argument("--lr", type=float, default=0.1)
    parser.add_argument("--server_lr", type=float, default=1.0)
    parser.add_argument("--num_actors", type=int, default=20,
                        help="accepted for compat; subsumed by rank runtime")
    parser.add_argument("--num_byzantine", type=int, default=8)
    parser.add_argument("--num_clients", type=int, default=20)
    parser.add_argument("--num_gpus", type=int, default=4)
    parser.add_argument("--model", type=str, default="cct",
                        help="mlp | cct | resnet18 | wrn28_10")
    parser.add_argument("--engine", type=str, default="auto",
                        help="auto | loop")
    parser.add_argument("--no_hip_graphs", action="store_true", default=False)
    parser.add_argument("--iid", action="store_true", default=True)
    parser.add_argument("--non_iid", dest="iid", action="store_false")
    parser.add_argument("--alpha", type=float, default=0.1,
                        help="Dirichlet alpha for non-iid partition")
    parser.add_argument("--validate_interval", type=int, default=1)
    parser.add_argument("--data_root", type=str, default="./data")
    options = parser.parse_args(argv)

    ROOT_DIR = os.path.dirname(os.path.abspath(__file__))
    EXP_DIR = os.path.join(ROOT_DIR, f"outputs/{options.dataset}")

    options.attack_args = {
        "signflipping": {},
        "labelflipping": {},
        "noise": {},
        "ipm": {"epsilon": 0.5},
        "alie": {"num_clients": options.num_clients,
                 "num_byzantine": options.num_byzantine},
    }
    options.agg_args = {
        "mean": {}, "median": {}, "geomed": {}, "autogm": {},
        "clustering": {}, "clippedclustering": {}, "centeredclipping": {},
        "fltrust": {},
        "trimmedmean": {"nb": options.num_byzantine},
        "krum": {"num_clients": options.num_clients,
                 "num_byzantine": options.num_byzantine},
        "multikrum": {"num_clients": options.num_clients,
                      "num_byzantine": options.num_byzantine},
    }

    # reference log-dir naming scheme (scripts/args.py:47-56)
    attack_kws = options.attack_args.get(options.attack, {})
    agg_kws = options.agg_args.get(options.agg, {})
    options.log_dir = (
        EXP_DIR
        + f"/b{options.num_byzantine}"
        + f"_{options.attack}"
        + ("_" + "_".join(k + str(v) for k, v in attack_kws.items())
           if attack_kws else "")
        + f"_{options.agg}"
        + ("_" + "_".join(k + str(v) for k, v in agg_kws.items())
           if agg_kws else "")
        + f"_lr{options.lr}"
        + f"_bz{options.batch_size}"
        + f"_seed{options.seed}"
    )

    options.use_cuda = torch.cuda.is_available() and (
        options.use_cuda or options.num_gpus > 0)
    if not torch.cuda.is_available():
        options.num_gpus = 0
    return options
