"""Aggregator sweep under a fixed attack + stats-file parsing.

MI355X equivalent of reference examples/"Simulation on MNIST.py": run
several robust aggregators against ALIE and read the accuracy curves back
from the JSON stats logs (one JSON object per line — the reference's
``replace("'", '"')`` consumer pattern also parses these).
"""
import json
import os

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from blades_amd import Simulator
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import MLP

num_clients, num_byzantine = 10, 3
aggs = ["mean", "median", "trimmedmean", "krum", "geomed"]
agg_kws = {
    "trimmedmean": {"nb": num_byzantine},
    "krum": {"num_clients": num_clients, "num_byzantine": num_byzantine},
}

curves = {}
for agg in aggs:
    log_path = f"./outputs/sweep/{agg}"
    ds = SyntheticFLDataset(num_clients=num_clients, samples_per_client=64,
                            batch_size=32, shape=(1, 28, 28), seed=0,
                            learnable=True)
    sim = Simulator(dataset=ds, aggregator=agg,
                    aggregator_kws=agg_kws.get(agg, {}),
                    num_byzantine=num_byzantine, attack="alie",
                    attack_kws={"num_clients": num_clients,
                                "num_byzantine": num_byzantine},
                    use_cuda=torch.cuda.is_available(), seed=1,
                    log_path=log_path)
    sim.run(MLP(), global_rounds=15, local_steps=2, server_lr=1.0,
            client_lr=0.2, validate_interval=5)

    # parse the stats file back (validation records)
    acc = []
    with open(os.path.join(log_path, "stats")) as f:
        for line in f:
            rec = json.loads(line.replace("'", '"'))
            if rec["_meta"]["type"] == "test":
                acc.append((rec["Round"], rec["top1"]))
    curves[agg] = acc
    print(f"{agg:>14}: " + "  ".join(f"r{r}={a:.1f}%" for r, a in acc))
