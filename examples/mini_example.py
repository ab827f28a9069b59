"""Minimal quickstart — ALIE attackers vs Mean aggregation.

MI355X equivalent of the reference README quickstart
(reference: src/blades/examples/mini_example.py, README.rst:64-70).
Runs on CPU or GPU; uses teacher-labeled synthetic MNIST-shaped data so it
works without downloads (swap in ``blades_amd.datasets.MNIST`` when the raw
idx files are available under ./data).
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from blades_amd import Simulator
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import MLP

num_clients = 10
num_byzantine = 3

dataset = SyntheticFLDataset(num_clients=num_clients, samples_per_client=64,
                             batch_size=32, shape=(1, 28, 28), num_classes=10,
                             seed=0, learnable=True)

simulator = Simulator(
    dataset=dataset,
    aggregator="mean",              # built-in aggregation scheme by name
    num_byzantine=num_byzantine,    # number of Byzantine clients
    attack="alie",                  # attack strategy
    attack_kws={"num_clients": num_clients, "num_byzantine": num_byzantine},
    use_cuda=torch.cuda.is_available(),
    seed=1,
    log_path="./outputs/mini_example",
)

model = MLP()
round_times = simulator.run(
    model,
    server_optimizer="SGD",
    client_optimizer="SGD",
    loss="crossentropy",
    global_rounds=20,
    local_steps=2,
    server_lr=1.0,
    client_lr=0.1,
    validate_interval=5,
)
print(f"ran {len(round_times)} rounds, "
      f"mean {sum(round_times) / len(round_times):.4f}s/round")
