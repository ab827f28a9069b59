"""Custom attack via subclassing — the three attack hook points.

MI355X equivalent of reference examples/customize_attack.py: subclass
``ByzantineClient`` and override any of

* ``local_training``        — malicious local optimization,
* ``on_train_batch_begin``  — per-batch data poisoning,
* ``omniscient_callback``   — post-gather update crafting with full
                              knowledge of the simulator.

Custom subclasses run on the reference-exact per-client loop engine
automatically; the built-in population stays on the fused batched engine.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from blades_amd import ByzantineClient, Simulator
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import MLP


class MedianSaboteurClient(ByzantineClient):
    """Crafts its update to sit just past the honest coordinate-wise median."""

    def __init__(self, shift: float = 1.0, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.shift = shift

    def omniscient_callback(self, simulator):
        honest = [c.get_update() for c in simulator.get_clients()
                  if not c.is_byzantine()]
        U = torch.stack(honest)
        med = U.median(dim=0).values
        self._state["saved_update"] = med + self.shift * U.std(dim=0)


num_clients = 12
dataset = SyntheticFLDataset(num_clients=num_clients, samples_per_client=64,
                             batch_size=32, shape=(1, 28, 28), seed=0,
                             learnable=True)

simulator = Simulator(dataset=dataset, aggregator="median",
                      use_cuda=torch.cuda.is_available(), seed=1,
                      log_path="./outputs/customize_attack")
# replace the first three clients with the custom attacker
simulator.register_attackers(
    [MedianSaboteurClient(shift=1.5) for _ in range(3)])

simulator.run(MLP(), global_rounds=10, local_steps=2,
              server_lr=1.0, client_lr=0.1, validate_interval=5)
print("custom attack simulation finished")
