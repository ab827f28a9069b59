"""Label-flipping attack (reference: attackers/labelflippingclient.py:12-26).

Trains on y → num_classes − 1 − y.  K14: on the fused engine this is a
per-client integer map applied to the stacked target tensor (no per-client
Python loop); on the loop engine the ``on_train_batch_begin`` hook fires
exactly as in the reference.
"""
from __future__ import annotations

from blades_amd.client import ByzantineClient


class LabelflippingClient(ByzantineClient):
    def __init__(self, num_classes: int = 10, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.num_classes = num_classes

    def on_train_batch_begin(self, data, target, logs=None):
        return data, self.num_classes - 1 - target

    # fused-engine metadata
    def fused_target_transform(self, target):
        return self.num_classes - 1 - target

    def __str__(self) -> str:
        return "LabelFlippingClient"
