"""Sign-flipping attack (reference: attackers/signflippingclient.py:6-21).

Gradient ascent: every local step negates all gradients before the
optimizer step (loss clamped to [0, 1e5] — the reference's tighter clamp).
K13: on the fused engine this is a per-client −1 multiplier on the gradient
slab row; on the loop engine the overridden ``local_training`` runs as-is.
"""
from __future__ import annotations

import torch

from blades_amd.client import ByzantineClient


class SignflippingClient(ByzantineClient):
    # fused-engine metadata
    fused_grad_sign: float = -1.0
    fused_loss_clamp: float = 1e5

    def local_training(self, data_batches: list) -> None:
        for data, target in data_batches:
            data, target = data.to(self.device), target.to(self.device)
            data, target = self.on_train_batch_begin(data=data, target=target)
            self.optimizer.zero_grad()
            output = self.model(data)
            loss = torch.clamp(self.loss_func(output, target), 0, 1e5)
            loss.backward()
            for _, p in self.model.named_parameters():
                if p.grad is not None:
                    p.grad.data = -p.grad.data
            self.optimizer.step()
