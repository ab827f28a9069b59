"""MNIST MLP — 784-64-128-10, log-softmax output.

Architecture parity with the reference model zoo
(reference: src/blades/models/mnist/dnn.py:5-18, d = 59,850 params);
implementation is this framework's own.
"""
from __future__ import annotations

import torch.nn.functional as F
from torch import nn


class MLP(nn.Module):
    def __init__(self, in_features: int = 28 * 28, num_classes: int = 10):
        super().__init__()
        self.flatten = nn.Flatten()
        self.layer1 = nn.Linear(in_features, 64)
        self.layer2 = nn.Linear(64, 128)
        self.layer3 = nn.Linear(128, num_classes)

    def forward(self, x):
        x = self.flatten(x)
        x = F.relu(self.layer1(x))
        x = F.relu(self.layer2(x))
        return F.log_softmax(self.layer3(x), dim=1)


def create_model():
    return MLP(), nn.CrossEntropyLoss()
