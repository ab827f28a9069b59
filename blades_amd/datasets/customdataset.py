"""Tensor-pair dataset (reference: datasets/customdataset.py:4-21)."""
from __future__ import annotations

import torch
from torch.utils.data import Dataset


class CustomTensorDataset(Dataset):
    def __init__(self, tensor_x: torch.Tensor, tensor_y: torch.Tensor,
                 transform_list=None):
        assert len(tensor_x) == len(tensor_y)
        self.x = tensor_x
        self.y = tensor_y
        self.transform = transform_list

    def __len__(self):
        return len(self.x)

    def __getitem__(self, idx):
        x = self.x[idx]
        if self.transform is not None:
            x = self.transform(x)
        return x, self.y[idx]
