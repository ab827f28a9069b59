"""Data layer: FL dataset plumbing, partitioners, raw-format loaders.

Reference surface: BaseDataset, FLDataset, CIFAR10, MNIST,
CustomTensorDataset; plus this framework's CIFAR100 and the device-resident
SyntheticFLDataset used by the benchmark configs.
"""
from .basedataset import BaseDataset
from .dataset import FLDataset
from .customdataset import CustomTensorDataset
from .mnist import MNIST
from .cifar10 import CIFAR10, CIFAR100
from .synthetic import SyntheticFLDataset
from .partition import dirichlet_partition, iid_partition

__all__ = [
    "BaseDataset", "FLDataset", "CustomTensorDataset", "MNIST",
    "CIFAR10", "CIFAR100", "SyntheticFLDataset",
    "dirichlet_partition", "iid_partition",
]
