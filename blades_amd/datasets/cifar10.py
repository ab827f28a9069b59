"""CIFAR-10 federated dataset (reference: datasets/cifar10.py:11-109).

Raw python-batch loader (no torchvision/network): expects
``cifar-10-batches-py/`` under ``data_root``.  Train transforms replicate
the reference pipeline with this framework's tensor-native transforms.
Fixes the reference's ``train_set.dat`` typo (SURVEY.md §2.1 known-bugs).
"""
from __future__ import annotations

import os
import pickle
from typing import Optional

import numpy as np

from .basedataset import BaseDataset
from .partition import build_client_dicts, dirichlet_partition, iid_partition
from .transforms import (Compose, Normalize, RandomErasing,
                         RandomHorizontalFlip, RandomResizedCrop)


def _load_cifar10(path: str):
    root = os.path.join(path, "cifar-10-batches-py")
    if not os.path.isdir(root):
        raise FileNotFoundError(
            f"{root} not found (no network to download; place the python "
            "batches there, or use SyntheticFLDataset)")

    def load_batch(name):
        with open(os.path.join(root, name), "rb") as f:
            d = pickle.load(f, encoding="bytes")
        return d[b"data"].reshape(-1, 3, 32, 32), np.array(d[b"labels"])

    xs, ys = zip(*[load_batch(f"data_batch_{i}") for i in range(1, 6)])
    x_train = np.concatenate(xs)
    y_train = np.concatenate(ys)
    x_test, y_test = load_batch("test_batch")
    return x_train, y_train, x_test, y_test


class CIFAR10(BaseDataset):
    num_classes = 10
    stats = {"mean": (0.4914, 0.4822, 0.4465), "std": (0.2023, 0.1994, 0.2010)}

    test_transform = Compose([Normalize(stats["mean"], stats["std"])])
    train_transform = Compose([
        RandomResizedCrop(32, scale=(0.75, 1.0), ratio=(1.0, 1.0)),
        RandomHorizontalFlip(0.5),
        Normalize(stats["mean"], stats["std"]),
        RandomErasing(0.25),
    ])

    def __init__(self, data_root: str = "./data", train_bs: Optional[int] = 32,
                 iid: Optional[bool] = True, alpha: Optional[float] = 0.1,
                 num_clients: Optional[int] = 20, seed: int = 1):
        super().__init__(data_root, train_bs, iid, alpha, num_clients, seed)

    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=20, seed=1):
        x_train, y_train, x_test, y_test = _load_cifar10(path)
        x_train = x_train.astype("float32") / 255.0
        x_test = x_test.astype("float32") / 255.0

        rng = np.random.RandomState(seed)
        perm = rng.permutation(len(x_train))
        x_train, y_train = x_train[perm], y_train[perm]
        perm = rng.permutation(len(x_test))
        x_test, y_test = x_test[perm], y_test[perm]

        if iid:
            splits = iid_partition(len(x_train), num_clients)
        else:
            splits = dirichlet_partition(y_train, num_clients, alpha,
                                         self.num_classes, rng=rng)
        train_ids, train_data = build_client_dicts(x_train, y_train, splits)
        test_splits = iid_partition(len(x_test), num_clients)
        test_ids, test_data = build_client_dicts(x_test, y_test, test_splits)
        return train_ids, train_data, test_ids, test_data


class CIFAR100(CIFAR10):
    """CIFAR-100 (new capability; the WRN-28-10 benchmark config needs it)."""

    num_classes = 100
    stats = {"mean": (0.5071, 0.4865, 0.4409), "std": (0.2673, 0.2564, 0.2762)}

    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=20, seed=1):
        root = os.path.join(path, "cifar-100-python")
        if not os.path.isdir(root):
            raise FileNotFoundError(f"{root} not found (no network to download)")

        def load(name):
            with open(os.path.join(root, name), "rb") as f:
                d = pickle.load(f, encoding="bytes")
            return d[b"data"].reshape(-1, 3, 32, 32), np.array(d[b"fine_labels"])

        x_train, y_train = load("train")
        x_test, y_test = load("test")
        x_train = x_train.astype("float32") / 255.0
        x_test = x_test.astype("float32") / 255.0

        rng = np.random.RandomState(seed)
        perm = rng.permutation(len(x_train))
        x_train, y_train = x_train[perm], y_train[perm]
        perm = rng.permutation(len(x_test))
        x_test, y_test = x_test[perm], y_test[perm]

        if iid:
            splits = iid_partition(len(x_train), num_clients)
        else:
            splits = dirichlet_partition(y_train, num_clients, alpha,
                                         self.num_classes, rng=rng)
        train_ids, train_data = build_client_dicts(x_train, y_train, splits)
        test_splits = iid_partition(len(x_test), num_clients)
        test_ids, test_data = build_client_dicts(x_test, y_test, test_splits)
        return train_ids, train_data, test_ids, test_data
