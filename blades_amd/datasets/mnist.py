"""MNIST federated dataset (reference: datasets/mnist.py:10-81).

Raw idx-format loader (no torchvision in this environment, no network):
expects ``train-images-idx3-ubyte`` etc. (optionally .gz) under
``<data_root>/MNIST/raw``.  Partition logic shared with CIFAR via
:mod:`partition`.
"""
from __future__ import annotations

import gzip
import os
import struct
from typing import Optional

import numpy as np

from .basedataset import BaseDataset
from .partition import build_client_dicts, dirichlet_partition, iid_partition


def _read_idx(path: str) -> np.ndarray:
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        zero, dtype_code, ndim = struct.unpack(">HBB", f.read(4))
        dims = struct.unpack(">" + "I" * ndim, f.read(4 * ndim))
        data = np.frombuffer(f.read(), dtype=np.uint8)
        return data.reshape(dims)


def _find(root: str, name: str) -> Optional[str]:
    for cand in (os.path.join(root, "MNIST", "raw", name),
                 os.path.join(root, "MNIST", "raw", name + ".gz"),
                 os.path.join(root, name),
                 os.path.join(root, name + ".gz")):
        if os.path.exists(cand):
            return cand
    return None


class MNIST(BaseDataset):
    num_classes = 10

    def __init__(self, data_root: str = "./data", train_bs: Optional[int] = 32,
                 iid: Optional[bool] = True, alpha: Optional[float] = 0.1,
                 num_clients: Optional[int] = 20, seed: int = 1):
        super().__init__(data_root, train_bs, iid, alpha, num_clients, seed)

    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=20, seed=1):
        files = {k: _find(path, n) for k, n in [
            ("xtr", "train-images-idx3-ubyte"), ("ytr", "train-labels-idx1-ubyte"),
            ("xte", "t10k-images-idx3-ubyte"), ("yte", "t10k-labels-idx1-ubyte")]}
        if any(v is None for v in files.values()):
            raise FileNotFoundError(
                f"MNIST raw idx files not found under {path} (no network in "
                "this environment to download; place train-images-idx3-ubyte "
                "etc. there, or use SyntheticFLDataset)")
        x_train = _read_idx(files["xtr"]).astype("float32") / 255.0
        y_train = _read_idx(files["ytr"]).astype("int64")
        x_test = _read_idx(files["xte"]).astype("float32") / 255.0
        y_test = _read_idx(files["yte"]).astype("int64")

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
