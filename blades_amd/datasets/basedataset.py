"""BaseDataset — client partition cache + per-client batch streams.

Compat contract (reference: datasets/basedataset.py:26-51,98-115): the
on-disk cache is one file ``<root>/<ClassName>.obj`` holding the pickle
stream ``meta_info, train_ids, train_data, test_ids, test_data`` keyed by
(num_clients, data_root, train_bs, iid, alpha, seed) — a cache written by
the reference loads here and vice versa (tests/test_datasets.py cache
round-trip) — and each client's train stream is an infinite shuffled batch
generator whose RNG sequence is: ``set_random_seed(seed)`` once at first
pull, then exactly one ``np.random.permutation`` per epoch, transforms
drawing from the global torch stream per batch.  Everything else
(structure, helpers) is our own; partition logic lives in :mod:`partition`
so every dataset class shares one tested implementation (the reference
duplicated the loop per dataset and had the ``.dat`` typo, SURVEY.md §2.1
known-bugs).  The reference applies ``train_transform`` to test data
(basedataset.py:95); here test data gets ``test_transform``.
"""
from __future__ import annotations

import os
import pickle
from abc import ABC, abstractmethod
from typing import Optional

import numpy as np
import torch

from blades_amd.utils import set_random_seed
from .customdataset import CustomTensorDataset

# pickle stream layout: meta dict first, then these four objects in order
_CACHE_FIELDS = ("train_ids", "train_data", "test_ids", "test_data")


def _read_cache(path: str):
    """Return the 5-object pickle stream as (meta, dict-of-fields)."""
    with open(path, "rb") as f:
        meta = pickle.load(f)
        body = {name: pickle.load(f) for name in _CACHE_FIELDS}
    return meta, body


def _write_cache(path: str, meta: dict, objects) -> None:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "wb") as f:
        pickle.dump(meta, f)
        for obj in objects:
            pickle.dump(obj, f)


class BaseDataset(ABC):
    train_transform = None
    test_transform = None

    def __init__(self, data_root: str = "./data", train_bs: Optional[int] = 32,
                 iid: Optional[bool] = True, alpha: Optional[float] = 0.1,
                 num_clients: Optional[int] = 20, seed: int = 1):
        self.train_bs = train_bs
        self._data_path = os.path.join(data_root, type(self).__name__ + ".obj")
        # key order matters for byte-compat with reference-written caches
        self._meta = {
            "num_clients": num_clients,
            "data_root": data_root,
            "train_bs": train_bs,
            "iid": iid,
            "alpha": alpha,
            "seed": seed,
        }
        if not self._cache_is_valid():
            parts = self.generate_datasets(data_root, iid, alpha,
                                           num_clients, seed)
            _write_cache(self._data_path, self._meta, parts)

    def _cache_is_valid(self) -> bool:
        if not os.path.exists(self._data_path):
            return False
        with open(self._data_path, "rb") as f:
            return pickle.load(f) == self._meta

    @abstractmethod
    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=20, seed=1):
        """Return (train_ids, train_data, test_ids, test_data) where
        *_data maps id -> {'x': array, 'y': array}."""

    # ------------------------------------------------------------- streams
    def _train_stream(self, data, labels, batch_size, seed=0):
        """Infinite shuffled batch generator.

        RNG contract (see module docstring): seed once, one permutation per
        epoch applied to the *current* order, batches sliced in order with
        a possibly-partial tail batch, transform drawn per batch.
        """
        set_random_seed(seed)
        n = len(labels)
        while True:
            order = np.random.permutation(n)
            data, labels = data[order], labels[order]
            for lo in range(0, n, batch_size):
                X = torch.Tensor(data[lo:lo + batch_size])
                if self.train_transform:
                    X = self.train_transform(X)
                yield X, torch.LongTensor(labels[lo:lo + batch_size])

    # kept under the reference's protected names so subclasses/tests that
    # poke them keep working
    _preprocess_train_data = _train_stream

    def _preprocess_test_data(self, data, labels) -> CustomTensorDataset:
        return CustomTensorDataset(torch.Tensor(data), torch.LongTensor(labels),
                                   transform_list=self.test_transform)

    def get_dls(self):
        if not os.path.isfile(self._data_path):
            raise FileNotFoundError(self._data_path)
        _, body = _read_cache(self._data_path)
        ids = body["train_ids"]
        if sorted(ids) != sorted(body["test_ids"]):
            raise ValueError("train/test client id sets differ in cache")
        train_dls = [
            self._train_stream(np.array(body["train_data"][u]["x"]),
                               np.array(body["train_data"][u]["y"]),
                               self.train_bs)
            for u in ids
        ]
        test_dls = [
            self._preprocess_test_data(np.array(body["test_data"][u]["x"]),
                                       np.array(body["test_data"][u]["y"]))
            for u in ids
        ]
        return train_dls, test_dls
