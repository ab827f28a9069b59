"""BaseDataset — partition + pickle cache + per-client generators.

Cache format parity with the reference (reference: datasets/
basedataset.py:26-51,98-115): one file ``<root>/<ClassName>.obj`` holding a
pickle stream ``[meta_info, train_ids, train_data, test_ids, test_data]``
keyed by (num_clients, data_root, train_bs, iid, alpha, seed) — a cache
written by the reference loads here and vice versa.

Partition helpers (iid split, Dirichlet(α) label-skew) live in
:mod:`partition` so every dataset class shares one tested implementation
(the reference duplicated the loop per dataset and had the ``.dat`` typo,
SURVEY.md §2.1 known-bugs).
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


class BaseDataset(ABC):
    train_transform = None
    test_transform = None

    def __init__(self, data_root: str = "./data", train_bs: Optional[int] = 32,
                 iid: Optional[bool] = True, alpha: Optional[float] = 0.1,
                 num_clients: Optional[int] = 20, seed: int = 1):
        self.train_bs = train_bs
        self._data_path = os.path.join(data_root, self.__class__.__name__ + ".obj")

        meta_info = {
            "num_clients": num_clients,
            "data_root": data_root,
            "train_bs": train_bs,
            "iid": iid,
            "alpha": alpha,
            "seed": seed,
        }

        regenerate = True
        if os.path.exists(self._data_path):
            with open(self._data_path, "rb") as f:
                loaded_meta_info = pickle.load(f)
                if loaded_meta_info == meta_info:
                    regenerate = False

        if regenerate:
            returns = self.generate_datasets(data_root, iid, alpha, num_clients, seed)
            os.makedirs(data_root, exist_ok=True)
            with open(self._data_path, "wb") as f:
                pickle.dump(meta_info, f)
                for obj in returns:
                    pickle.dump(obj, f)

    @abstractmethod
    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=20, seed=1):
        """Return (train_ids, train_data, test_ids, test_data) where
        *_data maps id -> {'x': array, 'y': array}."""

    def _preprocess_train_data(self, data, labels, batch_size, seed=0):
        """Infinite shuffled batch generator (reference: basedataset.py:58-86)."""
        i = 0
        set_random_seed(seed)
        idx = np.random.permutation(len(labels))
        data, labels = data[idx], labels[idx]

        while True:
            if i * batch_size >= len(labels):
                i = 0
                idx = np.random.permutation(len(labels))
                data, labels = data[idx], labels[idx]
                continue
            X = data[i * batch_size:(i + 1) * batch_size]
            y = labels[i * batch_size:(i + 1) * batch_size]
            i += 1
            X = torch.Tensor(X)
            if self.train_transform:
                X = self.train_transform(X)
            yield X, torch.LongTensor(y)

    def _preprocess_test_data(self, data, labels) -> CustomTensorDataset:
        return CustomTensorDataset(torch.Tensor(data), torch.LongTensor(labels),
                                   transform_list=self.test_transform)

    def get_dls(self):
        assert os.path.isfile(self._data_path)
        with open(self._data_path, "rb") as f:
            (_, train_clients, train_data, test_clients, test_data) = [
                pickle.load(f) for _ in range(5)
            ]
        assert sorted(train_clients) == sorted(test_clients)

        train_dls, test_dls = [], []
        for u_id in train_clients:
            train_dls.append(self._preprocess_train_data(
                data=np.array(train_data[u_id]["x"]),
                labels=np.array(train_data[u_id]["y"]),
                batch_size=self.train_bs,
            ))
            test_dls.append(self._preprocess_test_data(
                data=np.array(test_data[u_id]["x"]),
                labels=np.array(test_data[u_id]["y"]),
            ))
        return train_dls, test_dls
