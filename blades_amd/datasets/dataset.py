"""FLDataset — client id → (train generator, test Dataset)
(reference: datasets/dataset.py:80-115)."""
from __future__ import annotations

from warnings import warn


class FLDataset:
    def __init__(self, train_dataloaders: list, test_dataloaders: list = None):
        if not test_dataloaders:
            warn("No test data is given. Model evaluation will be based on train data.")
            test_dataloaders = train_dataloaders
        if len(train_dataloaders) != len(test_dataloaders):
            raise Exception(
                "Invalid Input: Numbers of train dataloaders and test "
                "dataloaders should be equal."
            )
        self._train_dls = {}
        self._test_dls = {}
        for idx, (traindl, testdl) in enumerate(zip(train_dataloaders, test_dataloaders)):
            self._train_dls[idx] = traindl
            self._test_dls[idx] = testdl
        self._clients = list(range(len(self._train_dls)))

    def get_clients(self):
        return self._clients

    def get_train_data(self, u_id, num_batches):
        return [next(self._train_dls[u_id]) for _ in range(num_batches)]

    def get_all_test_data(self, u_id):
        return self._test_dls[u_id]
