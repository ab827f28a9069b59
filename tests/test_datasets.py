"""Data layer: partitioners, cache format, FLDataset plumbing, synthetic."""
import os
import pickle

import numpy as np
import pytest
import torch

from blades_amd.datasets import (BaseDataset, CustomTensorDataset, FLDataset,
                                 SyntheticFLDataset, dirichlet_partition,
                                 iid_partition)


class TinyDataset(BaseDataset):
    """A BaseDataset over in-memory random arrays (tests the cache +
    generator machinery without real files)."""

    num_classes = 4

    def generate_datasets(self, path="./data", iid=True, alpha=0.1,
                          num_clients=4, seed=1):
        rng = np.random.RandomState(seed)
        x = rng.randn(64, 3, 8, 8).astype("float32")
        y = rng.randint(0, self.num_classes, 64)
        from blades_amd.datasets.partition import build_client_dicts
        splits = (iid_partition(64, num_clients) if iid else
                  dirichlet_partition(y, num_clients, alpha, self.num_classes,
                                      min_size_floor=2, rng=rng))
        ids, train = build_client_dicts(x, y, splits)
        _, test = build_client_dicts(x[:16], y[:16], iid_partition(16, num_clients))
        return ids, train, ids, test


def test_iid_partition_covers_all():
    splits = iid_partition(100, 7)
    assert sum(len(s) for s in splits) == 100
    assert len(np.unique(np.concatenate(splits))) == 100


def test_dirichlet_partition_covers_all_and_skews():
    rng = np.random.RandomState(0)
    y = rng.randint(0, 10, 1000)
    splits = dirichlet_partition(y, 8, alpha=0.1, num_classes=10, rng=rng)
    allidx = np.concatenate(splits)
    assert len(allidx) == 1000 and len(np.unique(allidx)) == 1000
    assert min(len(s) for s in splits) >= 10
    # low alpha → skewed label distributions (some client misses some class)
    missing = 0
    for s in splits:
        missing += 10 - len(np.unique(y[s]))
    assert missing > 0


def test_cache_roundtrip_and_meta_invalidation(tmp_path):
    root = str(tmp_path)
    ds = TinyDataset(data_root=root, train_bs=8, num_clients=4, seed=1)
    path = os.path.join(root, "TinyDataset.obj")
    assert os.path.exists(path)
    # reference cache layout: [meta, train_ids, train_data, test_ids, test_data]
    with open(path, "rb") as f:
        objs = [pickle.load(f) for _ in range(5)]
    assert isinstance(objs[0], dict) and objs[0]["num_clients"] == 4
    assert sorted(objs[1]) == sorted(objs[3])
    mtime = os.path.getmtime(path)
    # same meta → no regeneration
    TinyDataset(data_root=root, train_bs=8, num_clients=4, seed=1)
    assert os.path.getmtime(path) == mtime
    # different meta → regenerated
    TinyDataset(data_root=root, train_bs=8, num_clients=2, seed=1)
    with open(path, "rb") as f:
        meta = pickle.load(f)
    assert meta["num_clients"] == 2


def test_get_dls_and_fldataset(tmp_path):
    ds = TinyDataset(data_root=str(tmp_path), train_bs=4, num_clients=4, seed=1)
    train_dls, test_dls = ds.get_dls()
    fl = FLDataset(train_dls, test_dls)
    assert fl.get_clients() == [0, 1, 2, 3]
    batches = fl.get_train_data(0, 3)
    assert len(batches) == 3
    x, y = batches[0]
    assert x.shape == (4, 3, 8, 8) and y.dtype == torch.int64
    test = fl.get_all_test_data(1)
    assert len(test) == 4
    xt, yt = test[0]
    assert xt.shape == (3, 8, 8)


def test_infinite_generator_wraps(tmp_path):
    ds = TinyDataset(data_root=str(tmp_path), train_bs=8, num_clients=4, seed=1)
    fl = FLDataset(*ds.get_dls())
    # each client has 16 samples → 2 batches/epoch; draw 5 without StopIteration
    batches = fl.get_train_data(0, 5)
    assert len(batches) == 5


def test_synthetic_layout_invariance():
    """Per-client pools depend only on (seed, client id), not access order."""
    a = SyntheticFLDataset(num_clients=4, samples_per_client=8, batch_size=4,
                           shape=(2, 4, 4), seed=7)
    b = SyntheticFLDataset(num_clients=4, samples_per_client=8, batch_size=4,
                           shape=(2, 4, 4), seed=7)
    a.materialize([3, 1])
    xb = b.get_train_data(3, 1)[0][0]
    xa = a.get_train_data(3, 1)[0][0]
    assert torch.equal(xa, xb)


def test_synthetic_stacked_matches_per_client():
    ds1 = SyntheticFLDataset(num_clients=3, samples_per_client=8, batch_size=4,
                             shape=(2, 4, 4), seed=1)
    ds2 = SyntheticFLDataset(num_clients=3, samples_per_client=8, batch_size=4,
                             shape=(2, 4, 4), seed=1)
    stacked = ds1.get_stacked_train_data([0, 1, 2], 2)
    for s in range(2):
        for c in range(3):
            x, y = ds2.get_train_data(c, 1)[0]
            assert torch.equal(stacked[s][0][c], x)
            assert torch.equal(stacked[s][1][c], y)
        # note: ds2 cursor interleaving differs; re-fetch per step order
    # cursor advance: second stacked step ≠ first
    assert not torch.equal(stacked[0][0], stacked[1][0])


def test_transforms_shapes():
    from blades_amd.datasets.transforms import (Compose, Normalize,
                                                RandomErasing,
                                                RandomHorizontalFlip,
                                                RandomResizedCrop)
    t = Compose([RandomResizedCrop(32, scale=(0.75, 1.0), ratio=(1.0, 1.0)),
                 RandomHorizontalFlip(0.5),
                 Normalize((0.5, 0.5, 0.5), (0.2, 0.2, 0.2)),
                 RandomErasing(1.0)])
    x = torch.rand(3, 32, 32)
    out = t(x)
    assert out.shape == (3, 32, 32)
    xb = torch.rand(5, 3, 32, 32)
    assert t(xb).shape == (5, 3, 32, 32)


def test_leaf_cli_pipeline(tmp_path):
    """Full LEAF preprocessing pipeline on a tiny synthetic all_data set
    (reference: models/utils/preprocess.sh stages)."""
    import json
    import subprocess
    import sys

    from blades_amd.datasets import leaf_cli

    data_dir = tmp_path / "femnist" / "data"
    all_dir = data_dir / "all_data"
    all_dir.mkdir(parents=True)
    users = [f"u{i}" for i in range(20)]
    data = {u: {"x": [[float(i), 0.5]] * (i + 1), "y": [i % 3] * (i + 1)}
            for i, u in enumerate(users)}
    leaf_cli.write_leaf(str(all_dir / "all_data_0.json"), users, data)

    leaf_cli.preprocess(str(data_dir), sample_mode="niid", fraction=0.5,
                        min_samples=3, train_fraction=0.8)
    tr_users, tr_counts, tr_data = leaf_cli.load_dir(str(data_dir / "train"))
    te_users, te_counts, te_data = leaf_cli.load_dir(str(data_dir / "test"))
    assert tr_users and te_users
    assert set(te_users) <= set(tr_users)  # sample-split keeps users
    for u, c in zip(tr_users, tr_counts):
        assert c >= 1
        # x/y lengths consistent
        assert len(tr_data[u]["x"]) == len(tr_data[u]["y"]) == c
    # checksum manifest written and verifies
    manifest = data_dir / "meta" / "dir-checksum.md5.json"
    assert manifest.exists()
    assert leaf_cli.checksum(str(data_dir), str(manifest), verify=True)
    # seeds persisted (preprocess.sh meta behavior)
    assert (data_dir / "meta" / "sampling_seed.txt").exists()

    # stats + CLI entry
    out = leaf_cli.stats(str(data_dir / "train"))
    assert "users:" in out
    p = subprocess.run([sys.executable, "-m",
                        "blades_amd.datasets.leaf_cli", "stats",
                        "--src", str(all_dir)],
                       capture_output=True, text=True)
    assert p.returncode == 0 and "users: 20" in p.stdout

    # iid sampling path
    leaf_cli.sample(str(all_dir), str(tmp_path / "iid"), 0.3, iid=True,
                    iid_user_fraction=0.2, seed=1)
    iu, ic, _ = leaf_cli.load_dir(str(tmp_path / "iid"))
    assert len(iu) == 4 and sum(ic) == int(0.3 * sum(i + 1 for i in range(20)))
