"""LEAF preprocessing pipeline — MI-native reimplementation of the
reference's vendored LEAF CLI (reference: models/utils/{sample.py,
split_data.py, remove_users.py, stats.py, preprocess.sh, download_util.py,
constants.py} — standalone argparse scripts there; one typed pipeline
here).

Operates on the LEAF interchange format: JSON files with keys
``users`` (list of ids), ``num_samples`` (per-user counts) and
``user_data`` (id -> {"x": [...], "y": [...]}) under
``<root>/<dataset>/data/all_data/``.  Stages write to ``sampled_data`` /
``rem_user_data`` / ``train`` + ``test`` exactly like the upstream LEAF
layout, so artifacts interoperate.

Subcommands (``python -m blades_amd.datasets.leaf_cli <cmd> ...``):

* ``sample``        — iid or non-iid subsampling by data fraction
* ``remove-users``  — drop users with fewer than k samples
* ``split``         — train/test split by user or by sample
* ``stats``         — population/sample-count statistics (text histogram)
* ``checksum``      — write/verify an MD5 manifest (replaces the
  reference's gdrive download_util: this environment has no egress, so
  integrity checking of already-present data is the supported operation)
* ``preprocess``    — the full pipeline (the reference's preprocess.sh)
"""
from __future__ import annotations

import argparse
import hashlib
import json
import os
import random
from collections import OrderedDict
from typing import Dict, List, Tuple

from .leaf import iid_divide

DATASETS = ["sent140", "femnist", "shakespeare", "celeba", "synthetic"]
SEED_FILES = {"sampling": "sampling_seed.txt", "split": "split_seed.txt"}


# ------------------------------------------------------------------ IO layer
def _json_files(d: str) -> List[str]:
    return sorted(f for f in os.listdir(d) if f.endswith(".json"))


def load_dir(d: str) -> Tuple[List[str], List[int], Dict]:
    """Merge every all_data-format JSON in ``d``."""
    users: List[str] = []
    counts: List[int] = []
    data: Dict = OrderedDict()
    for f in _json_files(d):
        with open(os.path.join(d, f)) as fh:
            blob = json.load(fh)
        users.extend(blob["users"])
        counts.extend(blob["num_samples"])
        data.update(blob["user_data"])
    return users, counts, data


def write_leaf(path: str, users: List[str], data: Dict) -> None:
    os.makedirs(os.path.dirname(path), exist_ok=True)
    blob = {
        "users": users,
        "num_samples": [len(data[u]["y"]) for u in users],
        "user_data": {u: data[u] for u in users},
    }
    with open(path, "w") as fh:
        json.dump(blob, fh)


def _persist_seed(data_dir: str, kind: str, seed: int) -> None:
    meta = os.path.join(data_dir, "meta")
    os.makedirs(meta, exist_ok=True)
    with open(os.path.join(meta, SEED_FILES[kind]), "w") as fh:
        fh.write(str(seed))


# ------------------------------------------------------------------- stages
def sample(src: str, dst: str, fraction: float, iid: bool,
           iid_user_fraction: float = 0.01, seed: int = 0) -> None:
    """Subsample ``fraction`` of all datapoints.

    non-iid: keep whole users (accumulate randomly-ordered users until the
    sample budget is reached).  iid: pool every datapoint, shuffle, and
    deal the budget round-robin to ``iid_user_fraction``·|users| synthetic
    uniform users (f_0000…) — same semantics as the reference sampler.
    """
    rng = random.Random(seed)
    users, counts, data = load_dir(src)
    budget = int(sum(counts) * fraction)
    out: Dict = OrderedDict()
    if not iid:
        order = list(range(len(users)))
        rng.shuffle(order)
        taken = 0
        for i in order:
            if taken >= budget:
                break
            u = users[i]
            out[u] = data[u]
            taken += counts[i]
    else:
        xs, ys = [], []
        for u in users:
            xs.extend(data[u]["x"])
            ys.extend(data[u]["y"])
        order = list(range(len(ys)))
        rng.shuffle(order)
        order = order[:budget]
        n_users = max(1, int(len(users) * iid_user_fraction))
        for g, chunk in enumerate(iid_divide(order, n_users)):
            out[f"f_{g:07d}"] = {"x": [xs[i] for i in chunk],
                                 "y": [ys[i] for i in chunk]}
    write_leaf(os.path.join(dst, "all_data_sampled.json"),
               list(out.keys()), out)


def remove_users(src: str, dst: str, min_samples: int) -> None:
    """Drop users holding fewer than ``min_samples`` datapoints."""
    users, counts, data = load_dir(src)
    keep = [u for u, c in zip(users, counts) if c >= min_samples]
    write_leaf(os.path.join(dst, "all_data_niid_keep.json"), keep, data)


def split_train_test(src: str, train_dst: str, test_dst: str,
                     train_fraction: float = 0.9, by_user: bool = False,
                     seed: int = 0) -> None:
    """Train/test split.

    by_user: whole users go to one side (split on the user list);
    by sample (default): each user's samples are split
    ``train_fraction`` / rest, preserving at least one sample per side
    when possible.
    """
    rng = random.Random(seed)
    users, counts, data = load_dir(src)
    train: Dict = OrderedDict()
    test: Dict = OrderedDict()
    if by_user:
        order = list(range(len(users)))
        rng.shuffle(order)
        cut = int(len(users) * train_fraction)
        for j, i in enumerate(order):
            (train if j < cut else test)[users[i]] = data[users[i]]
    else:
        for u, c in zip(users, counts):
            idx = list(range(c))
            rng.shuffle(idx)
            cut = min(max(int(c * train_fraction), 1), c - 1) if c > 1 \
                else c
            tr, te = idx[:cut], idx[cut:]
            x, y = data[u]["x"], data[u]["y"]
            train[u] = {"x": [x[i] for i in tr], "y": [y[i] for i in tr]}
            if te:
                test[u] = {"x": [x[i] for i in te],
                           "y": [y[i] for i in te]}
    write_leaf(os.path.join(train_dst, "train.json"),
               list(train.keys()), train)
    write_leaf(os.path.join(test_dst, "test.json"), list(test.keys()), test)


def stats(src: str) -> str:
    """Population statistics + a text histogram of samples/user (the
    reference plotted with matplotlib; this image has none, and the
    numbers are what the pipeline needs)."""
    users, counts, _ = load_dir(src)
    import numpy as np

    c = np.asarray(counts)
    lines = [f"users: {len(users)}",
             f"samples: total {c.sum()}, mean {c.mean():.2f}, "
             f"std {c.std():.2f}, min {c.min()}, max {c.max()}"]
    hist, edges = np.histogram(c, bins=min(10, max(1, len(set(counts)))))
    peak = max(1, hist.max())
    for h, lo, hi in zip(hist, edges[:-1], edges[1:]):
        bar = "#" * int(round(40 * h / peak))
        lines.append(f"  [{lo:9.1f}, {hi:9.1f}) {h:6d} {bar}")
    out = "\n".join(lines)
    print(out)
    return out


def checksum(root: str, manifest: str, verify: bool = False) -> bool:
    """Write (or verify) an MD5 manifest over every JSON under ``root``."""
    digests = {}
    man_abs = os.path.abspath(manifest)
    for dirpath, _, files in sorted(os.walk(root)):
        for f in sorted(files):
            if f.endswith(".json"):
                p = os.path.join(dirpath, f)
                if os.path.abspath(p) == man_abs:
                    continue  # the manifest never digests itself
                rel = os.path.relpath(p, root)
                digests[rel] = hashlib.md5(open(p, "rb").read()).hexdigest()
    if verify:
        with open(manifest) as fh:
            want = json.load(fh)
        ok = want == digests
        print("checksum", "OK" if ok else "MISMATCH")
        return ok
    os.makedirs(os.path.dirname(manifest) or ".", exist_ok=True)
    with open(manifest, "w") as fh:
        json.dump(digests, fh, indent=1)
    return True


def preprocess(data_dir: str, sample_mode: str = "niid",
               fraction: float = 0.1, min_samples: int = 0,
               train_fraction: float = 0.9, split_by_user: bool = False,
               sampling_seed: int = 0, split_seed: int = 0,
               iid_user_fraction: float = 0.01,
               write_checksum: bool = True) -> None:
    """The full pipeline (the reference's preprocess.sh): all_data ->
    sampled_data -> rem_user_data -> train/ + test/ (+ meta checksum)."""
    all_dir = os.path.join(data_dir, "all_data")
    sampled = os.path.join(data_dir, "sampled_data")
    kept = os.path.join(data_dir, "rem_user_data")
    sample(all_dir, sampled, fraction, iid=sample_mode == "iid",
           iid_user_fraction=iid_user_fraction, seed=sampling_seed)
    _persist_seed(data_dir, "sampling", sampling_seed)
    remove_users(sampled, kept, min_samples)
    split_train_test(kept, os.path.join(data_dir, "train"),
                     os.path.join(data_dir, "test"),
                     train_fraction=train_fraction, by_user=split_by_user,
                     seed=split_seed)
    _persist_seed(data_dir, "split", split_seed)
    if write_checksum:
        checksum(data_dir, os.path.join(data_dir, "meta",
                                        "dir-checksum.md5.json"))


# ---------------------------------------------------------------------- CLI
def main(argv=None) -> None:
    ap = argparse.ArgumentParser(prog="leaf_cli", description=__doc__)
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("sample")
    p.add_argument("--src", required=True)
    p.add_argument("--dst", required=True)
    p.add_argument("--fraction", type=float, default=0.1)
    p.add_argument("--iid", action="store_true")
    p.add_argument("--u", type=float, default=0.01,
                   help="iid synthetic-user fraction")
    p.add_argument("--seed", type=int, default=0)

    p = sub.add_parser("remove-users")
    p.add_argument("--src", required=True)
    p.add_argument("--dst", required=True)
    p.add_argument("-k", "--min-samples", type=int, default=1)

    p = sub.add_parser("split")
    p.add_argument("--src", required=True)
    p.add_argument("--train-dst", required=True)
    p.add_argument("--test-dst", required=True)
    p.add_argument("--tf", type=float, default=0.9)
    p.add_argument("--by-user", action="store_true")
    p.add_argument("--seed", type=int, default=0)

    p = sub.add_parser("stats")
    p.add_argument("--src", required=True)

    p = sub.add_parser("checksum")
    p.add_argument("--root", required=True)
    p.add_argument("--manifest", required=True)
    p.add_argument("--verify", action="store_true")

    p = sub.add_parser("preprocess")
    p.add_argument("--data-dir", required=True)
    p.add_argument("-s", "--sample-mode", choices=["iid", "niid"],
                   default="niid")
    p.add_argument("--sf", type=float, default=0.1)
    p.add_argument("-k", "--min-samples", type=int, default=0)
    p.add_argument("--tf", type=float, default=0.9)
    p.add_argument("-t", "--split-by", choices=["user", "sample"],
                   default="sample")
    p.add_argument("--smplseed", type=int, default=0)
    p.add_argument("--spltseed", type=int, default=0)

    a = ap.parse_args(argv)
    if a.cmd == "sample":
        sample(a.src, a.dst, a.fraction, a.iid, a.u, a.seed)
    elif a.cmd == "remove-users":
        remove_users(a.src, a.dst, a.min_samples)
    elif a.cmd == "split":
        split_train_test(a.src, a.train_dst, a.test_dst, a.tf, a.by_user,
                         a.seed)
    elif a.cmd == "stats":
        stats(a.src)
    elif a.cmd == "checksum":
        ok = checksum(a.root, a.manifest, a.verify)
        raise SystemExit(0 if ok else 1)
    elif a.cmd == "preprocess":
        preprocess(a.data_dir, a.sample_mode, a.sf, a.min_samples, a.tf,
                   a.split_by == "user", a.smplseed, a.spltseed)


if __name__ == "__main__":
    main()
