"""Real-data ingestion for the reference's on-disk layouts.

The drift configs run on synthetic twins when nothing real is on disk
(data/generators.py — no network in this environment), but a user moving
from the reference brings its data directories along. This module reads
those layouts directly:

  * LEAF json (MNIST / FEMNIST / shakespeare style): train/test dirs of
    .json files with 'users' and 'user_data' keys — exact semantics of
    fedml_api/data_preprocessing/MNIST/data_loader.py:read_data (pooled
    users, sorted client ids). `LeafSampleSource` replicates the drift
    generator's MNIST_Data exactly (data_loader_cont.py:151-214): pool
    all users, one seed-100 legacy-numpy shuffle, sequential draws with
    wrap-around, and the label-swap concepts (1<->2, 3<->4, 5<->6).
  * FederatedEMNIST h5 (FederatedEMNIST/data_loader.py:28-60): 'pixels'
    / 'label' / 'id' datasets, natural clients grouped by id. Requires
    h5py (optional in this image; a clear error says so).
  * FMoW-style per-(client, iteration) index CSVs
    (fmow/data_loader.py:63-83): partitions/{P}/client_{c}_iter_{t}.csv
    holds row indices into the underlying dataset, including the
    reference's singleton special case. The WILDS package (and its
    download) is unavailable offline, so the backing pixels come from
    features.npy / labels.npy arrays in the dataset root — the same
    lazy index-follows-data layout, with the image decode swapped for a
    feature matrix.

generators.generate_data() switches to the real pixel source
automatically when the LEAF layout is present under the dataset dir.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Tuple

import numpy as np


# ---------------------------------------------------------------------------
# LEAF json
# ---------------------------------------------------------------------------
def read_leaf_json(train_data_dir: str, test_data_dir: str):
    """Reference read_data (MNIST/data_loader.py:14-54): returns
    (clients, groups, train_data, test_data); clients sorted by id."""
    clients: List[str] = []
    groups: List[str] = []
    train_data: Dict = {}
    test_data: Dict = {}
    for f in sorted(os.listdir(train_data_dir)):
        if not f.endswith(".json"):
            continue
        with open(os.path.join(train_data_dir, f)) as inf:
            cdata = json.load(inf)
        clients.extend(cdata["users"])
        if "hierarchies" in cdata:
            groups.extend(cdata["hierarchies"])
        train_data.update(cdata["user_data"])
    for f in sorted(os.listdir(test_data_dir)):
        if not f.endswith(".json"):
            continue
        with open(os.path.join(test_data_dir, f)) as inf:
            cdata = json.load(inf)
        test_data.update(cdata["user_data"])
    clients = list(sorted(train_data.keys()))
    return clients, groups, train_data, test_data


def leaf_layout_present(ds_dir: str) -> bool:
    train = os.path.join(ds_dir, "train")
    return os.path.isdir(train) and any(
        f.endswith(".json") for f in os.listdir(train))


_SWAPS = {1: (1.0, 2.0), 2: (3.0, 4.0), 3: (5.0, 6.0)}


class LeafSampleSource:
    """Drift-path pixel source over a LEAF layout — exact MNIST_Data
    semantics (data_loader_cont.py:151-214)."""

    def __init__(self, ds_dir: str):
        users, _, train_data, _ = read_leaf_json(
            os.path.join(ds_dir, "train"), os.path.join(ds_dir, "test"))
        X, Y = [], []
        for u in users:
            X.extend(train_data[u]["x"])
            Y.extend(train_data[u]["y"])
        nX = np.asarray(X, dtype=np.float64)
        nY = np.asarray(Y, dtype=np.float64)
        # the reference's exact shuffle: legacy np.random at seed 100,
        # same state replayed for X and Y (:167-171)
        np.random.seed(100)
        rng_state = np.random.get_state()
        np.random.shuffle(nX)
        np.random.set_state(rng_state)
        np.random.shuffle(nY)
        self.nX = nX
        self.nY = nY
        self.samples_used = 0

    def generate_sample(self, num_sample: int, concept_k: int,
                        rng=None) -> np.ndarray:
        """[n, D+1] rows; label swaps per concept; sequential draws with
        the reference's wrap-around quirk (:182-184)."""
        if self.samples_used + num_sample >= len(self.nX):
            self.samples_used = 0
        rows = []
        for i in range(self.samples_used, self.samples_used + num_sample):
            x = self.nX[i]
            y = self.nY[i]
            if concept_k in _SWAPS:
                a, b = _SWAPS[concept_k]
                if y == a:
                    y = b
                elif y == b:
                    y = a
            rows.append(np.concatenate((x, [y])))
        self.samples_used += num_sample
        return np.asarray(rows)


# ---------------------------------------------------------------------------
# FederatedEMNIST h5
# ---------------------------------------------------------------------------
def read_femnist_h5(h5_path: str,
                    max_clients: Optional[int] = None
                    ) -> Dict[int, Tuple[np.ndarray, np.ndarray]]:
    """Natural-client data from the reference h5 layout ('pixels',
    'label', 'id' — FederatedEMNIST/data_loader.py:28-41). Returns
    {client_index: (x [n, 784] f32, y [n] i64)} with clients ordered by
    first appearance of their id."""
    try:
        import h5py
    except ImportError as e:  # pragma: no cover
        raise RuntimeError(
            "reading FederatedEMNIST h5 files requires h5py, which is "
            "not installed in this image; convert the h5 to the LEAF "
            "json layout or install h5py") from e
    out: Dict[int, Tuple[np.ndarray, np.ndarray]] = {}
    with h5py.File(h5_path, "r") as f:
        pixels = np.asarray(f["pixels"])
        label = np.asarray(f["label"]).astype(np.int64)
        ids = np.asarray(f["id"])
        order = []
        seen = set()
        for i in ids:
            k = i.item() if hasattr(i, "item") else i
            if k not in seen:
                seen.add(k)
                order.append(k)
        for ci, k in enumerate(order):
            if max_clients is not None and ci >= max_clients:
                break
            sel = ids == k
            x = pixels[sel].reshape(int(sel.sum()), -1).astype(np.float32)
            out[ci] = (x, label[sel])
    return out


# ---------------------------------------------------------------------------
# FMoW-style lazy index partitions
# ---------------------------------------------------------------------------
def read_fmow_index(path: str) -> np.ndarray:
    """Index CSV for one (client, iteration) — including the reference's
    single-row special case (fmow/data_loader.py:66-69)."""
    subidxs = np.loadtxt(path, dtype=int, delimiter=",")
    if subidxs.size == 1:
        subidxs = np.asarray([subidxs.item()])
    return subidxs


class FmowIndexStore:
    """RawStore-compatible view over the reference FMoW partition layout:
    partitions/{P}/client_{c}_iter_{t}.csv index files resolved lazily
    against a features/labels backing store (features.npy / labels.npy
    in the dataset root, standing in for the offline-unavailable WILDS
    image archive)."""

    def __init__(self, data_dir: str, partition_name: str = "A",
                 num_client: int = 0):
        self.dataset = "fmow"
        self.dir = data_dir
        self.part = os.path.join(data_dir, "partitions", partition_name)
        self.num_client = num_client
        self.features = np.load(os.path.join(data_dir, "features.npy"),
                                mmap_mode="r")
        self.labels = np.load(os.path.join(data_dir, "labels.npy"))
        self._cache: Dict = {}

    @staticmethod
    def layout_present(data_dir: str, partition_name: str = "A") -> bool:
        return (os.path.isdir(os.path.join(data_dir, "partitions",
                                           partition_name))
                and os.path.exists(os.path.join(data_dir, "features.npy")))

    def get(self, c: int, t: int) -> Tuple[np.ndarray, np.ndarray]:
        key = (c, t)
        if key not in self._cache:
            path = os.path.join(self.part, f"client_{c}_iter_{t}.csv")
            if not os.path.exists(path):
                self._cache[key] = (
                    np.zeros((0, self.features.shape[1]), np.float32),
                    np.zeros((0,), np.int64))
            else:
                idx = read_fmow_index(path)
                self._cache[key] = (
                    np.asarray(self.features[idx], dtype=np.float32),
                    self.labels[idx].astype(np.int64))
        return self._cache[key]

    def put(self, c: int, t: int, x: np.ndarray, y: np.ndarray) -> None:
        self._cache[(c, t)] = (np.asarray(x, dtype=np.float32),
                               np.asarray(y, dtype=np.int64))
