"""Loading + batchification + the retrain-window DSL.

Reference semantics reproduced here:
  * batch_data (sea/data_loader.py:15-35): shuffle the rows, then chunk into
    consecutive batches of batch_size (last batch may be short).
  * retrain DSL (common/retrain.py:7-91): 'all', 'win-W', 'weight-linear',
    'weight-exp' (row duplication), 'sel-i,j,...', 'clientsel-<json>',
    'poisson' (Poisson(1) bootstrap). Test set is ALWAYS iteration t+1
    (prequential evaluation, retrain.py:79-83).
  * load_all_data (retrain.py:87-91 + load_all_data_sea:84-99): the full
    per-client x per-iteration history, batched.

Engine-facing representation: instead of python lists of torch batch tuples,
every (client, iteration) segment is shuffled once and stored as contiguous
numpy arrays; a batch is a (start, length) window into its segment. The
device arena (engine/arena.py) uploads the whole thing to HBM once.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from .generators import CLASS_NUM, FEATURE_NUM


@dataclass
class Segment:
    """One shuffled (client, iteration) data segment plus its batch windows."""
    x: np.ndarray            # float32 [n, D]
    y: np.ndarray            # int64 [n]
    windows: List[Tuple[int, int]] = field(default_factory=list)  # (start, len)

    @property
    def n(self) -> int:
        return len(self.y)

    def batches(self):
        for s, ln in self.windows:
            yield self.x[s:s + ln], self.y[s:s + ln]


def batchify(x: np.ndarray, y: np.ndarray, batch_size: int,
             rng: np.random.Generator) -> Segment:
    n = len(y)
    perm = rng.permutation(n)
    xs = np.ascontiguousarray(x[perm], dtype=np.float32)
    ys = np.ascontiguousarray(y[perm], dtype=np.int64)
    windows = [(i, min(batch_size, n - i)) for i in range(0, n, batch_size)]
    return Segment(xs, ys, windows)


def _read_client_iter_csv(path: str) -> Tuple[np.ndarray, np.ndarray]:
    arr = np.loadtxt(path, delimiter=",", skiprows=1, ndmin=2)
    if arr.size == 0:
        return np.zeros((0, 1), np.float32), np.zeros((0,), np.int64)
    return arr[:, :-1].astype(np.float32), arr[:, -1].astype(np.int64)


class RawStore:
    """Raw (un-batched) rows per (client, iteration); cached CSV reads."""

    def __init__(self, data_dir: str, dataset: str, num_client: int):
        self.dataset = "MNIST" if dataset.lower() == "mnist" else dataset
        self.dir = os.path.join(data_dir, self.dataset)
        self.num_client = num_client
        self._cache: Dict[Tuple[int, int], Tuple[np.ndarray, np.ndarray]] = {}

    def get(self, c: int, t: int) -> Tuple[np.ndarray, np.ndarray]:
        key = (c, t)
        if key not in self._cache:
            path = os.path.join(self.dir, f"client_{c}_iter_{t}.csv")
            self._cache[key] = _read_client_iter_csv(path)
        return self._cache[key]

    def put(self, c: int, t: int, x: np.ndarray, y: np.ndarray) -> None:
        """In-memory injection (synthetic benchmarks / tests — no CSV IO)."""
        self._cache[(c, t)] = (x.astype(np.float32), y.astype(np.int64))


def resolve_retrain_rows(store: RawStore, c: int, curr_iter: int,
                         method: str, rng: np.random.Generator
                         ) -> Tuple[np.ndarray, np.ndarray]:
    """Assemble one client's training rows per the retrain DSL."""
    xs: List[np.ndarray] = []
    ys: List[np.ndarray] = []

    def add(t: int, mult: int = 1):
        x, y = store.get(c, t)
        for _ in range(mult):
            xs.append(x)
            ys.append(y)

    if method == "all":
        for t in range(curr_iter + 1):
            add(t)
    elif method.startswith("win-"):
        w = int(method[len("win-"):])
        for t in range(max(0, curr_iter - w + 1), curr_iter + 1):
            add(t)
    elif method.startswith("weight-"):
        kind = method[len("weight-"):]
        for t in range(curr_iter + 1):
            add(t, (t + 1) if kind == "linear" else 2 ** t)
    elif method.startswith("sel-"):
        spec = method[len("sel-"):]
        for t in spec.split(","):
            if t != "":
                add(int(t))
    elif method.startswith("clientsel-"):
        table = json.loads(method[len("clientsel-"):])
        for t in table[c]:
            add(int(t))
    elif method.startswith("poisson"):
        x, y = store.get(c, curr_iter)
        w = rng.poisson(1.0, size=len(y))
        if w.sum() != 0:
            idx = rng.choice(len(y), size=len(y), replace=True,
                             p=w / w.sum())
            xs.append(x[idx])
            ys.append(y[idx])
        else:
            xs.append(x)
            ys.append(y)
    else:
        raise NameError(method)

    if not xs:
        d = store.get(c, 0)[0].shape[1]
        return np.zeros((0, d), np.float32), np.zeros((0,), np.int64)
    return np.concatenate(xs, 0).astype(np.float32), np.concatenate(ys, 0)


@dataclass
class RetrainView:
    """One dataset in the reference's 9-tuple sense: per-client train
    segments (retrain-DSL assembled) + per-client test segments (iter t+1)."""
    train: Dict[int, Segment]
    test: Dict[int, Segment]
    train_num: int
    test_num: int
    class_num: int
    feature_num: int

    def train_n(self, c: int) -> int:
        return self.train[c].n if c in self.train else 0


def load_retrain_data(store: RawStore, curr_iter: int, batch_size: int,
                      method: str, rng: np.random.Generator) -> RetrainView:
    train: Dict[int, Segment] = {}
    test: Dict[int, Segment] = {}
    train_num = test_num = 0
    for c in range(store.num_client):
        x, y = resolve_retrain_rows(store, c, curr_iter, method, rng)
        train_num += len(y)
        train[c] = batchify(x, y, batch_size, rng)
        tx, ty = store.get(c, curr_iter + 1)
        test_num += len(ty)
        test[c] = batchify(tx, ty, batch_size, rng)
    return RetrainView(train, test, train_num, test_num,
                       CLASS_NUM[store.dataset], FEATURE_NUM[store.dataset])


def load_all_data(store: RawStore, curr_iter: int, batch_size: int,
                  rng: np.random.Generator) -> List[List[Segment]]:
    """all_data[c][t] = batched history segment (reference load_all_data_sea)."""
    out = []
    for c in range(store.num_client):
        row = []
        for t in range(curr_iter + 1):
            x, y = store.get(c, t)
            row.append(batchify(x, y, batch_size, rng))
        out.append(row)
    return out


class ClientData:
    """Convenience bundle the engine passes around."""

    def __init__(self, store: RawStore, curr_iter: int, batch_size: int,
                 seed: int):
        self.store = store
        self.curr_iter = curr_iter
        self.batch_size = batch_size
        self.rng = np.random.default_rng(seed)
        self.all_data = load_all_data(store, curr_iter, batch_size, self.rng)

    def view(self, method: str) -> RetrainView:
        return load_retrain_data(self.store, self.curr_iter, self.batch_size,
                                 method, self.rng)


class DriftDataset:
    """Top-level handle: raw store + metadata. For fmow, the reference's
    lazy partition layout (partitions/{P}/client_{c}_iter_{t}.csv index
    files + a feature backing store) is consumed directly when present
    on disk (data/real.py:FmowIndexStore); otherwise the synthetic twin
    generates the same shape."""

    def __init__(self, data_dir: str, dataset: str, num_client: int,
                 partition: str = "A"):
        if dataset == "fmow":
            from .real import FmowIndexStore
            if FmowIndexStore.layout_present(data_dir, partition):
                self.store = FmowIndexStore(data_dir, partition,
                                            num_client)
                self.dataset = "fmow"
                self.feature_num = int(self.store.features.shape[1])
                self.class_num = int(self.store.labels.max()) + 1
                return
        self.store = RawStore(data_dir, dataset, num_client)
        self.dataset = self.store.dataset
        self.feature_num = FEATURE_NUM[self.dataset]
        self.class_num = CLASS_NUM[self.dataset]
