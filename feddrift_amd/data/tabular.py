"""Tabular dataset utilities: generic CSV loading, standardization,
horizontal (client) partitioning and vertical (feature) splitting.

Covers the reference's tabular loaders as one generic module instead of
one copy-pasted loader per dataset: UCI SUSY / Room-Occupancy
(fedml_api/data_preprocessing/UCI/), lending_club
(lending_club_loan/loan_preprocessing.py) and the NUS-WIDE two-party
split used by vertical FL (NUS_WIDE/nus_wide_dataset.py).  Those all
reduce to: read a feature matrix + label column, standardize, shard
rows across clients (horizontal FL) or shard columns across parties
(vertical FL).  Offline synthetic twins of the same shape are provided
for tests and benchmarks.
"""

from __future__ import annotations

import csv
from typing import Dict, List, Sequence, Tuple

import numpy as np
import torch

from .partition import partition


def load_csv(path: str, label_col: int = -1, skip_header: bool = True,
             ) -> Tuple[np.ndarray, np.ndarray]:
    """Read a numeric CSV into (features, labels). Non-float cells in the
    label column are mapped to categorical ids in order of appearance."""
    rows = []
    with open(path) as f:
        r = csv.reader(f)
        for i, row in enumerate(r):
            if skip_header and i == 0:
                try:
                    [float(v) for v in row]
                except ValueError:
                    continue
            rows.append(row)
    ncol = len(rows[0])
    lc = label_col % ncol
    label_map: Dict[str, int] = {}
    xs, ys = [], []
    for row in rows:
        feats = [float(v) for j, v in enumerate(row) if j != lc]
        lab = row[lc]
        try:
            ys.append(float(lab))
        except ValueError:
            ys.append(label_map.setdefault(lab, len(label_map)))
        xs.append(feats)
    return np.asarray(xs, dtype=np.float32), np.asarray(ys, dtype=np.float32)


def standardize(x: np.ndarray, eps: float = 1e-8,
                ) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    mu = x.mean(axis=0)
    sd = x.std(axis=0) + eps
    return (x - mu) / sd, mu, sd


def horizontal_shards(x: np.ndarray, y: np.ndarray, n_clients: int,
                      mode: str = "homo", alpha: float = 0.5,
                      seed: int = 0,
                      ) -> Dict[int, Tuple[torch.Tensor, torch.Tensor]]:
    """Row-sharded federated view (homo or Dirichlet label-skew via
    data/partition.py — the same machinery the image loaders use)."""
    idx_map = partition(mode, y.astype(np.int64), n_clients,
                        alpha=alpha, seed=seed)
    return {c: (torch.as_tensor(x[idx]), torch.as_tensor(y[idx]))
            for c, idx in idx_map.items()}


def vertical_split(x: np.ndarray, parts: Sequence[int],
                   ) -> List[np.ndarray]:
    """Column split across parties for vertical FL (NUS-WIDE style:
    party A holds one modality's features, party B the other's).
    `parts` gives each party's feature count; must sum to n_features."""
    assert sum(parts) == x.shape[1], (sum(parts), x.shape[1])
    out, off = [], 0
    for p in parts:
        out.append(x[:, off:off + p])
        off += p
    return out


def synthetic_susy(n: int = 4000, seed: int = 0,
                   ) -> Tuple[np.ndarray, np.ndarray]:
    """SUSY-shaped: 18 continuous physics-like features, binary label from
    a nonlinear rule on a few 'low-level' features + noise."""
    rng = np.random.default_rng(seed)
    x = rng.normal(size=(n, 18)).astype(np.float32)
    margin = x[:, 0] * x[:, 1] + 0.5 * x[:, 2] ** 2 - x[:, 3] - 0.2
    y = (margin + 0.3 * rng.normal(size=n) > 0).astype(np.float32)
    return x, y


def synthetic_lending(n: int = 4000, seed: int = 0,
                      ) -> Tuple[np.ndarray, np.ndarray]:
    """lending_club-shaped: mixed-scale loan features, default-probability
    label via a logistic rule (binary)."""
    rng = np.random.default_rng(seed)
    amount = rng.lognormal(9.0, 0.6, size=n)
    rate = rng.uniform(0.05, 0.3, size=n)
    dti = rng.uniform(0, 40, size=n)
    fico = rng.normal(690, 40, size=n)
    grade = rng.integers(0, 7, size=n)
    x = np.stack([amount / 1e4, rate * 10, dti / 10,
                  (fico - 600) / 100, grade.astype(np.float64)],
                 axis=1).astype(np.float32)
    logit = -1.5 + 2.5 * rate * 10 - 0.8 * (fico - 600) / 100 + 0.05 * dti / 10
    y = (rng.random(n) < 1 / (1 + np.exp(-logit))).astype(np.float32)
    return x, y


def synthetic_nus_wide(n: int = 2000, d_image: int = 634, d_text: int = 1000,
                       n_classes: int = 5, seed: int = 0,
                       ) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """NUS-WIDE-shaped two-party data: (image features, text features,
    label); both modalities correlate with a shared latent class."""
    rng = np.random.default_rng(seed)
    y = rng.integers(0, n_classes, size=n)
    centers_i = rng.normal(size=(n_classes, d_image))
    centers_t = rng.normal(size=(n_classes, d_text))
    xi = (centers_i[y] + rng.normal(scale=2.0, size=(n, d_image))) \
        .astype(np.float32)
    xt = (centers_t[y] + rng.normal(scale=2.0, size=(n, d_text))) \
        .astype(np.float32)
    return xi, xt, y.astype(np.int64)
