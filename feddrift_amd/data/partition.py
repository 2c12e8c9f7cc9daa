"""Non-drift dataset partitioning for the classic (one-shot) FedAvg
benchmarks.

Counterpart of the reference's per-dataset partitioners
(fedml_api/data_preprocessing/cifar10/data_loader.py:113-162 — 'homo'
uniform split and 'hetero' Dirichlet(alpha) label-skew split; the LEAF
json loaders use natural client splits). Returns per-client row indices.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np


def partition_homo(n_samples: int, n_clients: int,
                   rng: np.random.Generator) -> Dict[int, np.ndarray]:
    idx = rng.permutation(n_samples)
    return {c: np.sort(part) for c, part in
            enumerate(np.array_split(idx, n_clients))}


def partition_dirichlet(labels: np.ndarray, n_clients: int, alpha: float,
                        rng: np.random.Generator,
                        min_size_req: int = 10) -> Dict[int, np.ndarray]:
    """Label-skew Dirichlet partition (reference 'hetero',
    cifar10/data_loader.py:126-148): per class, split its samples across
    clients by Dirichlet(alpha) proportions, rebalancing until every client
    has at least min_size_req samples."""
    n = len(labels)
    classes = np.unique(labels)
    min_size = 0
    while min_size < min_size_req:
        idx_batch: List[List[int]] = [[] for _ in range(n_clients)]
        for k in classes:
            idx_k = np.where(labels == k)[0]
            rng.shuffle(idx_k)
            p = rng.dirichlet(np.repeat(alpha, n_clients))
            # balance: zero out clients already at the average size
            p = np.array([q * (len(b) < n / n_clients)
                          for q, b in zip(p, idx_batch)])
            p = p / p.sum()
            cuts = (np.cumsum(p) * len(idx_k)).astype(int)[:-1]
            for c, part in enumerate(np.split(idx_k, cuts)):
                idx_batch[c].extend(part.tolist())
        min_size = min(len(b) for b in idx_batch)
    return {c: np.sort(np.asarray(b)) for c, b in enumerate(idx_batch)}


def partition(method: str, labels: np.ndarray, n_clients: int,
              alpha: float = 0.5, seed: int = 0) -> Dict[int, np.ndarray]:
    rng = np.random.default_rng(seed)
    if method == "homo":
        return partition_homo(len(labels), n_clients, rng)
    if method in ("hetero", "dirichlet"):
        return partition_dirichlet(labels, n_clients, alpha, rng)
    raise NameError(method)
