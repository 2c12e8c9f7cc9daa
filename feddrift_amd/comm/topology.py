"""Topology managers for decentralized FL.

Counterpart of the reference fedml_core/distributed/topology/*: symmetric
ring-with-random-extra-edges and asymmetric random out-neighbor topologies,
as row-stochastic mixing matrices. Pure numpy (the reference uses
networkx); consumed by engine/decentralized.py.
"""

from __future__ import annotations

from typing import List

import numpy as np


class BaseTopologyManager:
    def __init__(self, n: int):
        self.n = n
        self.topology = np.zeros((n, n))

    def get_in_neighbor_idx_list(self, node: int) -> List[int]:
        return [i for i in range(self.n) if self.topology[i][node] > 0]

    def get_out_neighbor_idx_list(self, node: int) -> List[int]:
        return [j for j in range(self.n) if self.topology[node][j] > 0]

    def get_in_neighbor_weights(self, node: int) -> np.ndarray:
        return self.topology[:, node]

    def get_out_neighbor_weights(self, node: int) -> np.ndarray:
        return self.topology[node, :]


class SymmetricTopologyManager(BaseTopologyManager):
    """Ring of degree `neighbor_num` (symmetric), uniform mixing weights."""

    def __init__(self, n: int, neighbor_num: int = 2, seed: int = 0):
        super().__init__(n)
        self.neighbor_num = min(neighbor_num, n - 1)
        self.seed = seed
        self.generate_topology()

    def generate_topology(self) -> None:
        n, k = self.n, self.neighbor_num
        adj = np.eye(n)
        for i in range(n):
            for d in range(1, k // 2 + 1):
                adj[i][(i + d) % n] = 1
                adj[i][(i - d) % n] = 1
        if k % 2 == 1 and n > 2:
            rng = np.random.RandomState(self.seed)
            for i in range(n):
                j = rng.randint(n)
                if j != i:
                    adj[i][j] = adj[j][i] = 1
        # row-normalize to a doubly-stochastic-ish mixing matrix
        self.topology = adj / adj.sum(axis=1, keepdims=True)


class AsymmetricTopologyManager(BaseTopologyManager):
    """Random out-neighbors (directed), row-stochastic weights."""

    def __init__(self, n: int, neighbor_num: int = 2, seed: int = 0):
        super().__init__(n)
        self.neighbor_num = min(neighbor_num, n - 1)
        self.seed = seed
        self.generate_topology()

    def generate_topology(self) -> None:
        n, k = self.n, self.neighbor_num
        rng = np.random.RandomState(self.seed)
        adj = np.eye(n)
        for i in range(n):
            choices = [j for j in range(n) if j != i]
            for j in rng.choice(choices, size=k, replace=False):
                adj[i][j] = 1
        self.topology = adj / adj.sum(axis=1, keepdims=True)
