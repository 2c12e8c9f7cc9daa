"""Decentralized (server-less) FL: DSGD neighbor gossip + hierarchical FL.

Counterparts of the reference fedml_api/distributed/decentralized_framework
(neighbor gossip over a topology manager) and
fedml_api/standalone/hierarchical_fl (two-level group aggregation), built
on the engine's batched primitives: every worker's model is a row of a
flat [W, P] tensor; one gossip round = batched local training (the fused
HIP op / torch path) followed by a MIXING-MATRIX multiply — the reference's
per-neighbor message exchange becomes a single [W, W] x [W, P] GEMM
(rocBLAS), and group aggregation a segment mean.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..comm.topology import BaseTopologyManager
from ..models.packed import MLPSpec
from ..ops import mlp_torch


class DecentralizedDSGD:
    """n workers, each with private data windows; per round: E local SGD /
    Adam steps then weight mixing with topology neighbors."""

    def __init__(self, spec: MLPSpec, n_workers: int,
                 topology: BaseTopologyManager, init_flat: torch.Tensor,
                 x_arena: torch.Tensor, y_arena: torch.Tensor,
                 windows_per_worker: List[List[Tuple[int, int]]],
                 lr: float = 0.05, optimizer: str = "sgd",
                 epochs: int = 1, device: Optional[torch.device] = None,
                 seed: int = 0):
        self.spec = spec
        self.n = n_workers
        self.device = device or x_arena.device
        self.params = init_flat.unsqueeze(0).repeat(n_workers, 1) \
            .to(self.device)
        self.mix = torch.as_tensor(topology.topology, dtype=torch.float32,
                                   device=self.device)
        self.x = x_arena
        self.y = y_arena
        self.windows = windows_per_worker
        self.epochs = epochs
        self.opt = mlp_torch.make_opt_state(optimizer, n_workers,
                                            spec.n_params, lr, 0.0,
                                            self.device)
        self.rng = np.random.default_rng(seed)

    def round(self) -> None:
        E = self.epochs
        offs = np.zeros((self.n, E), dtype=np.int64)
        lens = np.zeros((self.n, E), dtype=np.int64)
        for w in range(self.n):
            wins = self.windows[w]
            picks = self.rng.integers(0, len(wins), size=E)
            offs[w] = [wins[p][0] for p in picks]
            lens[w] = [wins[p][1] for p in picks]
        mlp_torch.train_fused(
            self.spec, self.params,
            torch.arange(self.n, device=self.device),
            self.x, self.y,
            torch.as_tensor(offs, device=self.device),
            torch.as_tensor(lens, device=self.device), self.opt)
        # gossip mixing: one GEMM replaces the reference's per-neighbor
        # message send/receive cycle
        self.params = self.mix @ self.params

    def consensus_distance(self) -> float:
        mean = self.params.mean(dim=0, keepdim=True)
        return float((self.params - mean).norm(dim=1).max())


class PushSumDSGD:
    """Stochastic gradient push over a DIRECTED topology (reference
    fedml_api/standalone/decentralized: push-sum gossip; asymmetric
    out-neighbor graphs from AsymmetricTopologyManager).

    Each worker keeps a numerator row x and a scalar weight w; the
    de-biased estimate z = x / w is what local SGD steps update.  Both
    x and w are mixed with a COLUMN-stochastic matrix each round, so
    z converges to the network average even though the directed graph
    is not doubly stochastic (the symmetric-mixing assumption DSGD
    needs).  As with DSGD above, the whole per-neighbor push/receive
    cycle of the reference is one [W, W] x [W, P] GEMM on rocBLAS.
    """

    def __init__(self, spec: MLPSpec, n_workers: int,
                 topology: BaseTopologyManager, init_flat: torch.Tensor,
                 x_arena: torch.Tensor, y_arena: torch.Tensor,
                 windows_per_worker: List[List[Tuple[int, int]]],
                 lr: float = 0.05, epochs: int = 1,
                 device: Optional[torch.device] = None, seed: int = 0):
        self.spec = spec
        self.n = n_workers
        self.device = device or x_arena.device
        self.num = init_flat.unsqueeze(0).repeat(n_workers, 1) \
            .to(self.device)
        self.w = torch.ones(n_workers, device=self.device)
        # column-stochastic mixing: node j splits its mass equally over
        # its out-neighbors (incl. itself) -> column j of A sums to 1
        out = torch.as_tensor(topology.topology, dtype=torch.float32,
                              device=self.device)
        out = (out > 0).float()
        out.fill_diagonal_(1.0)
        self.mix = out / out.sum(dim=0, keepdim=True)
        self.x = x_arena
        self.y = y_arena
        self.windows = windows_per_worker
        self.epochs = epochs
        self.opt = mlp_torch.make_opt_state("sgd", n_workers,
                                            spec.n_params, lr, 0.0,
                                            self.device)
        self.rng = np.random.default_rng(seed)

    def round(self) -> None:
        E = self.epochs
        offs = np.zeros((self.n, E), dtype=np.int64)
        lens = np.zeros((self.n, E), dtype=np.int64)
        for k in range(self.n):
            wins = self.windows[k]
            picks = self.rng.integers(0, len(wins), size=E)
            offs[k] = [wins[p][0] for p in picks]
            lens[k] = [wins[p][1] for p in picks]
        z = self.num / self.w.unsqueeze(1)
        mlp_torch.train_fused(
            self.spec, z, torch.arange(self.n, device=self.device),
            self.x, self.y,
            torch.as_tensor(offs, device=self.device),
            torch.as_tensor(lens, device=self.device), self.opt)
        # push step: re-scale the stepped estimate back into numerator
        # space, then one column-stochastic mixing GEMM for both x and w
        self.num = self.mix @ (z * self.w.unsqueeze(1))
        self.w = self.mix @ self.w

    def estimates(self) -> torch.Tensor:
        return self.num / self.w.unsqueeze(1)

    def consensus_distance(self) -> float:
        z = self.estimates()
        mean = z.mean(dim=0, keepdim=True)
        return float((z - mean).norm(dim=1).max())


class HierarchicalFL:
    """Two-level FedAvg: clients -> group aggregation every round,
    groups -> global aggregation every `group_comm_round` rounds
    (reference fedml_api/standalone/hierarchical_fl)."""

    def __init__(self, spec: MLPSpec, groups: Sequence[Sequence[int]],
                 init_flat: torch.Tensor, x_arena: torch.Tensor,
                 y_arena: torch.Tensor,
                 windows_per_client: List[List[Tuple[int, int]]],
                 lr: float = 0.05, epochs: int = 1,
                 group_comm_round: int = 2,
                 device: Optional[torch.device] = None, seed: int = 0):
        self.spec = spec
        self.groups = [list(g) for g in groups]
        self.n_clients = sum(len(g) for g in self.groups)
        self.device = device or x_arena.device
        self.group_params = init_flat.unsqueeze(0).repeat(
            len(self.groups), 1).to(self.device)
        self.client_params = torch.zeros(self.n_clients, spec.n_params,
                                         device=self.device)
        self.x = x_arena
        self.y = y_arena
        self.windows = windows_per_client
        self.epochs = epochs
        self.group_comm_round = group_comm_round
        self.lr = lr
        self.opt = mlp_torch.make_opt_state("sgd", self.n_clients,
                                            spec.n_params, lr, 0.0,
                                            self.device)
        self.rng = np.random.default_rng(seed)
        self._round = 0

    def round(self) -> None:
        # push group model to members, local train, group-average
        for gi, members in enumerate(self.groups):
            for c in members:
                self.client_params[c] = self.group_params[gi]
        E = self.epochs
        offs = np.zeros((self.n_clients, E), dtype=np.int64)
        lens = np.zeros((self.n_clients, E), dtype=np.int64)
        for c in range(self.n_clients):
            wins = self.windows[c]
            picks = self.rng.integers(0, len(wins), size=E)
            offs[c] = [wins[p][0] for p in picks]
            lens[c] = [wins[p][1] for p in picks]
        mlp_torch.train_fused(
            self.spec, self.client_params,
            torch.arange(self.n_clients, device=self.device),
            self.x, self.y,
            torch.as_tensor(offs, device=self.device),
            torch.as_tensor(lens, device=self.device), self.opt)
        for gi, members in enumerate(self.groups):
            self.group_params[gi] = self.client_params[members].mean(dim=0)
        self._round += 1
        if self._round % self.group_comm_round == 0:
            self.group_params[:] = self.group_params.mean(dim=0,
                                                          keepdim=True)
