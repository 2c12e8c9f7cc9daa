"""Split learning: client-side lower layers, server-side upper layers.

Counterpart of the reference fedml_api/distributed/split_nn: each client
computes activations of its local partial model, ships them to the server,
which finishes the forward, computes the loss, and returns activation
gradients. Here both halves live on one GPU and the "wire" is a tensor
hand-off, but the computational structure (alternating clients, gradient
cut at the split layer) is the same.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn.functional as F
from torch import nn


class SplitNN:
    def __init__(self, client_models: List[nn.Module],
                 server_model: nn.Module, lr: float = 0.01,
                 device: torch.device = torch.device("cpu")):
        self.client_models = [m.to(device) for m in client_models]
        self.server_model = server_model.to(device)
        self.device = device
        self.client_opts = [torch.optim.SGD(m.parameters(), lr=lr)
                            for m in self.client_models]
        self.server_opt = torch.optim.SGD(self.server_model.parameters(),
                                          lr=lr)

    def train_step(self, client_idx: int, x: torch.Tensor,
                   y: torch.Tensor) -> float:
        cm = self.client_models[client_idx]
        copt = self.client_opts[client_idx]
        copt.zero_grad()
        self.server_opt.zero_grad()
        # client forward -> cut -> server forward
        acts = cm(x)
        acts_cut = acts.detach().requires_grad_(True)
        logits = self.server_model(acts_cut)
        loss = F.cross_entropy(logits, y)
        loss.backward()                      # server grads + grad at cut
        acts.backward(acts_cut.grad)         # client backward from the cut
        self.server_opt.step()
        copt.step()
        return float(loss.item())

    @torch.no_grad()
    def evaluate(self, client_idx: int, x: torch.Tensor,
                 y: torch.Tensor) -> float:
        logits = self.server_model(self.client_models[client_idx](x))
        return float((logits.argmax(-1) == y).float().mean().item())


class SplitNNRelay(SplitNN):
    """The reference's relay protocol (fedml_api/distributed/split_nn):
    clients form a ring and pass a turn token; the active client runs
    `epochs_per_node` local epochs against the SHARED server-side model
    (activations up, cut-layer gradients down), then a validation pass,
    then hands the token to node_right (client.py:12-14,
    client_manager.py semaphore flow). One lap = every client's turn."""

    def run_lap(self, data, epochs_per_node: int = 1,
                batch_size: int = 32) -> dict:
        losses = {}
        n_clients = len(self.client_models)
        for turn in range(n_clients):       # ring order 0 -> 1 -> ... -> 0
            x, y = data[turn]
            n = x.shape[0]
            loss_sum = steps = 0
            for _ in range(epochs_per_node):
                for i in range(0, n, batch_size):
                    loss_sum += self.train_step(turn, x[i:i + batch_size],
                                                y[i:i + batch_size])
                    steps += 1
            losses[turn] = loss_sum / max(steps, 1)
        return losses
