"""Federated neural architecture search (FedNAS / DARTS-style).

Counterpart of the reference fedml_api/distributed/fednas +
model/cv/darts: clients hold a supernet whose cells mix candidate
operations with softmax-weighted architecture parameters (alpha); each
round clients take weight steps on their train split and (first-order
DARTS) alpha steps on their validation split; the server averages BOTH
weight and alpha tensors. A compact MLP supernet keeps the search
machinery testable; the search loop is model-agnostic.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn.functional as F
from torch import nn


class MixedOp(nn.Module):
    """Softmax-weighted mixture of candidate ops (DARTS mixed edge)."""

    def __init__(self, d_in: int, d_out: int):
        super().__init__()
        self.ops = nn.ModuleList([
            nn.Sequential(nn.Linear(d_in, d_out), nn.ReLU()),
            nn.Sequential(nn.Linear(d_in, d_out), nn.Tanh()),
            nn.Linear(d_in, d_out),
        ])

    def forward(self, x, alpha_edge):
        w = torch.softmax(alpha_edge, dim=0)
        return sum(wi * op(x) for wi, op in zip(w, self.ops))


class SuperNet(nn.Module):
    def __init__(self, d_in: int, d_hidden: int, n_classes: int,
                 n_cells: int = 2):
        super().__init__()
        dims = [d_in] + [d_hidden] * n_cells
        self.cells = nn.ModuleList(
            [MixedOp(dims[i], dims[i + 1]) for i in range(n_cells)])
        self.head = nn.Linear(d_hidden, n_classes)
        self.alpha = nn.Parameter(1e-3 * torch.randn(n_cells, 3))

    def forward(self, x):
        for i, cell in enumerate(self.cells):
            x = cell(x, self.alpha[i])
        return self.head(x)

    def weight_parameters(self):
        return [p for n, p in self.named_parameters() if n != "alpha"]

    def genotype(self) -> List[int]:
        return self.alpha.argmax(dim=1).tolist()


class FedNAS:
    """First-order DARTS search federated across clients."""

    def __init__(self, n_clients: int, d_in: int, n_classes: int,
                 d_hidden: int = 16, w_lr: float = 0.05,
                 a_lr: float = 0.01, device=torch.device("cpu")):
        self.device = device
        self.global_model = SuperNet(d_in, d_hidden, n_classes).to(device)
        self.clients = [SuperNet(d_in, d_hidden, n_classes).to(device)
                        for _ in range(n_clients)]
        self.w_lr = w_lr
        self.a_lr = a_lr

    def round(self, train_data, val_data, epochs: int = 1):
        gsd = self.global_model.state_dict()
        uploads = []
        for c, model in enumerate(self.clients):
            model.load_state_dict(gsd)
            w_opt = torch.optim.SGD(model.weight_parameters(), lr=self.w_lr)
            a_opt = torch.optim.Adam([model.alpha], lr=self.a_lr)
            xt, yt = train_data[c]
            xv, yv = val_data[c]
            for _ in range(epochs):
                # architecture step on validation (first-order DARTS)
                a_opt.zero_grad()
                F.cross_entropy(model(xv), yv).backward()
                a_opt.step()
                # weight step on train
                w_opt.zero_grad()
                F.cross_entropy(model(xt), yt).backward()
                w_opt.step()
            uploads.append(model.state_dict())
        avg = {k: torch.stack([u[k].float() for u in uploads]).mean(0)
               for k in gsd}
        self.global_model.load_state_dict(avg)

    def genotype(self):
        return self.global_model.genotype()

    @torch.no_grad()
    def evaluate(self, x, y) -> float:
        pred = self.global_model(x).argmax(-1)
        return float((pred == y).float().mean())


class FedNASDarts:
    """FedNAS over the FULL 8-op DARTS cell search space
    (models/darts.py; reference fedml_api/distributed/fednas/ +
    model/cv/darts/model_search.py): clients share a conv supernet,
    take first-order alternating weight/alpha steps, and the server
    averages both — same protocol as the MLP-space FedNAS above, at the
    reference's search-space scale."""

    def __init__(self, n_clients: int, in_ch: int = 3, n_classes: int = 10,
                 c: int = 8, layers: int = 4, w_lr: float = 0.025,
                 a_lr: float = 3e-4, device=torch.device("cpu")):
        from ..models.darts import DartsNetwork
        self.device = device
        self.global_model = DartsNetwork(
            c=c, num_classes=n_classes, layers=layers,
            in_ch=in_ch).to(device)
        self.clients = [
            DartsNetwork(c=c, num_classes=n_classes, layers=layers,
                         in_ch=in_ch).to(device)
            for _ in range(n_clients)]
        self.w_lr = w_lr
        self.a_lr = a_lr

    def round(self, train_data, val_data, epochs: int = 1):
        gsd = self.global_model.state_dict()
        uploads = []
        for c, model in enumerate(self.clients):
            model.load_state_dict(gsd)
            w_opt = torch.optim.SGD(model.weight_parameters(),
                                    lr=self.w_lr, momentum=0.9)
            a_opt = torch.optim.Adam(model.arch_parameters(), lr=self.a_lr,
                                     betas=(0.5, 0.999), weight_decay=1e-3)
            xt, yt = train_data[c]
            xv, yv = val_data[c]
            for _ in range(epochs):
                a_opt.zero_grad()
                F.cross_entropy(model(xv), yv).backward()
                a_opt.step()
                w_opt.zero_grad()
                F.cross_entropy(model(xt), yt).backward()
                w_opt.step()
            uploads.append(model.state_dict())
        avg = {k: torch.stack([u[k].float() for u in uploads]).mean(0)
               for k in gsd}
        self.global_model.load_state_dict(avg)

    def genotype(self):
        return self.global_model.genotype()

    @torch.no_grad()
    def evaluate(self, x, y) -> float:
        self.global_model.eval()
        pred = self.global_model(x).argmax(-1)
        self.global_model.train()
        return float((pred == y).float().mean())
