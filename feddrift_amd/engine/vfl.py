"""Classical two-party vertical federated learning.

Counterpart of the reference fedml_api/distributed/classical_vertical_fl /
standalone VFL: a guest party holds labels + its feature slice, a host
party holds only a feature slice; both train linear towers whose logits
are summed. Only logits and logit-gradients cross the party boundary.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn


class VFLParty(nn.Module):
    def __init__(self, d_in: int, d_out: int):
        super().__init__()
        self.tower = nn.Linear(d_in, d_out)

    def forward(self, x):
        return self.tower(x)


class TwoPartyVFL:
    def __init__(self, d_guest: int, d_host: int, n_classes: int,
                 lr: float = 0.05, device=torch.device("cpu")):
        self.guest = VFLParty(d_guest, n_classes).to(device)
        self.host = VFLParty(d_host, n_classes).to(device)
        self.opt_g = torch.optim.SGD(self.guest.parameters(), lr=lr)
        self.opt_h = torch.optim.SGD(self.host.parameters(), lr=lr)

    def train_step(self, x_guest: torch.Tensor, x_host: torch.Tensor,
                   y: torch.Tensor) -> float:
        self.opt_g.zero_grad()
        self.opt_h.zero_grad()
        zg = self.guest(x_guest)
        zh = self.host(x_host)
        zh_wire = zh.detach().requires_grad_(True)   # the party boundary
        loss = F.cross_entropy(zg + zh_wire, y)      # guest holds labels
        loss.backward()
        zh.backward(zh_wire.grad)                    # host gets only dL/dz
        self.opt_g.step()
        self.opt_h.step()
        return float(loss.item())

    @torch.no_grad()
    def predict(self, x_guest, x_host):
        return (self.guest(x_guest) + self.host(x_host)).argmax(-1)
