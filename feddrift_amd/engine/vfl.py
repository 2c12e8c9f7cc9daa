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


class VFLTower(nn.Module):
    """Feature extractor + scalar classifier head — the per-party model
    of the reference classical_vertical_fl (guest_trainer.py:36-45,
    host_trainer.py:26-37)."""

    def __init__(self, d_in: int, d_hidden: int):
        super().__init__()
        self.extractor = nn.Sequential(nn.Linear(d_in, d_hidden),
                                       nn.ReLU())
        self.classifier = nn.Linear(d_hidden, 1)

    def forward(self, x):
        return self.classifier(self.extractor(x))


class MultiPartyVFL:
    """Classical vertical FL at the reference's working shape
    (fedml_api/distributed/classical_vertical_fl): ONE guest holding the
    binary labels + its feature slice, K HOSTS holding feature slices
    only. Each party runs extractor+classifier producing scalar logits;
    the guest sums its logits with every host's, takes
    BCEWithLogitsLoss, and sends each host ONLY dL/d(host logits)
    (guest_trainer.py:73-110). SGD(momentum=0.9, weight_decay=0.01) per
    the reference; prequential AUC/accuracy via sklearn."""

    def __init__(self, d_guest: int, host_dims, d_hidden: int = 16,
                 lr: float = 0.05, device=torch.device("cpu")):
        self.device = device
        self.guest = VFLTower(d_guest, d_hidden).to(device)
        self.hosts = [VFLTower(d, d_hidden).to(device) for d in host_dims]
        mk = lambda m: torch.optim.SGD(m.parameters(), lr=lr,  # noqa: E731
                                       momentum=0.9, weight_decay=0.01)
        self.opt_g = mk(self.guest)
        self.opt_h = [mk(h) for h in self.hosts]
        self.crit = nn.BCEWithLogitsLoss()

    def train_step(self, x_guest, host_xs, y) -> float:
        self.opt_g.zero_grad()
        z_total = self.guest(x_guest).squeeze(-1)
        host_outs = []
        for h, opt, xh in zip(self.hosts, self.opt_h, host_xs):
            opt.zero_grad()
            zh = h(xh).squeeze(-1)
            host_outs.append(zh)
            z_total = z_total + zh.detach()    # only logits cross the wire
        z_wire = z_total.detach().requires_grad_(True)
        loss = self.crit(z_wire, y.float())
        loss.backward()
        g = z_wire.grad                        # dL/d(summed logits)
        # guest backprop through its own tower
        self.guest(x_guest).squeeze(-1).backward(g)
        self.opt_g.step()
        # each host receives ONLY the logit gradient
        for zh, opt in zip(host_outs, self.opt_h):
            zh.backward(g)
            opt.step()
        return float(loss.item())

    @torch.no_grad()
    def scores(self, x_guest, host_xs):
        z = self.guest(x_guest).squeeze(-1)
        for h, xh in zip(self.hosts, host_xs):
            z = z + h(xh).squeeze(-1)
        return torch.sigmoid(z)

    def evaluate(self, x_guest, host_xs, y):
        from sklearn.metrics import accuracy_score, roc_auc_score
        p = self.scores(x_guest, host_xs).cpu().numpy()
        yy = y.cpu().numpy()
        acc = accuracy_score(yy, p > 0.5)
        auc = roc_auc_score(yy, p) if len(set(yy.tolist())) > 1 else 0.5
        return acc, auc
