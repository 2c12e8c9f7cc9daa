"""Secure aggregation, split learning, vertical FL equivalents."""

import numpy as np
import torch
import torch.nn.functional as F
from torch import nn

from feddrift_amd.comm.secure_agg import mask_uploads
from feddrift_amd.engine.splitnn import SplitNN
from feddrift_amd.engine.vfl import TwoPartyVFL


def test_secure_agg_masks_cancel():
    torch.manual_seed(0)
    ups = torch.randn(5, 40)
    masked = mask_uploads(ups, clients=[3, 7, 11, 20, 21], base_seed=42)
    # individual uploads are hidden ...
    assert (masked - ups).abs().max() > 0.5
    # ... but the aggregate is exact
    assert torch.allclose(masked.sum(0), ups.sum(0), atol=1e-4)


def test_splitnn_learns():
    torch.manual_seed(1)
    x = torch.rand(400, 3) * 8
    y = (x[:, 1] + x[:, 2] > 8).long()
    clients = [nn.Sequential(nn.Linear(3, 8), nn.ReLU()) for _ in range(2)]
    server = nn.Linear(8, 2)
    sn = SplitNN(clients, server, lr=0.1)
    for step in range(150):
        c = step % 2
        sl = slice(c * 200, (c + 1) * 200)
        sn.train_step(c, x[sl], y[sl])
    assert sn.evaluate(0, x[:200], y[:200]) > 0.8


def test_vfl_two_party_learns():
    torch.manual_seed(2)
    x = torch.rand(500, 4)
    y = (x[:, 0] + x[:, 3] > 1.0).long()
    vfl = TwoPartyVFL(d_guest=2, d_host=2, n_classes=2, lr=0.5)
    for _ in range(200):
        vfl.train_step(x[:, :2], x[:, 2:], y)
    acc = (vfl.predict(x[:, :2], x[:, 2:]) == y).float().mean().item()
    assert acc > 0.85, acc
