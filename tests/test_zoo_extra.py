"""Extended model zoo + FedGKT + FedNAS."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from feddrift_amd.engine.fedgkt import FedGKT
from feddrift_amd.engine.fednas import FedNAS
from feddrift_amd.models.cv_extra import (MobileNet, densenet121, resnet56,
                                          resnet110)
from feddrift_amd.models.rnn import CharLSTM, StackOverflowRNN
from feddrift_amd.models.zoo import create_model


@pytest.mark.parametrize("name", ["resnet56", "resnet_gn", "mobilenet"])
def test_cifar_models_forward(name):
    m = create_model(name, 10, 3072)
    out = m(torch.rand(2, 3072))
    assert out.shape == (2, 10)


def test_densenet_forward():
    m = densenet121(num_classes=7)
    out = m(torch.rand(1, 3, 64, 64))
    assert out.shape == (1, 7)


def test_resnet110_depth():
    m = resnet110(num_classes=10)
    convs = sum(1 for mod in m.modules()
                if isinstance(mod, torch.nn.Conv2d))
    assert convs >= 109   # 6n+2 with n=18 -> 110 layers (incl. downsamples)


def test_char_lstm_forward_backward():
    m = CharLSTM()
    x = torch.randint(0, 90, (3, 20)).float()
    logits = m(x)
    assert logits.shape == (3, 90)
    F.cross_entropy(logits, torch.randint(0, 90, (3,))).backward()


def test_stackoverflow_rnn_forward():
    m = StackOverflowRNN(vocab_size=100)
    x = torch.randint(0, 100, (2, 12)).float()
    out = m(x)
    assert out.shape == (2, 12, 104)


def test_fedgkt_learns():
    torch.manual_seed(0)
    x = torch.rand(300, 3) * 8
    y = (x[:, 1] + x[:, 2] > 8).long()
    gkt = FedGKT(n_clients=2, d_in=3, d_feat=8, n_classes=2, lr=0.1)
    data = {0: (x[:150], y[:150]), 1: (x[150:], y[150:])}
    for _ in range(40):
        gkt.round(data, epochs=1)
    assert gkt.evaluate(0, x[:150], y[:150]) > 0.8


def test_fednas_search_converges():
    torch.manual_seed(1)
    x = torch.rand(400, 4)
    y = (x[:, 0] + x[:, 1] > 1.0).long()
    nas = FedNAS(n_clients=2, d_in=4, n_classes=2, d_hidden=8, w_lr=0.2)
    train = {0: (x[:100], y[:100]), 1: (x[100:200], y[100:200])}
    val = {0: (x[200:300], y[200:300]), 1: (x[300:], y[300:])}
    for _ in range(60):
        nas.round(train, val, epochs=3)
    assert nas.evaluate(x, y) > 0.8
    geno = nas.genotype()
    assert len(geno) == 2 and all(0 <= g < 3 for g in geno)


def test_darts_network_full_space():
    """The full 8-op DARTS search supernet (models/darts.py): 14 mixed
    edges x 8 primitives per cell, reduction cells at 1/3 and 2/3,
    genotype derivation picks 2 input edges per node."""
    from feddrift_amd.models.darts import PRIMITIVES, DartsNetwork
    torch.manual_seed(0)
    assert len(PRIMITIVES) == 8
    net = DartsNetwork(c=4, num_classes=10, layers=3, in_ch=3)
    assert net.alphas_normal.shape == (14, 8)
    x = torch.randn(2, 3, 16, 16)
    out = net(x)
    assert out.shape == (2, 10)
    out.sum().backward()
    assert net.alphas_normal.grad is not None
    gene_n, gene_r = net.genotype()
    assert len(gene_n) == 8 and len(gene_r) == 8   # 2 edges x 4 nodes
    for name, j in gene_n:
        assert name in PRIMITIVES and name != "none"
        assert 0 <= j < 5
    # reduction structure: exactly one cell at layers//3 and 2*layers//3
    reductions = [c.reduction for c in net.cells]
    assert sum(reductions) == 2


def test_fednas_darts_round_moves_alphas():
    from feddrift_amd.engine.fednas import FedNASDarts
    torch.manual_seed(2)
    nas = FedNASDarts(n_clients=2, in_ch=3, n_classes=4, c=4, layers=3,
                      w_lr=0.05, a_lr=0.01)
    x = torch.randn(80, 3, 12, 12)
    y = torch.randint(0, 4, (80,))
    train = {0: (x[:20], y[:20]), 1: (x[20:40], y[20:40])}
    val = {0: (x[40:60], y[40:60]), 1: (x[60:], y[60:])}
    a0 = nas.global_model.alphas_normal.detach().clone()
    w0 = nas.global_model.stem[0].weight.detach().clone()
    nas.round(train, val, epochs=2)
    assert (nas.global_model.alphas_normal - a0).abs().max() > 0
    assert (nas.global_model.stem[0].weight - w0).abs().max() > 0
    acc = nas.evaluate(x, y)
    assert 0.0 <= acc <= 1.0
    gene_n, gene_r = nas.genotype()
    assert len(gene_n) == 8 and len(gene_r) == 8


def test_fedgkt_resnet_split():
    """The reference-scale GKT split: client stem features (16ch) feed
    the bottleneck server; bidirectional KD trains both sides."""
    from feddrift_amd.engine.fedgkt import FedGKTResNet
    from feddrift_amd.models.cv_extra import GKTClientNet, GKTServerNet
    torch.manual_seed(3)
    cl = GKTClientNet(10)
    sv = GKTServerNet(10, n_blocks=2)
    x = torch.randn(4, 3, 16, 16)
    logits, feats = cl(x)
    assert logits.shape == (4, 10) and feats.shape == (4, 16, 16, 16)
    assert sv(feats).shape == (4, 10)

    # tiny learnable task: class = dominant color channel
    n = 60
    y = torch.randint(0, 3, (n,))
    xs = torch.randn(n, 3, 8, 8) * 0.1
    for i in range(n):
        xs[i, y[i]] += 1.0
    gkt = FedGKTResNet(n_clients=2, n_classes=3, client_blocks=1,
                       server_blocks=1, lr=0.05)
    data = {0: (xs[:30], y[:30]), 1: (xs[30:], y[30:])}
    first = None
    for r in range(8):
        gkt.round(data, epochs=2)
        acc = gkt.evaluate(0, xs[:30], y[:30])
        if first is None:
            first = acc
    assert acc > max(0.5, first - 0.1), (first, acc)


def test_multiparty_vfl_learns():
    """Reference-shape vertical FL: 1 guest + 3 hosts over disjoint
    feature slices of a linearly separable binary task; only logits and
    logit-gradients cross party boundaries."""
    from feddrift_amd.engine.vfl import MultiPartyVFL
    torch.manual_seed(4)
    n = 400
    x = torch.randn(n, 12)
    w = torch.randn(12)
    y = ((x @ w) > 0).long()
    slices = [x[:, :3], x[:, 3:6], x[:, 6:9], x[:, 9:]]
    vfl = MultiPartyVFL(d_guest=3, host_dims=[3, 3, 3], lr=0.1)
    for epoch in range(60):
        vfl.train_step(slices[0], slices[1:], y)
    acc, auc = vfl.evaluate(slices[0], slices[1:], y)
    assert acc > 0.85 and auc > 0.9, (acc, auc)


def test_splitnn_relay_protocol():
    from feddrift_amd.engine.splitnn import SplitNNRelay
    import torch.nn as tnn
    torch.manual_seed(5)
    n = 240
    x = torch.randn(n, 6)
    y = ((x[:, 0] + x[:, 1]) > 0).long()
    clients = [tnn.Sequential(tnn.Linear(6, 8), tnn.ReLU())
               for _ in range(3)]
    server = tnn.Linear(8, 2)
    nn_ = SplitNNRelay(clients, server, lr=0.1)
    data = {c: (x[c * 80:(c + 1) * 80], y[c * 80:(c + 1) * 80])
            for c in range(3)}
    first = None
    for lap in range(8):
        losses = nn_.run_lap(data, epochs_per_node=2)
        if first is None:
            first = np.mean(list(losses.values()))
    accs = [nn_.evaluate(c, *data[c]) for c in range(3)]
    assert np.mean(list(losses.values())) < first
    assert min(accs) > 0.7, accs
