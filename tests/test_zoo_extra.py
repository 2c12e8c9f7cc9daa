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
