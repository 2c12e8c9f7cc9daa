"""Numerics tests: the batched ops must match plain torch autograd +
torch.optim exactly (the reference's eager training loop semantics,
FedAvgEnsTrainer.py:65-85)."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from feddrift_amd.models.packed import MLPSpec, PackedMLP, spec_for
from feddrift_amd.models.zoo import FeedForwardNN, LogisticRegression
from feddrift_amd.ops import mlp_torch


def _make_data(n, d, o, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, d, generator=g) * 10
    y = torch.randint(0, o, (n,), generator=g)
    return x, y


def _eager_train(model, opt, x, y, picks, batch):
    crit = torch.nn.CrossEntropyLoss()
    for p in picks:
        xb = x[p * batch:(p + 1) * batch]
        yb = y[p * batch:(p + 1) * batch]
        opt.zero_grad()
        loss = crit(model(xb), yb)
        loss.backward()
        opt.step()


@pytest.mark.parametrize("optname", ["adam", "sgd"])
def test_train_fused_matches_eager_fnn(optname):
    torch.manual_seed(7)
    d, o, h, batch, E = 3, 2, 6, 50, 5
    spec = spec_for("fnn", d, o)
    packer = PackedMLP(spec)
    x, y = _make_data(200, d, o)

    model = FeedForwardNN(d, o, h)
    sd0 = {k: v.clone() for k, v in model.state_dict().items()}
    if optname == "adam":
        opt = torch.optim.Adam(model.parameters(), lr=0.01,
                               weight_decay=0.001, amsgrad=True)
    else:
        opt = torch.optim.SGD(model.parameters(), lr=0.01)
    picks = [1, 0, 3, 2, 1]
    _eager_train(model, opt, x, y, picks, batch)
    ref = packer.flatten(model.state_dict())

    params = packer.flatten(sd0).unsqueeze(0).clone()
    ost = mlp_torch.make_opt_state(optname, 1, spec.n_params, 0.01, 0.001,
                                   "cpu")
    off = torch.tensor([[p * batch for p in picks]])
    ln = torch.full((1, E), batch)
    mlp_torch.train_fused(spec, params, torch.tensor([0]), x, y, off, ln, ost)
    assert torch.allclose(params[0], ref, atol=2e-6, rtol=1e-5), \
        (params[0] - ref).abs().max()


def test_train_fused_matches_eager_lr():
    torch.manual_seed(3)
    d, o, batch, E = 4, 3, 40, 4
    spec = spec_for("lr", d, o)
    packer = PackedMLP(spec)
    x, y = _make_data(160, d, o, seed=5)

    model = LogisticRegression(d, o)
    sd0 = {k: v.clone() for k, v in model.state_dict().items()}
    opt = torch.optim.Adam(model.parameters(), lr=0.02, weight_decay=0.001,
                           amsgrad=True)
    picks = [0, 2, 1, 3]
    _eager_train(model, opt, x, y, picks, batch)
    ref = packer.flatten(model.state_dict())

    params = packer.flatten(sd0).unsqueeze(0).clone()
    ost = mlp_torch.make_opt_state("adam", 1, spec.n_params, 0.02, 0.001,
                                   "cpu")
    off = torch.tensor([[p * batch for p in picks]])
    ln = torch.full((1, E), batch)
    mlp_torch.train_fused(spec, params, torch.tensor([0]), x, y, off, ln, ost)
    assert torch.allclose(params[0], ref, atol=2e-6, rtol=1e-5)


def test_train_fused_multi_pair_independent():
    """Pairs in one fused call must equal pairs trained one at a time."""
    torch.manual_seed(11)
    d, o, batch, E, G = 3, 2, 30, 3, 4
    spec = spec_for("fnn", d, o)
    x, y = _make_data(300, d, o, seed=2)
    base = torch.randn(G, spec.n_params) * 0.1
    offs = torch.randint(0, 8, (G, E)) * batch
    lens = torch.full((G, E), batch)

    fused = base.clone()
    ost = mlp_torch.make_opt_state("adam", G, spec.n_params, 0.01, 0.001,
                                   "cpu")
    mlp_torch.train_fused(spec, fused, torch.arange(G), x, y, offs, lens, ost)

    for g in range(G):
        solo = base[g:g + 1].clone()
        ost1 = mlp_torch.make_opt_state("adam", 1, spec.n_params, 0.01,
                                        0.001, "cpu")
        mlp_torch.train_fused(spec, solo, torch.tensor([0]), x, y,
                              offs[g:g + 1], lens[g:g + 1], ost1)
        assert torch.allclose(fused[g], solo[0], atol=1e-6)


def test_train_fused_skip_step_preserves_adam_t():
    """A zero-length step must not advance optimizer state
    (reference Exp trainer `continue`, FedAvgEnsTrainerExp.py:73-74)."""
    torch.manual_seed(1)
    spec = spec_for("fnn", 3, 2)
    x, y = _make_data(100, 3, 2)
    params = torch.randn(2, spec.n_params) * 0.1
    ost = mlp_torch.make_opt_state("adam", 2, spec.n_params, 0.01, 0.001,
                                   "cpu")
    off = torch.tensor([[0, 50], [0, 0]])
    ln = torch.tensor([[50, 50], [50, 0]])   # pair 1 skips step 2
    mlp_torch.train_fused(spec, params, torch.arange(2), x, y, off, ln, ost)
    assert ost["t"][0] == 2
    assert ost["t"][1] == 1


def test_eval_tasks_matches_eager():
    torch.manual_seed(4)
    spec = spec_for("fnn", 3, 2)
    model = FeedForwardNN(3, 2, 6)
    packer = PackedMLP(spec)
    params = packer.flatten(model.state_dict()).unsqueeze(0)
    x, y = _make_data(120, 3, 2, seed=9)

    logits = model(x)
    pred = logits.argmax(-1)
    ref_correct = (pred == y).sum().item()
    ref_loss = F.cross_entropy(logits, y, reduction="sum").item()
    prob = torch.softmax(logits, -1)
    ref_mse = ((1 - prob.gather(1, y.unsqueeze(1)).squeeze(1)) ** 2).sum()

    # two windows of different lengths accumulated into one task
    correct, total, loss, mse = mlp_torch.eval_tasks(
        spec, params, x, y,
        task_row=torch.tensor([0, 0]), task_id=torch.tensor([0, 0]),
        win_off=torch.tensor([0, 70]), win_len=torch.tensor([70, 50]),
        n_tasks=1, want_mse=True)
    assert correct[0].item() == ref_correct
    assert total[0].item() == 120
    assert abs(loss[0].item() - ref_loss) < 1e-3
    assert abs(mse[0].item() - ref_mse.item()) < 1e-3


def test_ens_vote_eval_hard():
    """AUE weighted argmax vote (FedAvgEnsAggregatorAue.py:256-283)."""
    torch.manual_seed(5)
    spec = spec_for("fnn", 3, 2)
    params = torch.randn(3, spec.n_params) * 0.5
    x, y = _make_data(64, 3, 2, seed=1)
    w = torch.tensor([0.5, 0.3, 0.2])

    logits = mlp_torch.forward_logits(spec, params, x.unsqueeze(1).expand(
        -1, 3, -1).transpose(0, 1).contiguous())
    preds = logits.argmax(-1)       # [3, 64]
    votes = torch.zeros(64, 2)
    for m in range(3):
        for i in range(64):
            votes[i, preds[m, i]] += w[m]
    ref_correct = (votes.argmax(-1) == y).sum().item()

    correct, total = mlp_torch.ens_vote_eval(
        spec, params, w, x, y, [(0, 64)], mode="hard")
    assert correct == ref_correct and total == 64


def test_confusion_tasks():
    torch.manual_seed(6)
    spec = spec_for("fnn", 3, 2)
    params = torch.randn(1, spec.n_params) * 0.5
    x, y = _make_data(90, 3, 2, seed=3)
    logits = mlp_torch.forward_logits(spec, params, x.unsqueeze(0)).squeeze(0)
    pred = logits.argmax(-1)
    ref = np.zeros((2, 2))
    for i in range(90):
        ref[y[i], pred[i]] += 1
    A = mlp_torch.confusion_tasks(
        spec, params, x, y, torch.tensor([0]), torch.tensor([0]),
        torch.tensor([0]), torch.tensor([90]), 1, 2)
    assert np.allclose(A[0].cpu().numpy(), ref)


def test_kue_mask_applied():
    torch.manual_seed(8)
    spec = spec_for("fnn", 3, 2)
    params = torch.randn(1, spec.n_params) * 0.5
    x, y = _make_data(50, 3, 2)
    mask = torch.tensor([[1.0, 0.0, 1.0]])
    c1, _, _, _ = mlp_torch.eval_tasks(
        spec, params, x, y, torch.tensor([0]), torch.tensor([0]),
        torch.tensor([0]), torch.tensor([50]), 1, x_mask=mask)
    c2, _, _, _ = mlp_torch.eval_tasks(
        spec, params, x * mask, y, torch.tensor([0]), torch.tensor([0]),
        torch.tensor([0]), torch.tensor([50]), 1)
    assert c1[0].item() == c2[0].item()
