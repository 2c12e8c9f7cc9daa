"""Synthetic text (shakespeare/stackoverflow-shaped) and generic tabular
data paths: shape contracts, federated structure, and that the matching
zoo models actually learn from them."""

import numpy as np
import torch
import torch.nn.functional as F

from feddrift_amd.data import tabular, text_synthetic
from feddrift_amd.models.rnn import CharLSTM, StackOverflowRNN


def test_char_sequences_shape_and_noniid():
    data = text_synthetic.char_sequences(3, 40, seq_len=16, vocab=30,
                                         seed=1)
    assert set(data) == {0, 1, 2}
    x, y = data[0]
    assert x.shape == (40, 16) and y.shape == (40,)
    assert x.min() >= 1 and x.max() < 30      # 0 reserved for padding
    # clients draw from different chains -> different marginals
    h0 = torch.bincount(data[0][0].long().flatten(), minlength=30).float()
    h1 = torch.bincount(data[1][0].long().flatten(), minlength=30).float()
    assert (h0 / h0.sum() - h1 / h1.sum()).abs().sum() > 0.3
    # concept permutation changes the distribution for the same client
    alt = text_synthetic.char_sequences(1, 40, seq_len=16, vocab=30,
                                        seed=1, concept=3)
    # (a permutation of the chain moves the stationary distribution less
    # than an independent chain does — weaker threshold)
    ha = torch.bincount(alt[0][0].long().flatten(), minlength=30).float()
    assert (h0 / h0.sum() - ha / ha.sum()).abs().sum() > 0.15


def test_char_lstm_learns_synthetic_chain():
    torch.manual_seed(0)
    data = text_synthetic.char_sequences(1, 256, seq_len=12, vocab=20,
                                         seed=5)
    x, y = data[0]
    model = CharLSTM(vocab_size=20, embedding_dim=8, hidden_size=32)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    first = last = None
    for step in range(30):
        opt.zero_grad()
        loss = F.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        if step == 0:
            first = loss.item()
        last = loss.item()
    assert last < first - 0.1, (first, last)


def test_word_sequences_and_stackoverflow_rnn():
    data = text_synthetic.word_sequences(2, 16, seq_len=10, vocab=50,
                                         seed=2)
    x, y = data[0]
    assert x.shape == (16, 10) and y.shape == (16, 10)
    assert int(x.max()) <= 50 + 3 + 1
    model = StackOverflowRNN(vocab_size=50, embedding_size=12,
                             latent_size=16)
    logits = model(x)
    assert logits.shape == (16, 10, 50 + 3 + 1)
    mask = y > 0                              # pad positions excluded
    loss = F.cross_entropy(logits[mask], y[mask])
    assert torch.isfinite(loss)


def test_load_csv_and_standardize(tmp_path):
    p = tmp_path / "t.csv"
    p.write_text("f1,f2,label\n1.0,2.0,cat\n3.0,4.0,dog\n5.0,6.0,cat\n")
    x, y = tabular.load_csv(str(p), label_col=-1)
    assert x.shape == (3, 2)
    assert y.tolist() == [0.0, 1.0, 0.0]      # categorical ids in order
    xs, mu, sd = tabular.standardize(x)
    assert np.allclose(xs.mean(axis=0), 0, atol=1e-6)
    assert np.allclose(xs.std(axis=0), 1, atol=1e-3)


def test_horizontal_shards_and_vertical_split():
    x, y = tabular.synthetic_susy(n=600, seed=0)
    shards = tabular.horizontal_shards(x, y, 4, mode="homo")
    assert sum(len(v[1]) for v in shards.values()) == 600
    het = tabular.horizontal_shards(x, y, 4, mode="hetero", alpha=0.3,
                                    seed=1)
    assert sum(len(v[1]) for v in het.values()) == 600
    parts = tabular.vertical_split(x, [10, 8])
    assert parts[0].shape == (600, 10) and parts[1].shape == (600, 8)


def test_synthetic_tabular_learnable():
    torch.manual_seed(0)
    for gen in [tabular.synthetic_susy, tabular.synthetic_lending]:
        x, y = gen(n=2000, seed=3)
        xs, _, _ = tabular.standardize(x)
        xt = torch.as_tensor(xs)
        yt = torch.as_tensor(y).long()
        lin = torch.nn.Sequential(torch.nn.Linear(x.shape[1], 16),
                                  torch.nn.ReLU(), torch.nn.Linear(16, 2))
        opt = torch.optim.Adam(lin.parameters(), lr=0.02)
        for _ in range(60):
            opt.zero_grad()
            F.cross_entropy(lin(xt), yt).backward()
            opt.step()
        acc = (lin(xt).argmax(1) == yt).float().mean().item()
        assert acc > 0.65, (gen.__name__, acc)


def test_nus_wide_two_party_shapes():
    xi, xt, y = tabular.synthetic_nus_wide(n=50, d_image=20, d_text=30,
                                           n_classes=3, seed=0)
    assert xi.shape == (50, 20) and xt.shape == (50, 30)
    assert y.min() >= 0 and y.max() < 3


def test_text_drift_timeline_end_to_end(tmp_path):
    """Concept drift over TEXT: FedDrift-Eager on the synthetic char-LM
    dataset with the CharLSTM through the sequential module engine (LSTM
    has no vmap batching rule) — the drift machinery is modality-agnostic."""
    import os

    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import generate_data
    from feddrift_amd.engine.timeline import run_timeline

    d = str(tmp_path / "data")
    os.makedirs(os.path.join(d, "changepoints"))
    mat = np.zeros((3, 3), dtype=int)
    mat[2:, :2] = 1                       # 2 clients drift at t=2
    np.savetxt(os.path.join(d, "changepoints", "T.cp"), mat, fmt="%u")
    np.random.seed(5)
    generate_data("text", d, 2, 3, 0, 128, 0.0, 1, "T")
    cfg = Config(model="rnn", dataset="text", data_dir=d,
                 client_num_in_total=3, client_num_per_round=3,
                 batch_size=64, lr=0.02, epochs=3, comm_round=8,
                 total_train_iteration=2, concept_num=2,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06",
                 change_points="T", dummy_arg=0, log_dir=str(tmp_path),
                 report_client=0)
    out = run_timeline(cfg)
    # next-char prediction on a 30-symbol Markov chain: clearly above the
    # 1/30 chance level after a few rounds
    assert out["avg_test_acc"] > 0.10, out["avg_test_acc"]
