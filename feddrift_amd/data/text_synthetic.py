"""Synthetic federated text data for the recurrent models.

The reference ships LEAF Shakespeare (char-level, 90-symbol vocab,
80-char context windows — fedml_api/data_preprocessing/shakespeare/
data_loader.py and fed_shakespeare/) and StackOverflow next-word
prediction (10k-word vocab + pad/bos/eos/oov —
stackoverflow_nwp/data_loader.py), both fetched over the network.  This
container is offline, so these generators produce data of the same SHAPE
and federated structure from client-specific Markov chains: each client
draws from its own order-1 transition matrix (non-IID by construction,
like LEAF's per-speaker/per-user partitions), and an optional `concept`
id permutes the matrix so the drift timeline machinery can run on text
too.

Output matches the engine's eager-batch convention: per client, integer
id tensors `x [n, seq_len]` (accepted as float rows by the models, cast
inside forward) and next-token targets.
"""

from __future__ import annotations

from typing import Dict, Tuple

import numpy as np
import torch


def _client_chain(rng: np.random.Generator, vocab: int,
                  concentration: float = 0.3) -> np.ndarray:
    """Sparse-ish Dirichlet transition matrix — every client/world its own."""
    mat = rng.dirichlet(np.full(vocab, concentration), size=vocab)
    return mat


def _sample_chain(rng: np.random.Generator, chain: np.ndarray,
                  n: int, seq_len: int) -> np.ndarray:
    vocab = chain.shape[0]
    out = np.empty((n, seq_len + 1), dtype=np.int64)
    state = rng.integers(1, vocab, size=n)      # 0 is the pad id: never emit
    for t in range(seq_len + 1):
        out[:, t] = state
        # vectorized categorical draw per row of the chain
        cdf = np.cumsum(chain[state], axis=1)
        u = rng.random(n)[:, None]
        state = np.maximum(1, (u > cdf).sum(axis=1))
    return out


def char_sequences(n_clients: int, n_per_client: int, seq_len: int = 80,
                   vocab: int = 90, seed: int = 0, concept: int = 0,
                   ) -> Dict[int, Tuple[torch.Tensor, torch.Tensor]]:
    """Shakespeare-shaped: x = 80-char context, y = next char
    (CharLSTM predicts the single next symbol, models/rnn.py)."""
    data = {}
    for c in range(n_clients):
        rng = np.random.default_rng(seed * 1_000_003 + c)
        chain = _client_chain(rng, vocab)
        if concept:
            perm = np.random.default_rng(concept).permutation(vocab)
            chain = chain[perm][:, perm]
        seqs = _sample_chain(rng, chain, n_per_client, seq_len)
        x = torch.as_tensor(seqs[:, :seq_len], dtype=torch.float32)
        y = torch.as_tensor(seqs[:, seq_len], dtype=torch.long)
        data[c] = (x, y)
    return data


def word_sequences(n_clients: int, n_per_client: int, seq_len: int = 20,
                   vocab: int = 10000, num_oov_buckets: int = 1,
                   seed: int = 0,
                   ) -> Dict[int, Tuple[torch.Tensor, torch.Tensor]]:
    """StackOverflow-shaped: x = [bos, w1..w_{L-1}] padded, y = per-position
    next word over the extended vocab (pad=0, bos/eos/oov appended after
    the word ids, as in the reference's id layout)."""
    extended = vocab + 3 + num_oov_buckets
    bos = vocab + 1
    eos = vocab + 2
    data = {}
    for c in range(n_clients):
        rng = np.random.default_rng(seed * 2_000_003 + c)
        # cheap word model: client-specific unigram over a topic slice
        width = min(200, vocab)
        lo = 1 + (c * 97) % max(1, vocab - width)
        probs = rng.dirichlet(np.full(width, 0.2))
        lens = rng.integers(max(2, seq_len // 2), seq_len, size=n_per_client)
        x = np.zeros((n_per_client, seq_len), dtype=np.int64)
        y = np.zeros((n_per_client, seq_len), dtype=np.int64)
        for i, L in enumerate(lens):
            words = lo + rng.choice(width, size=L, p=probs)
            seq = np.concatenate([[bos], words, [eos]])
            x[i, :L + 1] = seq[:L + 1]
            y[i, :L + 1] = seq[1:L + 2]
        data[c] = (torch.as_tensor(x, dtype=torch.float32),
                   torch.as_tensor(y, dtype=torch.long))
    return data
