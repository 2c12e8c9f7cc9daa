"""Recurrent models of the FedML zoo.

Counterparts of the reference fedml_api/model/nlp/rnn.py:
  * CharLSTM — the original FedAvg-paper Shakespeare next-character model
    (RNN_OriginalFedAvg: 8-dim char embedding, 2x LSTM(256), dense to the
    90-char vocabulary).
  * StackOverflowRNN — next-word prediction (RNN_StackOverFlow: 96-dim
    embedding over 10k words + special tokens, LSTM(670), projection 96,
    dense back to the vocabulary).
Inputs arrive as integer id sequences (accepted as float rows from the
flat data layer and cast inside forward).
"""

from __future__ import annotations

from torch import nn


class CharLSTM(nn.Module):
    def __init__(self, vocab_size: int = 90, embedding_dim: int = 8,
                 hidden_size: int = 256):
        super().__init__()
        self.embeddings = nn.Embedding(vocab_size, embedding_dim, padding_idx=0)
        self.lstm = nn.LSTM(embedding_dim, hidden_size, num_layers=2,
                            batch_first=True)
        self.fc = nn.Linear(hidden_size, vocab_size)

    def forward(self, x):
        ids = x.long()
        emb = self.embeddings(ids)
        out, _ = self.lstm(emb)
        return self.fc(out[:, -1, :])   # next-char logits


class StackOverflowRNN(nn.Module):
    def __init__(self, vocab_size: int = 10000, num_oov_buckets: int = 1,
                 embedding_size: int = 96, latent_size: int = 670,
                 num_layers: int = 1):
        super().__init__()
        extended = vocab_size + 3 + num_oov_buckets   # pad/bos/eos + oov
        self.word_embeddings = nn.Embedding(extended, embedding_size,
                                            padding_idx=0)
        self.lstm = nn.LSTM(embedding_size, latent_size,
                            num_layers=num_layers, batch_first=True)
        self.fc1 = nn.Linear(latent_size, embedding_size)
        self.fc2 = nn.Linear(embedding_size, extended)

    def forward(self, x):
        ids = x.long()
        emb = self.word_embeddings(ids)
        out, _ = self.lstm(emb)
        return self.fc2(self.fc1(out))   # per-position next-word logits
