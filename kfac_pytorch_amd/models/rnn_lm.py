"""LSTM language model (WikiText-2 shape).

Capability analog of the reference's RNN example
(reference: examples/wikitext_models.py; its trainer calls a stale K-FAC
API, examples/pytorch_wikitext_rnn.py:196-202 -- this one works against
the current API: the decoder Linear carries K-FAC factors, LSTM weights
stay first-order)."""

from __future__ import annotations

import torch
import torch.nn as nn

__all__ = ["LSTMLanguageModel"]


class LSTMLanguageModel(nn.Module):
    def __init__(self, vocab_size: int = 33278, emb: int = 256,
                 hidden: int = 256, layers: int = 2, dropout: float = 0.5,
                 tie_weights: bool = True):
        super().__init__()
        self.drop = nn.Dropout(dropout)
        self.embed = nn.Embedding(vocab_size, emb)
        self.lstm = nn.LSTM(emb, hidden, layers, dropout=dropout,
                            batch_first=True)
        self.decoder = nn.Linear(hidden, vocab_size)
        if tie_weights:
            assert emb == hidden
            self.decoder.weight = self.embed.weight
        self.vocab_size = vocab_size

    def forward(self, x, hidden=None):
        emb = self.drop(self.embed(x))
        out, hidden = self.lstm(emb, hidden)
        return self.decoder(self.drop(out)), hidden
