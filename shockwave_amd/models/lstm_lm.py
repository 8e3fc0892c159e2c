"""LSTM language model for the language-modeling workload.

Wikitext-2-shaped 2-layer LSTM (reference workloads/pytorch/language_modeling:
emsize/nhid 650 per the stock word_language_model settings).
"""

from __future__ import annotations

import torch
import torch.nn as nn


class LSTMLanguageModel(nn.Module):
    def __init__(self, vocab=33278, emsize=650, nhid=650, nlayers=2, dropout=0.5):
        super().__init__()
        self.drop = nn.Dropout(dropout)
        self.encoder = nn.Embedding(vocab, emsize)
        self.rnn = nn.LSTM(emsize, nhid, nlayers, dropout=dropout, batch_first=False)
        self.decoder = nn.Linear(nhid, vocab)
        self.nhid = nhid
        self.nlayers = nlayers
        self.init_weights()

    def init_weights(self):
        rng = 0.1
        nn.init.uniform_(self.encoder.weight, -rng, rng)
        nn.init.zeros_(self.decoder.bias)
        nn.init.uniform_(self.decoder.weight, -rng, rng)

    def init_hidden(self, bsz, device):
        w = next(self.parameters())
        return (
            w.new_zeros(self.nlayers, bsz, self.nhid),
            w.new_zeros(self.nlayers, bsz, self.nhid),
        )

    def forward(self, x, hidden):
        emb = self.drop(self.encoder(x))
        out, hidden = self.rnn(emb, hidden)
        out = self.drop(out)
        return self.decoder(out), hidden

    @staticmethod
    def repackage_hidden(h):
        if isinstance(h, torch.Tensor):
            return h.detach()
        return tuple(LSTMLanguageModel.repackage_hidden(v) for v in h)
