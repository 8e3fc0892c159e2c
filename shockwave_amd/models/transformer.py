"""Encoder-decoder Transformer for the translation workload.

Multi30k-shaped seq2seq model (reference workloads/pytorch/translation:
"Attention is All You Need" settings — d_model 512, 6+6 layers, 8 heads).
Built on torch.nn.Transformer (MIOpen/hipBLASLt-backed on ROCm).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn


class PositionalEncoding(nn.Module):
    def __init__(self, d_model, max_len=512, dropout=0.1):
        super().__init__()
        self.dropout = nn.Dropout(dropout)
        pe = torch.zeros(max_len, d_model)
        pos = torch.arange(max_len).unsqueeze(1).float()
        div = torch.exp(
            torch.arange(0, d_model, 2).float() * (-math.log(10000.0) / d_model)
        )
        pe[:, 0::2] = torch.sin(pos * div)
        pe[:, 1::2] = torch.cos(pos * div)
        self.register_buffer("pe", pe.unsqueeze(0))

    def forward(self, x):
        return self.dropout(x + self.pe[:, : x.size(1)])


class TranslationTransformer(nn.Module):
    def __init__(
        self,
        src_vocab=9521,
        tgt_vocab=17851,
        d_model=512,
        nhead=8,
        num_layers=6,
        dim_ff=2048,
        dropout=0.1,
        share_proj_weight=True,
        max_len=128,
    ):
        super().__init__()
        self.d_model = d_model
        self.src_emb = nn.Embedding(src_vocab, d_model)
        self.tgt_emb = nn.Embedding(tgt_vocab, d_model)
        self.pos = PositionalEncoding(d_model, max_len, dropout)
        self.transformer = nn.Transformer(
            d_model=d_model,
            nhead=nhead,
            num_encoder_layers=num_layers,
            num_decoder_layers=num_layers,
            dim_feedforward=dim_ff,
            dropout=dropout,
            batch_first=True,
        )
        self.proj = nn.Linear(d_model, tgt_vocab, bias=False)
        if share_proj_weight:
            self.proj.weight = self.tgt_emb.weight

    def forward(self, src, tgt):
        scale = math.sqrt(self.d_model)
        src_e = self.pos(self.src_emb(src) * scale)
        tgt_e = self.pos(self.tgt_emb(tgt) * scale)
        tgt_mask = nn.Transformer.generate_square_subsequent_mask(
            tgt.size(1), device=tgt.device
        )
        out = self.transformer(src_e, tgt_e, tgt_mask=tgt_mask)
        return self.proj(out)
