"""Autoencoder recommender for the recommendation workload.

ML-20M-shaped denoising autoencoder (reference
workloads/pytorch/recommendation, the Recoder project's autoencoder):
sparse item-interaction vectors -> bottleneck -> reconstruction.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class RecommendationAutoencoder(nn.Module):
    def __init__(self, num_items=20108, hidden=(600, 200), noise_prob=0.5):
        super().__init__()
        dims = [num_items] + list(hidden)
        self.encode_layers = nn.ModuleList(
            nn.Linear(dims[i], dims[i + 1]) for i in range(len(dims) - 1)
        )
        self.decode_layers = nn.ModuleList(
            nn.Linear(dims[i + 1], dims[i]) for i in reversed(range(len(dims) - 1))
        )
        self.noise_prob = noise_prob

    def forward(self, x):
        if self.training and self.noise_prob > 0:
            x = F.dropout(x, p=self.noise_prob)
        z = x
        for layer in self.encode_layers:
            z = torch.tanh(layer(z))
        for i, layer in enumerate(self.decode_layers):
            z = layer(z)
            if i < len(self.decode_layers) - 1:
                z = torch.tanh(z)
        return z

    @staticmethod
    def loss(recon, target):
        """Masked MSE over observed interactions + negative sampling over
        the rest (simplified Recoder loss)."""
        return F.mse_loss(recon, target)


def recall_at_k(scores: torch.Tensor, targets: torch.Tensor,
                k: int = 20) -> torch.Tensor:
    """Per-user Recall@k: hits in the top-k / min(#relevant, k)
    (reference workloads/pytorch/recommendation recoder/metrics.py)."""
    topk = scores.topk(k, dim=1).indices
    hits = targets.gather(1, topk).sum(1)
    denom = targets.sum(1).clamp(max=float(k)).clamp(min=1.0)
    return hits / denom


def ndcg_at_k(scores: torch.Tensor, targets: torch.Tensor,
              k: int = 20) -> torch.Tensor:
    """Per-user NDCG@k with binary relevance."""
    topk = scores.topk(k, dim=1).indices
    gains = targets.gather(1, topk)
    discounts = 1.0 / torch.log2(
        torch.arange(2, k + 2, dtype=scores.dtype, device=scores.device)
    )
    dcg = (gains * discounts).sum(1)
    ideal_counts = targets.sum(1).clamp(max=float(k)).long()
    cum = torch.cat(
        [torch.zeros(1, dtype=scores.dtype, device=scores.device),
         discounts.cumsum(0)]
    )
    idcg = cum[ideal_counts]
    return dcg / idcg.clamp(min=1e-8)
