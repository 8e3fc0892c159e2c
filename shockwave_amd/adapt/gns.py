"""Gradient-noise-scale estimator — shared adaptation library.

One implementation of the GNS logic the reference copies into each
gns_workloads tree (gns cifar10 main.py:329-383 et al.), backed by the
``swq_gns_window_stats`` CDNA4 kernel:

maintain a sliding window of W flattened gradients (W = world size, or 2
single-GPU); per step compute

    |G_big|^2  = || mean_w g_w ||^2        (large-batch gradient estimate)
    |G_small|^2 = || g_latest ||^2          (small-batch gradient)

then the unbiased estimators (OpenAI GNS):

    |G|^2 = (B_big |G_big|^2 - B_small |G_small|^2) / (B_big - B_small)
    S     = (|G_small|^2 - |G_big|^2) / (1/B_small - 1/B_big)
    GNS   = S / |G|^2
"""

from __future__ import annotations

from collections import deque
from typing import List, Optional

import torch

from .. import ops


class GNSEstimator:
    def __init__(self, model: torch.nn.Module, batch_size: int,
                 window: int = 2, ema: float = 0.9):
        self.window = max(2, window)
        self.batch_size = batch_size
        self.params = [p for p in model.parameters() if p.requires_grad]
        self._grads: deque = deque(maxlen=self.window)
        self.ema = ema
        self._g2_avg: Optional[float] = None
        self._s_avg: Optional[float] = None
        self.gns_by_epoch = {}

    def _flat_grad(self) -> torch.Tensor:
        return torch.cat(
            [
                p.grad.detach().reshape(-1).float()
                for p in self.params
                if p.grad is not None
            ]
        )

    def on_step(self) -> Optional[float]:
        """Push the current gradient; return the running GNS estimate once
        the window is full."""
        self._grads.append(self._flat_grad())
        if len(self._grads) < self.window:
            return None
        big_sq, small_sq = ops.gns_window_stats(list(self._grads))
        big_sq = float(big_sq)
        small_sq = float(small_sq)
        b_small = self.batch_size
        b_big = self.batch_size * self.window
        g2 = (b_big * big_sq - b_small * small_sq) / (b_big - b_small)
        s = (small_sq - big_sq) / (1.0 / b_small - 1.0 / b_big)
        if self._g2_avg is None:
            self._g2_avg, self._s_avg = g2, s
        else:
            self._g2_avg = self.ema * self._g2_avg + (1 - self.ema) * g2
            self._s_avg = self.ema * self._s_avg + (1 - self.ema) * s
        if self._g2_avg <= 0:
            return None
        return self._s_avg / self._g2_avg

    def on_epoch(self, epoch: int) -> None:
        gns = None
        if self._g2_avg and self._g2_avg > 0:
            gns = self._s_avg / self._g2_avg
        self.gns_by_epoch[epoch] = gns

    def should_double(self, epoch: int, lookback: int = 10) -> bool:
        """Every ``lookback`` epochs: double bs if current GNS exceeds the
        trailing average (reference gns main.py:520-556)."""
        if epoch < lookback or epoch % lookback != lookback - 1:
            return False
        recent = [
            v
            for e, v in self.gns_by_epoch.items()
            if epoch - lookback <= e < epoch and v is not None
        ]
        cur = self.gns_by_epoch.get(epoch)
        if cur is None or not recent:
            return False
        return cur > sum(recent) / len(recent)

    def state_dict(self):
        return {
            "g2_avg": self._g2_avg,
            "s_avg": self._s_avg,
            "gns_by_epoch": self.gns_by_epoch,
            "batch_size": self.batch_size,
        }

    def load_state_dict(self, state):
        self._g2_avg = state["g2_avg"]
        self._s_avg = state["s_avg"]
        self.gns_by_epoch = {int(k): v for k, v in state["gns_by_epoch"].items()}
        self.batch_size = state["batch_size"]
