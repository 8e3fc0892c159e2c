"""Accordion critical-regime detector — shared adaptation library.

The reference replicates this logic per workload
(accordion_workloads/pytorch/image_classification/cifar10/main.py:276-429
and six copies); here it is ONE implementation backed by the CDNA4
multi-tensor kernels:

* per step: accumulate gradients of all >=2-D parameters into fp32
  buffers (``swq_multi_tensor_accum``),
* per epoch: per-layer L2 norms of the accumulated gradients
  (``swq_multi_tensor_l2norm_sq``), reset buffers,
* every ``interval`` epochs: compare the summed norms against the previous
  checkpoint — relative change < threshold means the job left the critical
  regime (switch to the large batch size); >= threshold means it is back
  (switch to the small batch size).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from .. import ops


class AccordionDetector:
    def __init__(
        self,
        model: torch.nn.Module,
        interval: int = 10,
        threshold: float = 0.5,
        min_dims: int = 2,
    ):
        self.interval = interval
        self.threshold = threshold
        self.params: List[torch.Tensor] = [
            p for p in model.parameters() if p.dim() >= min_dims
        ]
        self.accum: List[torch.Tensor] = [
            torch.zeros_like(p, dtype=torch.float32) for p in self.params
        ]
        self.norms_by_epoch: Dict[int, List[float]] = {}
        self._prev_checkpoint_sum: Optional[float] = None
        self.in_critical_regime = True

    def on_step(self) -> None:
        """Accumulate current gradients (call after backward)."""
        grads = [
            p.grad for p in self.params if p.grad is not None
        ]
        if len(grads) != len(self.params):
            ps = [p for p in self.params if p.grad is not None]
            accums = [
                a for a, p in zip(self.accum, self.params) if p.grad is not None
            ]
        else:
            ps, accums = self.params, self.accum
        ops.multi_tensor_accum(accums, [p.grad for p in ps], alpha=1.0)

    def on_epoch(self, epoch: int) -> Optional[bool]:
        """Record per-layer norms; every ``interval`` epochs decide the
        regime.  Returns True (entered critical regime), False (left), or
        None (no change decision this epoch)."""
        norms = ops.multi_tensor_l2norm(self.accum)
        self.norms_by_epoch[epoch] = norms.tolist()
        for a in self.accum:
            a.zero_()

        if epoch % self.interval != self.interval - 1:
            return None
        total = sum(self.norms_by_epoch[epoch])
        decision = None
        if self._prev_checkpoint_sum is not None and self._prev_checkpoint_sum > 0:
            rel_change = abs(total - self._prev_checkpoint_sum) / (
                self._prev_checkpoint_sum
            )
            was_critical = self.in_critical_regime
            self.in_critical_regime = rel_change >= self.threshold
            if self.in_critical_regime != was_critical:
                decision = self.in_critical_regime
        self._prev_checkpoint_sum = total
        return decision

    # -- checkpointing ------------------------------------------------------

    def state_dict(self):
        return {
            "norms_by_epoch": self.norms_by_epoch,
            "prev_checkpoint_sum": self._prev_checkpoint_sum,
            "in_critical_regime": self.in_critical_regime,
            "accum": [a.cpu() for a in self.accum],
        }

    def load_state_dict(self, state):
        self.norms_by_epoch = {
            int(k): v for k, v in state["norms_by_epoch"].items()
        }
        self._prev_checkpoint_sum = state["prev_checkpoint_sum"]
        self.in_critical_regime = state["in_critical_regime"]
        for a, saved in zip(self.accum, state["accum"]):
            a.copy_(saved.to(a.device))

    def reset(self) -> None:
        """Fresh-job state, in place (session-cache reuse)."""
        for a in self.accum:
            a.zero_()
        self.norms_by_epoch = {}
        self._prev_checkpoint_sum = None
        self.in_critical_regime = True


def hardcoded_critical_regime(model: str, original_bs: int, epoch: int) -> bool:
    """The hard-coded regime tables the reference uses in practice
    (check_critical_regime_hardcode; also the simulator twin,
    scheduler.py:1658-1726)."""
    if model == "Transformer":
        return True  # accordion not applicable -> always "critical"
    if model == "LM":
        return epoch < 10
    if model == "Recommendation":
        if original_bs in (512, 1024):
            return epoch < 30
        if original_bs == 2048:
            return epoch < 40
        return epoch < 10
    if model == "ResNet-50":
        return (epoch % 30) < 10
    if model == "ResNet-18":
        head = 20 if original_bs == 256 else 10
        return epoch < head or 150 <= epoch < 160 or 250 <= epoch < 260
    return True
