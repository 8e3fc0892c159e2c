"""Fused multi-tensor ops: CDNA4 HIP kernels with torch fp32 reference.

Dispatch rule:

* device (ROCm) tensors -> the compiled ``shockwave_amd.ops._C`` extension.
  If the extension is missing on a GPU machine the call raises — there is
  NO silent eager fallback on GPU (the HIP path must be the one that runs).
* CPU tensors -> pure-torch reference implementations of the same math,
  used by CPU tests and as the numerics baseline for the GPU kernels.
"""

from __future__ import annotations

import math
from typing import List, Tuple

import torch

try:
    from . import _C  # compiled by setup.py build_ext --inplace

    HAVE_EXT = True
except ImportError:  # pragma: no cover - exercised on CPU-only boxes
    _C = None
    HAVE_EXT = False


def _require_ext():
    if not HAVE_EXT:
        raise RuntimeError(
            "shockwave_amd.ops._C extension not built; run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
            "GPU tensors are never silently handled in eager mode."
        )


# ---------------------------------------------------------------------------
# fused SGD
# ---------------------------------------------------------------------------

def fused_sgd(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    momentum_bufs: List[torch.Tensor],
    lr: float,
    momentum: float = 0.0,
    dampening: float = 0.0,
    weight_decay: float = 0.0,
    nesterov: bool = False,
    buf_initialized: bool = True,
) -> None:
    if params[0].is_cuda:
        _require_ext()
        _C.fused_sgd(
            params, grads, momentum_bufs, lr, momentum, dampening,
            weight_decay, nesterov, buf_initialized,
        )
        return
    for p, g, b in zip(params, grads, momentum_bufs):
        d_p = g.clone()
        if weight_decay != 0:
            d_p.add_(p, alpha=weight_decay)
        if momentum != 0:
            if not buf_initialized:
                b.copy_(d_p)
            else:
                b.mul_(momentum).add_(d_p, alpha=1 - dampening)
            d_p = d_p.add(b, alpha=momentum) if nesterov else b.clone()
        p.add_(d_p, alpha=-lr)


# ---------------------------------------------------------------------------
# fused Adam
# ---------------------------------------------------------------------------

def fused_adam(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    exp_avgs: List[torch.Tensor],
    exp_avg_sqs: List[torch.Tensor],
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    weight_decay: float = 0.0,
    step: int = 1,
    adamw: bool = False,
) -> None:
    if params[0].is_cuda:
        _require_ext()
        _C.fused_adam(
            params, grads, exp_avgs, exp_avg_sqs, lr, beta1, beta2, eps,
            weight_decay, step, adamw,
        )
        return
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
        g = g.clone()
        if adamw:
            p.mul_(1 - lr * weight_decay)
        elif weight_decay != 0:
            g.add_(p, alpha=weight_decay)
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
        p.addcdiv_(m, denom, value=-lr / bc1)


# ---------------------------------------------------------------------------
# Accordion helpers
# ---------------------------------------------------------------------------

def multi_tensor_accum(
    dsts: List[torch.Tensor], srcs: List[torch.Tensor], alpha: float = 1.0
) -> None:
    """dst += alpha * src elementwise for each tensor pair."""
    if dsts[0].is_cuda:
        _require_ext()
        _C.multi_tensor_accum(dsts, srcs, alpha)
        return
    for d, s in zip(dsts, srcs):
        d.add_(s, alpha=alpha)


def multi_tensor_l2norm(tensors: List[torch.Tensor]) -> torch.Tensor:
    """Per-tensor L2 norms, returned as a 1-D tensor of len(tensors)."""
    if tensors[0].is_cuda:
        _require_ext()
        out = torch.empty(
            len(tensors), dtype=torch.float32, device=tensors[0].device
        )
        _C.multi_tensor_l2norm_sq(tensors, out)
        return out.sqrt()
    return torch.stack([t.norm() for t in tensors])


# ---------------------------------------------------------------------------
# GNS estimator
# ---------------------------------------------------------------------------

def gns_window_stats(
    window_grads: List[torch.Tensor],
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(||mean of window||^2, ||last grad||^2) over flat fp32 grads."""
    if window_grads[0].is_cuda:
        _require_ext()
        out = torch.empty(
            2, dtype=torch.float32, device=window_grads[0].device
        )
        _C.gns_window_stats(window_grads, out)
        return out[0], out[1]
    mean = torch.stack(window_grads).mean(dim=0)
    return (mean * mean).sum(), (window_grads[-1] * window_grads[-1]).sum()
