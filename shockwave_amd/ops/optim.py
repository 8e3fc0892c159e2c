"""torch.optim-compatible optimizers backed by the fused CDNA4 kernels.

Drop-in for the reference workloads' ``optim.SGD(momentum=0.9, wd=5e-4)``
and Adam optimizers: one multi-tensor kernel per
step instead of per-parameter elementwise ops.
"""

from __future__ import annotations

import torch

from . import fused_adam, fused_sgd


def _preinit_grads(param_groups):
    """Create zero gradients with each param's own memory layout before
    the first backward: autograd then accumulates into them, so grad
    layout always matches the param (a backward kernel is otherwise free
    to emit a different-layout grad, e.g. NCHW grads for channels_last
    1x1-conv weights) and grad addresses stay stable for the metadata
    cache and hipGraphs."""
    for group in param_groups:
        for p in group["params"]:
            if p.requires_grad and p.grad is None:
                p.grad = torch.zeros_like(p)


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr, momentum=0.0, dampening=0.0,
                 weight_decay=0.0, nesterov=False):
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)
        _preinit_grads(self.param_groups)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            params, grads, bufs = [], [], []
            momentum = group["momentum"]
            buf_initialized = True
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if momentum != 0 and "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p)
                    buf_initialized = False
                params.append(p)
                grads.append(p.grad)
                bufs.append(
                    state["momentum_buffer"] if momentum != 0 else p.grad
                )
            if not params:
                continue
            fused_sgd(
                params, grads, bufs,
                lr=group["lr"], momentum=momentum,
                dampening=group["dampening"],
                weight_decay=group["weight_decay"],
                nesterov=group["nesterov"],
                buf_initialized=buf_initialized,
            )
        return loss


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adamw=False, amsgrad=False):
        assert not amsgrad, "amsgrad not supported by the fused kernel"
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)
        _preinit_grads(self.param_groups)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            params, grads, avgs, sqs = [], [], [], []
            step = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                    state["step"] = 0
                state["step"] += 1
                step = state["step"]
                params.append(p)
                grads.append(p.grad)
                avgs.append(state["exp_avg"])
                sqs.append(state["exp_avg_sq"])
            if not params:
                continue
            beta1, beta2 = group["betas"]
            fused_adam(
                params, grads, avgs, sqs,
                lr=group["lr"], beta1=beta1, beta2=beta2, eps=group["eps"],
                weight_decay=group["weight_decay"], step=step,
                adamw=group["adamw"],
            )
        return loss


class FusedAdamW(FusedAdam):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        super().__init__(params, lr, betas, eps, weight_decay, adamw=True)
