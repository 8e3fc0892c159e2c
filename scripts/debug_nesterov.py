#!/usr/bin/env python3
"""Isolate the GPU nesterov mismatch: single-tensor cases vs hand math."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from shockwave_amd import ops

assert torch.cuda.is_available()
DEV = "cuda:0"


def run_case(n, steps=3, momentum=0.9, nesterov=True):
    torch.manual_seed(0)
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)
    p_ref = p.clone().cpu()
    g_ref = g.clone().cpu()
    buf_ref = torch.zeros(n)
    buf = torch.zeros(n, device=DEV)
    for step in range(steps):
        # manual torch-semantics reference on CPU
        d_p = g_ref.clone()
        if step == 0:
            buf_ref.copy_(d_p)
        else:
            buf_ref.mul_(momentum).add_(d_p)
        d_ref = d_p + momentum * buf_ref if nesterov else buf_ref.clone()
        p_ref.add_(d_ref, alpha=-0.1)
        ops.fused_sgd([p], [g], [buf], lr=0.1, momentum=momentum,
                      nesterov=nesterov, buf_initialized=(step > 0))
    torch.cuda.synchronize()
    diff = (p.cpu() - p_ref).abs().max().item()
    bufdiff = (buf.cpu() - buf_ref).abs().max().item()
    print(f"n={n:7d} steps={steps} nesterov={nesterov}: "
          f"p diff={diff:.3e} buf diff={bufdiff:.3e}")
    return diff


for n in (3, 4, 8, 1000, 32769):
    run_case(n, steps=1)
for n in (3, 4, 8, 1000, 32769):
    run_case(n, steps=3)
for n in (3, 1000):
    run_case(n, steps=3, nesterov=False)
