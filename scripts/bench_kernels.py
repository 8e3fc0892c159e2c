#!/usr/bin/env python3
"""Microbenchmark the CDNA4 fused kernels: achieved HBM bandwidth.

Each kernel is HBM-bound; the table reports moved bytes / time against the
~6.3 TB/s achievable ceiling (MI355X_MICROARCH.md).  Run under rocprofv3
for per-kernel confirmation; results land in profiles/kernel_bandwidth.md
when --out is given.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from shockwave_amd import ops


def time_kernel(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1e-3  # seconds


def split_sizes(total, pieces=60, seed=0):
    """Realistic multi-tensor shape mix (ResNet-like layer sizes)."""
    import random

    rng = random.Random(seed)
    cuts = sorted(rng.sample(range(1, total), pieces - 1))
    return [b - a for a, b in zip([0] + cuts, cuts + [total])]


def run(total_params, label, out_lines):
    dev = "cuda:0"
    sizes = split_sizes(total_params)
    p = [torch.randn(s, device=dev) for s in sizes]
    g = [torch.randn(s, device=dev) for s in sizes]
    m = [torch.zeros(s, device=dev) for s in sizes]
    v = [torch.zeros(s, device=dev) for s in sizes]
    B = 4 * total_params  # bytes per tensor list

    rows = []

    t = time_kernel(lambda: ops.fused_sgd(p, g, m, lr=0.1, momentum=0.9,
                                          weight_decay=5e-4))
    rows.append(("swq_fused_sgd", (3 + 2) * B, t))  # r:p,g,m w:p,m

    t = time_kernel(lambda: ops.fused_adam(p, g, m, v, lr=1e-3, step=10))
    rows.append(("swq_fused_adam", (4 + 3) * B, t))  # r:p,g,m,v w:p,m,v

    t = time_kernel(lambda: ops.multi_tensor_accum(m, g, 1.0))
    rows.append(("swq_multi_tensor_accum", 3 * B, t))  # r:m,g w:m

    t = time_kernel(lambda: ops.multi_tensor_l2norm(g))
    rows.append(("swq_multi_tensor_l2norm_sq", 1 * B, t))

    flat = [torch.randn(total_params, device=dev) for _ in range(4)]
    t = time_kernel(lambda: ops.gns_window_stats(flat))
    rows.append(("swq_gns_window_stats (W=4)", 4 * B, t))

    out_lines.append(f"\n### {label} ({total_params/1e6:.1f} M params)\n")
    out_lines.append("| kernel | bytes moved | time (us) | GB/s |")
    out_lines.append("|---|---|---|---|")
    for name, bytes_, secs in rows:
        gbs = bytes_ / secs / 1e9
        line = f"| {name} | {bytes_/1e6:.0f} MB | {secs*1e6:.1f} | {gbs:.0f} |"
        out_lines.append(line)
        print(f"{label:10s} {name:30s} {secs*1e6:8.1f} us  {gbs:7.0f} GB/s")


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    lines = ["# Fused-kernel achieved bandwidth (MI355X, fp32)",
             "", "Ceiling: ~6300 GB/s (measured float4 copy,",
             "MI355X_MICROARCH.md).  Bytes counted as reads+writes of each",
             "tensor list touched."]
    run(11_200_000, "ResNet-18", lines)   # ~ResNet-18 param count
    run(25_600_000, "ResNet-50", lines)
    run(100_000_000, "100M", lines)
    if args.out:
        with open(args.out, "w") as f:
            f.write("\n".join(lines) + "\n")
        print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
