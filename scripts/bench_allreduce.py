#!/usr/bin/env python3
"""Measure the gloo host-staged all-reduce cost of each family's exact
DDP bucket layout (2 ranks time-slicing one MI355X).

Companion to scripts/measure_sf2.py: its `oh` (per-rank step-time excess
over pure 2x time-slicing) bundles the transport cost with sync/
bookkeeping.  Measuring the transport term alone — the same bucket
tensors, same gloo path, no compute — lets SF2_BOUNDS.md decompose

    oh  =  t_comm_gloo  +  residual(sync, hooks, scheduling)

and estimate the true 2-GPU xGMI efficiency by swapping the measured
gloo transport for the xGMI ring model (2(n-1)/n * bytes / 153 GB/s
per link):

    e_est(2) = t1 / (t1 + residual + t_comm_xgmi)

Writes gpurun_out/allreduce_gloo.json (run under gpurun).
"""

import argparse
import json
import os
import socket
import subprocess
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

# model -> constructor producing the training module (same ones the
# families train; bucket layout identical to BucketedDataParallel's)
MODELS = ["ResNet-18", "ResNet-50", "Transformer", "LM"]


def build_model(name, device):
    import torch

    from shockwave_amd import models as M

    if name == "ResNet-18":
        return M.resnet18_cifar().to(device)
    if name == "ResNet-50":
        return M.resnet50_imagenet().to(device)
    if name == "Transformer":
        return M.TranslationTransformer().to(device)
    if name == "LM":
        return M.LSTMLanguageModel().to(device)
    raise ValueError(name)


def worker(args):
    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(args.port)
    dist.init_process_group("gloo", rank=args.rank, world_size=2)
    device = (
        torch.device("cuda", 0) if torch.cuda.is_available()
        else torch.device("cpu")
    )
    out = {}
    for name in MODELS:
        from shockwave_amd.parallel import BucketedDataParallel

        model = build_model(name, device)
        ddp = BucketedDataParallel(model)
        bufs = ddp.grad_buffers
        total_mb = sum(b.numel() * b.element_size() for b in bufs) / 2**20
        # warm
        for _ in range(3):
            for b in bufs:
                dist.all_reduce(b)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.time()
        for _ in range(args.iters):
            works = [dist.all_reduce(b, async_op=True) for b in bufs]
            for w in works:
                w.wait()
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = (time.time() - t0) / args.iters
        out[name] = {"total_mb": total_mb, "buckets": len(bufs),
                     "allreduce_s": dt}
        if args.rank == 0:
            print(f"{name}: {total_mb:.1f} MB in {len(bufs)} buckets -> "
                  f"{dt*1e3:.2f} ms/step", flush=True)
        del ddp, model
        if device.type == "cuda":
            torch.cuda.empty_cache()
    if args.rank == 0:
        os.makedirs(args.outdir, exist_ok=True)
        with open(os.path.join(args.outdir, "allreduce_gloo.json"), "w") as f:
            json.dump(out, f, indent=1)
    dist.destroy_process_group()


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--worker", action="store_true")
    p.add_argument("--rank", type=int, default=0)
    p.add_argument("--port", type=int, default=29710)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--outdir", default="gpurun_out")
    args = p.parse_args()
    if args.worker:
        worker(args)
        return
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [
        subprocess.Popen(
            [sys.executable, os.path.abspath(__file__), "--worker",
             "--rank", str(r), "--port", str(port),
             "--iters", str(args.iters), "--outdir", args.outdir],
        )
        for r in range(2)
    ]
    rc = [p2.wait(timeout=900) for p2 in procs]
    sys.exit(max(rc))


if __name__ == "__main__":
    main()
