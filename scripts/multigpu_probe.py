#!/usr/bin/env python3
"""World>1 probe on limited hardware: N ranks time-sharing one MI355X.

VERDICT r1 item 1: prove BucketedDataParallel + the partial-graph
fallback work over RCCL at world>1 on real hardware, and measure
all-reduce bandwidth for DDP-sized buckets — within a 1-GPU lease by
running both ranks on cuda:0 (HIP time-slicing).  If RCCL refuses a
duplicate-device communicator the probe falls back to gloo with CUDA
tensors, which still exercises the wrapper's GPU path and gives a
time-sliced sf=2 throughput lower bound.

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port 29617 \
      scripts/multigpu_probe.py --steps 30
Writes gpurun_out/multigpu_probe.json from rank 0.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def log(msg):
    print(f"[rank {os.environ.get('RANK','?')}] {msg}", flush=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--bs", type=int, default=16)
    p.add_argument("--out", default="gpurun_out/multigpu_probe.json")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    assert world > 1, "launch with torchrun --nproc-per-node 2"
    # every rank shares GPU 0 (1-GPU lease)
    torch.cuda.set_device(0)
    device = torch.device("cuda", 0)

    result = {"world_size": world, "device": torch.cuda.get_device_name(0)}

    backend = "nccl"
    try:
        dist.init_process_group(backend="nccl")
        # a tiny collective proves the communicator actually works —
        # init can succeed lazily and fail on first use
        t = torch.ones(4, device=device)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert t[0].item() == world
        log("RCCL communicator with duplicate device works")
    except Exception as e:
        log(f"RCCL duplicate-device refused: {e!r}; falling back to gloo")
        if dist.is_initialized():
            dist.destroy_process_group()
        backend = "gloo"
        dist.init_process_group(backend="gloo")
    result["backend"] = backend

    # ------------------------------------------------------------------
    # all-reduce bandwidth sweep, DDP bucket sizes (GPU tensors)
    # ------------------------------------------------------------------
    bw = {}
    for mb in (1, 8, 16, 32, 45):
        n = mb * 1024 * 1024 // 4
        x = torch.randn(n, device=device)
        for _ in range(3):
            dist.all_reduce(x)
        torch.cuda.synchronize()
        dist.barrier()
        iters = 10
        t0 = time.time()
        for _ in range(iters):
            dist.all_reduce(x)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / iters
        # ring all-reduce moves 2(w-1)/w of the buffer per link
        algbw = mb / 1024 / dt  # GiB/s algorithmic
        bw[f"{mb}MB"] = {"ms": round(dt * 1e3, 3), "algbw_GiB_s": round(algbw, 2)}
        log(f"all_reduce {mb} MB: {dt*1e3:.2f} ms ({algbw:.1f} GiB/s alg)")
    result["allreduce"] = bw

    # ------------------------------------------------------------------
    # world-2 DDP training step: ResNet-18 bs16 through the real wrapper
    # ------------------------------------------------------------------
    from shockwave_amd.models import resnet18_cifar
    from shockwave_amd.ops.optim import FusedSGD
    from shockwave_amd.parallel import BucketedDataParallel
    from shockwave_amd.workloads import common

    torch.manual_seed(0)
    model = resnet18_cifar().to(device).to(memory_format=torch.channels_last)
    ddp = BucketedDataParallel(model)
    opt = FusedSGD(ddp.parameters(), lr=0.1, momentum=0.9, weight_decay=5e-4)
    crit = torch.nn.CrossEntropyLoss().to(device)
    x = torch.randn(args.bs, 3, 32, 32, device=device).to(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 10, (args.bs,), device=device)

    def step():
        common.zero_grads(ddp)
        loss = crit(ddp(x), y)
        loss.backward()
        ddp.finish_gradient_sync()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dist.barrier()
    eager_ms = (time.time() - t0) / args.steps * 1e3
    log(f"world-{world} eager DDP step (time-sliced 1 GPU): {eager_ms:.2f} ms")
    result["ddp_eager_ms_per_step"] = round(eager_ms, 3)

    # gradient-correctness check: both ranks must hold identical averaged
    # gradients after a step with rank-dependent data
    xr = torch.randn(args.bs, 3, 32, 32, device=device,
                     generator=torch.Generator(device="cuda").manual_seed(rank)
                     ).to(memory_format=torch.channels_last)
    common.zero_grads(ddp)
    crit(ddp(xr), y).backward()
    ddp.finish_gradient_sync()
    gsum = torch.stack([b.double().sum() for b in ddp.grad_buffers]).sum()
    gather = [torch.zeros_like(gsum) for _ in range(world)]
    dist.all_gather(gather, gsum)
    spread = max(abs(float(g - gather[0])) for g in gather)
    result["grad_sync_spread"] = spread
    assert spread < 1e-6 * max(1.0, abs(float(gather[0]))), gather
    log(f"grad sync OK (spread {spread:.2e})")

    # ------------------------------------------------------------------
    # partial-graph fallback: capture fwd+bwd, collectives outside
    # ------------------------------------------------------------------
    from shockwave_amd.parallel.graphs import try_graph_step

    ddp.sync_mode = "manual"

    def fwd_bwd(xx, yy):
        common.zero_grads(ddp)
        crit(ddp(xx), yy).backward()

    partial = try_graph_step(fwd_bwd, [x, y])
    if partial is None:
        result["partial_graph"] = "capture failed"
        log("partial-graph capture failed")
    else:
        def gstep():
            partial(x, y)
            ddp.finish_gradient_sync()
            opt.step()

        for _ in range(args.warmup):
            gstep()
        torch.cuda.synchronize()
        dist.barrier()
        t0 = time.time()
        for _ in range(args.steps):
            gstep()
        torch.cuda.synchronize()
        dist.barrier()
        graph_ms = (time.time() - t0) / args.steps * 1e3
        result["ddp_partial_graph_ms_per_step"] = round(graph_ms, 3)
        log(f"world-{world} partial-graph DDP step: {graph_ms:.2f} ms")

    # full-graph attempt (RCCL collectives inside hipGraph)
    ddp.sync_mode = "hook"
    full = try_graph_step(lambda xx, yy: (step(), None)[1], [x, y])
    result["full_graph_with_collectives"] = full is not None
    log(f"full-graph capture incl. collectives: "
        f"{'OK' if full is not None else 'not capturable'}")

    if rank == 0:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(result, f, indent=2)
        log(f"wrote {args.out}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
