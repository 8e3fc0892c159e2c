"""Shared workload plumbing: distributed init, checkpoints, lease wiring.

The reference carries three near-identical trees of seven apps
(workloads/, accordion_workloads/, gns_workloads/ — ~28 kLoC of copies);
here every family calls into this one module, and the adaptation mode
(static / accordion / gns) is a runtime flag.  Thin entry scripts under
``workloads/<tree>/<family>/`` keep the reference's CLI + directory layout
so its traces dispatch unchanged.
"""

from __future__ import annotations

import argparse
import os
import time
import torch
import torch.distributed as dist

from ..parallel import BucketedDataParallel
from ..runtime.lease_iterator import LeaseIterator


def add_scheduler_args(parser: argparse.ArgumentParser, num_steps_arg: str):
    """The arguments the dispatcher appends to every command
    (reference dispatcher.py:179-206, scheduler.py:2538-2552)."""
    parser.add_argument(num_steps_arg, dest="num_steps", type=int,
                        default=None, help="steps to run this lease")
    parser.add_argument("--local_rank", type=int, default=0)
    parser.add_argument("--checkpoint_dir", type=str, default=None)
    parser.add_argument("--enable_gavel_iterator", action="store_true")
    parser.add_argument("--master_addr", type=str, default=None)
    parser.add_argument("--master_port", type=int, default=None)
    parser.add_argument("--world_size", type=int, default=1)
    parser.add_argument("--rank", type=int, default=0)
    parser.add_argument("--mode", type=str, default=None,
                        choices=[None, "static", "accordion", "gns"],
                        help="adaptation mode override (else from env/tree)")
    parser.add_argument("--throughput_estimation_interval", type=int,
                        default=100)
    return parser


def init_device_and_distributed(args) -> torch.device:
    """Bind the GPU and bring up RCCL (gloo on CPU) when distributed."""
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        # the dispatcher both pins HIP_VISIBLE_DEVICES to this job's GPU
        # and passes --local_rank <gpu>: inside the process the pinned
        # GPU is device 0, so clamp to the visible range
        local = min(args.local_rank, torch.cuda.device_count() - 1)
        torch.cuda.set_device(local)
        device = torch.device("cuda", local)
    else:
        device = torch.device("cpu")
    if args.world_size > 1 and not dist.is_initialized():
        import datetime

        # explicit dispatcher args take precedence over a stale env (a
        # warm runner re-inits per lease; reusing the previous lease's
        # port can connect to its dying TCPStore and hang rendezvous)
        if args.master_addr:
            os.environ["MASTER_ADDR"] = args.master_addr
        else:
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if args.master_port:
            os.environ["MASTER_PORT"] = str(args.master_port)
        else:
            os.environ.setdefault("MASTER_PORT", "29500")
        # bounded rendezvous: a peer rank whose GPU slot is still held by
        # an over-running lease may never arrive this round; failing fast
        # turns the round into a zero-step micro-task failure (retried by
        # the scheduler) instead of wedging the slot for the backend's
        # default timeout
        timeout_s = float(os.environ.get("SWQ_RENDEZVOUS_TIMEOUT", "60"))
        # SWQ_DIST_BACKEND forces the backend: RCCL refuses two ranks on
        # one device (documented in profiles/MULTIGPU_PROBE.md), so
        # time-sliced measurement runs world>1 on a single GPU over gloo
        backend = os.environ.get(
            "SWQ_DIST_BACKEND", "nccl" if use_cuda else "gloo"
        )
        dist.init_process_group(
            backend=backend,
            world_size=args.world_size,
            rank=args.rank,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return device


def wrap_distributed(model, args):
    if args.world_size > 1:
        return BucketedDataParallel(model)
    return model


def finish_sync(model):
    if isinstance(model, BucketedDataParallel):
        model.finish_gradient_sync()


def zero_grads(model):
    """Zero gradients WITHOUT setting them to None: keeps gradient-buffer
    addresses stable so the fused-optimizer metadata cache and the DDP
    bucket views stay valid across steps."""
    if isinstance(model, BucketedDataParallel):
        model.zero_grad()
        return
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    if grads:
        torch._foreach_zero_(grads)


def checkpoint_path(checkpoint_dir: str) -> str:
    return os.path.join(checkpoint_dir, "model.chkpt")


def default_load_checkpoint(path):
    from ..parallel.ckpt_stream import CheckpointStore

    store = CheckpointStore(os.path.dirname(path))

    def load():
        return store.load()

    return load


def default_save_checkpoint(path):
    from ..parallel.ckpt_stream import CheckpointStore

    store = CheckpointStore(os.path.dirname(path))

    def save(state):
        store.save(state)

    return save


def make_lease_iterator(loader, args, synthetic_data=True, client=None):
    """Wrap a data loader in the lease iterator when running under the
    scheduler; otherwise return (loader, None)."""
    ckpt_dir = args.checkpoint_dir or "/tmp/swq_ckpt"
    path = checkpoint_path(ckpt_dir)
    load_fn = default_load_checkpoint(path)
    save_fn = default_save_checkpoint(path)
    if args.enable_gavel_iterator:
        it = LeaseIterator(
            loader, ckpt_dir, load_fn, save_fn,
            synthetic_data=synthetic_data, client=client,
        )
        return it, it
    return loader, None


class ThroughputReporter:
    """Prints the [THROUGHPUT_ESTIMATION] lines the reference workloads
    emit every N steps (cifar10 main.py:227-230)."""

    def __init__(self, interval, rank=0):
        self.interval = interval
        self.rank = rank
        self.total_steps = 0

    def step(self):
        self.total_steps += 1
        if (
            self.rank == 0
            and self.interval
            and self.total_steps % self.interval == 0
        ):
            print(
                "[THROUGHPUT_ESTIMATION]\t%s\t%d"
                % (time.time(), self.total_steps),
                flush=True,
            )
