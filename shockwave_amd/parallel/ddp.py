"""Bucketed data parallelism over RCCL/xGMI.

MI355X-first replacement for the torch DDP wrapper used by the reference
workloads.  Design points for this fabric:

* xGMI is point-to-point (7 links x ~153 GB/s per GPU); ring all-reduce is
  per-link bound, so bucket sizes are chosen large (default 32 MiB) to
  amortize per-collective latency while still overlapping with backward.
* gradients are **views into per-bucket flat buffers** — autograd
  accumulates straight into the communication buffer, so there is no
  pack/unpack pass (one less read+write of every gradient per step, which
  matters when the whole model is ~45 MB against 8 TB/s of HBM).
* each bucket all-reduces asynchronously as soon as its last gradient is
  produced (post-accumulate-grad hooks), overlapping communication with
  the rest of backward; ``finish_gradient_sync`` waits and averages.
* the flat-buffer layout also keeps the fused-optimizer metadata cache
  stable across steps (shockwave_amd/ops), and makes the whole
  fwd+bwd+step capturable in a hipGraph.

Works on the ``gloo`` backend too (CPU tests, world_size > 1).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class Bucket:
    def __init__(self, params: List[torch.nn.Parameter], buffer: torch.Tensor):
        self.params = params
        self.buffer = buffer
        self.pending = 0
        self.work = None


class BucketedDataParallel(torch.nn.Module):
    def __init__(
        self,
        module: torch.nn.Module,
        bucket_bytes: int = 32 * 1024 * 1024,
        process_group=None,
        average: bool = True,
    ):
        super().__init__()
        self.module = module
        self.group = process_group
        self.world_size = dist.get_world_size(process_group)
        self.average = average

        params = [p for p in module.parameters() if p.requires_grad]
        # bucket in reverse registration order ~ backward completion order
        self.buckets: List[Bucket] = []
        self._param_bucket: Dict[torch.nn.Parameter, Bucket] = {}
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in reversed(params):
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= bucket_bytes:
                self._make_bucket(cur)
                cur, cur_bytes = [], 0
        if cur:
            self._make_bucket(cur)

        # one initial broadcast so all ranks start from rank 0's weights
        with torch.no_grad():
            for p in params:
                dist.broadcast(p.data, src=0, group=self.group)

        self._hooks = []
        for p in params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._grad_ready_hook)
            )
        self._require_finish = False
        self._accumulating = False

    def _make_bucket(self, params: List[torch.nn.Parameter]):
        device = params[0].device
        dtype = params[0].dtype
        # the fused optimizer kernels (ops/csrc/fused_ops.hip) vectorize as
        # float4 (16 B); every grad view must start 16-B aligned, so round
        # each param's offset up to an element multiple of 16 bytes
        align = max(1, 16 // params[0].element_size())
        offsets = []
        offset = 0
        for p in params:
            offsets.append(offset)
            offset += -(-p.numel() // align) * align
        buffer = torch.zeros(offset, dtype=dtype, device=device)
        for p, off in zip(params, offsets):
            # match the param's own dense layout (NCHW or channels_last) so
            # autograd accumulates into the bucket with the param's strides
            # and the flat fused-optimizer kernels see aligned layouts
            p.grad = buffer.as_strided(p.shape, p.stride(), off)
        bucket = Bucket(params, buffer)
        for p in params:
            self._param_bucket[p] = bucket
        self.buckets.append(bucket)

    # "hook": each bucket all-reduces asynchronously as soon as its last
    # gradient lands (overlaps with the rest of backward).  "manual": the
    # hooks only count; finish_gradient_sync() issues every bucket's
    # all-reduce afterwards — used when backward is hipGraph-captured but
    # the RCCL collective must stay outside the graph.
    sync_mode = "hook"

    def _grad_ready_hook(self, param):
        bucket = self._param_bucket[param]
        bucket.pending -= 1
        if (
            bucket.pending == 0
            and self.world_size > 1
            and self.sync_mode == "hook"
            and not self._accumulating
        ):
            bucket.work = dist.all_reduce(
                bucket.buffer, op=dist.ReduceOp.SUM, group=self.group,
                async_op=True,
            )

    def forward(self, *args, **kwargs):
        # arm the countdown lazily: a second grad-enabled forward before
        # finish_gradient_sync (GAN-style multi-forward, grad accumulation)
        # must NOT reset pending mid-flight, or a bucket could all-reduce a
        # partial gradient
        if torch.is_grad_enabled() and not self._require_finish:
            for b in self.buckets:
                b.pending = len(b.params)
                b.work = None
            self._require_finish = True
        return self.module(*args, **kwargs)

    def no_sync(self):
        """Context manager: skip hook-triggered all-reduce (gradient
        accumulation).  The final (synchronizing) step runs outside it."""
        import contextlib

        @contextlib.contextmanager
        def _ctx():
            prev = self._accumulating
            self._accumulating = True
            try:
                yield
            finally:
                self._accumulating = prev

        return _ctx()

    def finish_gradient_sync(self):
        """Call after backward, before optimizer.step()."""
        # manual mode is driven externally (e.g. after a hipGraph replay,
        # where the python-side forward/hooks did not re-run)
        if self.sync_mode != "manual":
            if not self._require_finish:
                return
        self._require_finish = False
        if self.world_size > 1:
            if self.sync_mode == "manual":
                works = [
                    dist.all_reduce(
                        b.buffer, op=dist.ReduceOp.SUM, group=self.group,
                        async_op=True,
                    )
                    for b in self.buckets
                ]
                for w in works:
                    w.wait()
            else:
                for b in self.buckets:
                    if b.work is not None:
                        b.work.wait()
                    else:
                        # the hook never fired the collective: partial
                        # bucket (unused params), or hooks ran under
                        # no_sync / multi-backward accumulation
                        dist.all_reduce(
                            b.buffer, op=dist.ReduceOp.SUM, group=self.group
                        )
                    b.work = None
            if self.average:
                scale = 1.0 / self.world_size
                torch._foreach_mul_(
                    [b.buffer for b in self.buckets], scale
                )

    def zero_grad(self, set_to_none: bool = False):
        # gradients are bucket views: zero the flat buffers (never None them)
        for b in self.buckets:
            b.buffer.zero_()

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)

    @property
    def grad_buffers(self) -> List[torch.Tensor]:
        return [b.buffer for b in self.buckets]


def setup_distributed(backend: Optional[str] = None) -> int:
    """Initialize torch.distributed from torchrun/scheduler env; returns
    local rank.  backend defaults to nccl (=RCCL) on GPU, gloo on CPU."""
    import os

    if dist.is_initialized():
        return int(os.environ.get("LOCAL_RANK", 0))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return local_rank
