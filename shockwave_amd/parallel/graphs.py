"""hipGraph capture of the training step.

MI355X-first replacement for a tracing compiler: the flagship jobs
(ResNet-18 @ small batch, LSTM LM) are launch-bound — a bs-16 ResNet-18
step issues ~400 kernels totalling <3 ms of GPU work but >30 ms of wall
from eager dispatch.  Capturing fwd+bwd+fused-optimizer in one hipGraph
collapses that to a single ~10-16 us replay plus kernel time.

Works because the rest of the stack keeps addresses stable:
* gradients are zeroed in place (never set to None),
* the fused optimizer's tensor metadata is cached on device,
* DDP gradients live in persistent flat bucket buffers.

Scalars captured in the graph (e.g. the learning rate passed to the fused
kernel) are frozen; call ``recapture()`` after changing them (the
workloads' epoch-level LR schedules re-capture at epoch boundaries).
"""

from __future__ import annotations

from typing import Callable, Sequence

import torch


class GraphedTrainStep:
    """Capture ``step_fn(*static_inputs)`` into a hipGraph.

    ``static_inputs`` are device tensors owned by this object; callers
    copy fresh data into them (``copy_inputs``) and ``replay()``.
    """

    def __init__(
        self,
        step_fn: Callable,
        static_inputs: Sequence[torch.Tensor],
        warmup_iters: int = 3,
        pool=None,
    ):
        assert torch.cuda.is_available(), "hipGraph capture requires a GPU"
        self.step_fn = step_fn
        self.static_inputs = list(static_inputs)

        # warm up in a side stream (allocator + MIOpen algo selection)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self.step_fn(*self.static_inputs)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        self._pool = pool
        self._capture()

    def _capture(self):
        with torch.cuda.graph(self.graph, pool=self._pool):
            self.out = self.step_fn(*self.static_inputs)

    def recapture(self):
        self.graph.reset()
        self._capture()

    def copy_inputs(self, *tensors: torch.Tensor):
        for dst, src in zip(self.static_inputs, tensors):
            dst.copy_(src, non_blocking=True)

    def replay(self):
        self.graph.replay()
        return self.out

    def __call__(self, *tensors: torch.Tensor):
        self.copy_inputs(*tensors)
        return self.replay()


def agree_capture(graphed, group=None):
    """Collective agreement on hipGraph use at world>1.

    Capture success is timing-dependent per rank (the RCCL watchdog
    polling another rank's in-flight work can invalidate a capture), but
    every rank must run the SAME per-step collective schedule: a full
    graph replays its captured RCCL ops while the fallbacks issue them
    eagerly, and mixing the two across ranks deadlocks the communicator.
    MIN-reduce a success flag and keep the graph only if every rank
    captured; the discarded graph is simply never replayed (no real
    collective ran during its capture), so discarding is safe.

    Returns ``graphed`` when all ranks captured, else ``None``.
    """
    import torch.distributed as dist

    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return graphed
    backend = dist.get_backend(group)
    device = torch.device("cuda") if backend == "nccl" else torch.device("cpu")
    ok = torch.tensor([1.0 if graphed is not None else 0.0], device=device)
    dist.all_reduce(ok, op=dist.ReduceOp.MIN, group=group)
    if ok.item() < 1.0:
        if graphed is not None:
            import logging

            logging.getLogger("shockwave_amd.graphs").warning(
                "hipGraph captured here but failed on a peer rank; "
                "discarding for a uniform collective schedule"
            )
        return None
    return graphed


def try_graph_step(step_fn, static_inputs, warmup_iters=3):
    """Capture if possible; return None when capture is unsupported for
    this step (e.g. RCCL build without graph support) so callers keep the
    eager path."""
    try:
        return GraphedTrainStep(step_fn, static_inputs, warmup_iters)
    except Exception as e:  # capture failure must not kill the job
        import logging

        logging.getLogger("shockwave_amd.graphs").warning(
            "hipGraph capture failed (%s); staying eager", e
        )
        return None
