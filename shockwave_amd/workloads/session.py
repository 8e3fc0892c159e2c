"""Workload session cache — warm re-dispatch inside a persistent runner.

A *session* is the compiled executor for one workload configuration:
model (device-resident, NHWC), fused optimizer with its metadata cache,
adaptation detectors, and (on GPU) the captured hipGraph of the whole
training step.  Inside a warm runner process
(shockwave_amd/runtime/warm_runner.py) sessions persist across leases,
so re-dispatching a job costs a checkpoint load into EXISTING tensors
instead of model build + MIOpen find + capture (~30 s on MI355X).

Job state and session state are strictly separated: the session is keyed
by configuration (family, batch size, world size, ...), while everything
job-specific — weights, optimizer moments, adaptation state, epoch
counters — rides in the job's checkpoint and is loaded IN PLACE each
lease (tensor addresses never change, so the captured graph stays
valid).  Two different jobs with the same configuration can therefore
share one session: each lease starts by restoring that job's checkpoint,
or, for a job with no checkpoint yet, by restoring the session's initial
weights and zeroing optimizer/adaptation state.

Constraint for graph reuse: a spec that enables capture
(make_static_batch) must not carry per-run tensors through the ``state``
dict in ``spec.step`` (true of the capture-enabled families; LM's hidden
carry is eager-only).

The reference has no counterpart — it re-execs Python per lease and
pays full startup every round (modeled at scheduler.py:1936-1968).
"""

from __future__ import annotations

import logging
import os
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import torch

logger = logging.getLogger("shockwave_amd.session")

# args that change per lease but do not change the compiled executor
VOLATILE_ARGS = {
    "num_steps",
    "checkpoint_dir",
    "enable_gavel_iterator",
    "master_addr",
    "master_port",
    "throughput_estimation_interval",
}


@dataclass
class Session:
    key: tuple
    loader: Any
    model: Any
    model_train: Any
    optimizer: Any
    accordion: Any = None
    gns: Any = None
    graphed: Any = None
    static_batch: Any = None
    capture_lr: Optional[float] = None
    graph_failed: bool = False
    init_model_state: Dict = field(default_factory=dict)
    uses: int = 0


_CACHE: "OrderedDict[tuple, Session]" = OrderedDict()


def _max_sessions() -> int:
    return int(os.environ.get("SWQ_SESSION_CACHE_SIZE", "4"))


def enabled() -> bool:
    return os.environ.get("SWQ_SESSION_CACHE", "1") != "0"


def make_key(family: str, args, mode: str) -> tuple:
    items = tuple(
        sorted(
            (k, repr(v))
            for k, v in vars(args).items()
            if k not in VOLATILE_ARGS
        )
    )
    return (family, mode, items)


def get(key: tuple) -> Optional[Session]:
    sess = _CACHE.get(key)
    if sess is not None:
        _CACHE.move_to_end(key)
        sess.uses += 1
    return sess


def put(sess: Session) -> None:
    _CACHE[sess.key] = sess
    _CACHE.move_to_end(sess.key)
    evicted = False
    while len(_CACHE) > _max_sessions():
        old_key, old = _CACHE.popitem(last=False)
        logger.info("evicting session %s", old_key[:2])
        del old
        evicted = True
    if evicted and torch.cuda.is_available():
        # evicted sessions' blocks return to the allocator pool
        torch.cuda.empty_cache()


def clear() -> None:
    _CACHE.clear()


def snapshot_initial_state(sess: Session) -> None:
    """CPU copy of the freshly-initialized weights; restoring it in
    place is the fresh-job path when a session is reused by a job with
    no checkpoint."""
    sess.init_model_state = {
        k: v.detach().cpu().clone()
        for k, v in sess.model.state_dict().items()
    }


@torch.no_grad()
def reinit_for_fresh_job(sess: Session, args) -> None:
    """Return the session to its initial state IN PLACE (addresses
    stable, graph stays valid)."""
    from ..ops.optim import reset_optimizer_state_inplace

    sess.model.load_state_dict(sess.init_model_state)
    reset_optimizer_state_inplace(sess.optimizer)
    # zero the gradient buffers too (bucket views / preinit grads)
    from . import common

    common.zero_grads(sess.model_train)
    if sess.accordion is not None:
        sess.accordion.reset()
    if sess.gns is not None:
        sess.gns.reset(getattr(args, "batch_size", None))
    if torch.distributed.is_initialized() and (
        torch.distributed.get_world_size() > 1
    ):
        for p in sess.model.parameters():
            torch.distributed.broadcast(p.data, src=0)
        for b in sess.model.buffers():
            torch.distributed.broadcast(b.data, src=0)
