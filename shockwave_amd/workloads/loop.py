"""Generic lease-preemptible training loop.

One implementation of the epoch/lease/checkpoint/adaptation state machine
all families share (the reference repeats it in 21 mains across three
workload trees).  A family provides a ``WorkloadSpec``; the loop handles:

* device + RCCL (or gloo) distributed bring-up,
* checkpoint restore/save (model, optimizer, epoch, steps, adaptation
  state) through the LeaseIterator hooks,
* the Accordion / GNS shared adaptation library and
  ``update_resource_requirement`` reporting,
* per-step gradient sync + fused-optimizer stepping + throughput lines.
"""

from __future__ import annotations

import dataclasses
from typing import Callable, Optional

import torch

from ..adapt import AccordionDetector, GNSEstimator, hardcoded_critical_regime
from ..core import datasets as ds_tables
from . import common


@dataclasses.dataclass
class WorkloadSpec:
    family: str                      # "ResNet-18", "LM", ...
    build_model: Callable            # (args, device) -> nn.Module
    build_loader: Callable           # (args) -> iterable of batches
    build_optimizer: Callable        # (args, params) -> optimizer
    step: Callable                   # (model, batch, device, state) -> loss
    supports_accordion: bool = True
    supports_gns: bool = True
    synthetic_data: bool = True
    extra_state: Callable = None     # (state) -> dict to checkpoint
    restore_state: Callable = None   # (state, dict) -> None
    # hipGraph support: build a device-resident batch with the loader's
    # structure + a copier from a host batch into it.  Static-mode jobs on
    # GPU then run one graph replay per step instead of ~400 eager
    # launches (shockwave_amd/parallel/graphs.py).
    make_static_batch: Callable = None   # (args, device) -> batch
    copy_batch: Callable = None          # (static_batch, batch) -> None


class _Preempted(SystemExit):
    """Raised by the SIGTERM handler: the dispatcher's watchdog kill sends
    SIGTERM 10 s before SIGKILL (dispatcher.kill_job) — enough to
    checkpoint and flush progress so the round's steps are not lost.
    (The reference kills by SIGKILL and loses the round's work.)"""


def run(spec: WorkloadSpec, args, mode: Optional[str] = None, client=None,
        max_steps_override: Optional[int] = None):
    import os
    import signal

    mode = args.mode or mode or os.environ.get("SWQ_MODE", "static")
    device = common.init_device_and_distributed(args)

    # session cache: inside a warm runner the compiled executor (model,
    # fused optimizer, adaptation buffers, captured graph) persists
    # across leases; job identity lives in the checkpoint (session.py)
    from . import session as session_mod

    sess = None
    sess_key = session_mod.make_key(spec.family, args, mode)
    if session_mod.enabled():
        sess = session_mod.get(sess_key)
    if sess is None:
        loader = spec.build_loader(args)
        model = spec.build_model(args, device)
        model_train = common.wrap_distributed(model, args)
        optimizer = spec.build_optimizer(args, model_train.parameters())
        accordion = gns = None
        if mode == "accordion" and spec.supports_accordion:
            accordion = AccordionDetector(model)
        elif mode == "gns" and spec.supports_gns:
            window = args.world_size if args.world_size > 1 else 2
            gns = GNSEstimator(
                model, getattr(args, "batch_size", 1), window=window
            )
        sess = session_mod.Session(
            key=sess_key, loader=loader, model=model,
            model_train=model_train, optimizer=optimizer,
            accordion=accordion, gns=gns,
        )
        session_mod.snapshot_initial_state(sess)
        if session_mod.enabled():
            session_mod.put(sess)
        fresh_session = True
    else:
        loader, model = sess.loader, sess.model
        model_train, optimizer = sess.model_train, sess.optimizer
        accordion, gns = sess.accordion, sess.gns
        fresh_session = False

    trainloader, lease_it = common.make_lease_iterator(
        loader, args, synthetic_data=spec.synthetic_data, client=client
    )

    state = {
        "args": args,
        "device": device,
        "model": model,
        "epoch": 0,
        "cumulative_steps": 0,
        "original_bs": getattr(args, "batch_size", 0),
    }

    ckpt = lease_it.load_checkpoint() if lease_it is not None else None
    if ckpt:
        model.load_state_dict(ckpt["model"])
        try:
            optimizer.load_state_dict(ckpt["optimizer"])
        except (ValueError, KeyError):
            pass
        state["epoch"] = ckpt.get("epoch", 0)
        state["cumulative_steps"] = ckpt.get("cumulative_steps", 0)
        state["original_bs"] = ckpt.get("original_bs", state["original_bs"])
        if accordion is not None and "accordion" in ckpt:
            accordion.load_state_dict(ckpt["accordion"])
        if gns is not None and "gns" in ckpt:
            gns.load_state_dict(ckpt["gns"])
        if spec.restore_state and "extra" in ckpt:
            spec.restore_state(state, ckpt["extra"])
    elif not fresh_session:
        # reused session, but this job has no checkpoint: restore the
        # session's initial weights + zeroed state, in place
        session_mod.reinit_for_fresh_job(sess, args)

    steps_per_epoch = max(1, len(loader))
    target_steps = (
        max_steps_override or args.num_steps or steps_per_epoch * 5
    )
    reporter = common.ThroughputReporter(
        args.throughput_estimation_interval, args.rank
    )

    def save_checkpoint():
        if args.rank != 0 or lease_it is None:
            return
        out = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict(),
            "epoch": state["epoch"],
            "cumulative_steps": state["cumulative_steps"],
            "original_bs": state["original_bs"],
        }
        if accordion is not None:
            out["accordion"] = accordion.state_dict()
        if gns is not None:
            out["gns"] = gns.state_dict()
        if spec.extra_state:
            out["extra"] = spec.extra_state(state)
        lease_it.save_checkpoint(out)

    def maybe_request_rescale(epoch):
        """Epoch-boundary adaptation: always run the detectors (per-layer
        norms / GNS read); report a batch-size change only when scheduled
        under a lease."""
        bs = getattr(args, "batch_size", 0)
        max_bs = ds_tables.max_batch_size(spec.family, bs)
        if accordion is not None:
            accordion.on_epoch(epoch)
            in_cr = hardcoded_critical_regime(
                spec.family, state["original_bs"], epoch + 1
            )
            if lease_it is not None:
                if not in_cr and bs == state["original_bs"] and bs != max_bs:
                    lease_it.update_resource_requirement(True, False)
                    return True
                if in_cr and bs != state["original_bs"]:
                    lease_it.update_resource_requirement(False, True)
                    return True
        if gns is not None:
            gns.on_epoch(epoch)
            if (
                lease_it is not None
                and gns.should_double(epoch)
                and bs < max_bs
            ):
                lease_it.update_resource_requirement(True, False)
                return True
        return False

    model_train.train()

    def eager_step(batch):
        common.zero_grads(model_train)
        loss = spec.step(model_train, batch, device, state)
        if loss is not None and loss.requires_grad:
            loss.backward()
        common.finish_sync(model_train)
        if accordion is not None:
            accordion.on_step()
        if gns is not None:
            gns.on_step()
        optimizer.step()

    # hipGraph the whole compute step.  All three modes are capturable:
    # accordion's multi-tensor accumulate is in-place, and GNS's window is
    # a preallocated address-stable ring (adapt/gns.py) — include the
    # adaptation hooks inside the captured body so one replay is the whole
    # step.  (Epoch-boundary reads — norms, GNS scalar — happen outside.)
    # The captured graph lives in the SESSION: later leases of this
    # configuration replay it directly (capture paid once per config per
    # warm runner).
    graphed = None
    static_batch = None
    # capture costs seconds (warmup + MIOpen find on first process); only
    # worth it when this lease will run enough steps to amortize it
    expected_steps = target_steps - state["cumulative_steps"]
    if lease_it is not None and lease_it._lease.max_steps < 1e9:
        expected_steps = min(expected_steps, lease_it._lease.max_steps)
    if (
        device.type == "cuda"
        and spec.make_static_batch is not None
        and expected_steps >= int(os.environ.get("SWQ_GRAPH_MIN_STEPS", "100"))
        and os.environ.get("SWQ_GRAPHS", "1") != "0"
        and not sess.graph_failed
    ):
        cur_lr = optimizer.param_groups[0]["lr"]
        if sess.graphed is not None:
            if sess.capture_lr == cur_lr:
                graphed, static_batch = sess.graphed, sess.static_batch
            # else: lr changed since capture — run eager (scalars are
            # frozen in the graph; see parallel/graphs.py docstring)
        else:
            from ..parallel.graphs import try_graph_step

            static_batch = spec.make_static_batch(args, device)

            def graph_body():
                common.zero_grads(model_train)
                loss = spec.step(model_train, static_batch, device, state)
                loss.backward()
                common.finish_sync(model_train)
                if accordion is not None:
                    accordion.on_step()
                if gns is not None:
                    gns.on_step()
                optimizer.step()

            # capture warmup trains a few steps on the zero-filled static
            # batch; restore the weights afterwards so the job resumes
            # exactly where its checkpoint left it
            pre = {
                k: v.detach().clone()
                for k, v in model.state_dict().items()
            }
            # GNS: the ring-fill and EMA-init branches must reach steady
            # state before capture (gns.py docstring): window+1 warmups
            warmup = 3 if gns is None else max(3, gns.window + 1)
            graphed = try_graph_step(lambda: graph_body(), [],
                                     warmup_iters=warmup)
            if args.world_size > 1:
                from ..parallel.graphs import agree_capture

                graphed = agree_capture(graphed)
            with torch.no_grad():
                model.load_state_dict(pre)
            sess.graphed, sess.static_batch = graphed, static_batch
            sess.capture_lr = cur_lr
            sess.graph_failed = graphed is None

    def _on_sigterm(signum, frame):
        raise _Preempted(0)

    prev_sigterm = None
    try:
        prev_sigterm = signal.signal(signal.SIGTERM, _on_sigterm)
    except ValueError:
        pass  # not the main thread (in-process tests)

    done = False
    try:
        while not done and state["cumulative_steps"] < target_steps:
            hit_target = False
            for batch in trainloader:
                if graphed is not None:
                    spec.copy_batch(static_batch, batch)
                    graphed.replay()
                else:
                    eager_step(batch)
                state["cumulative_steps"] += 1
                reporter.step()
                if state["cumulative_steps"] >= target_steps:
                    hit_target = True
                    break
            if hit_target or (lease_it is not None and lease_it.done):
                break
            if maybe_request_rescale(state["epoch"]):
                done = True
            state["epoch"] += 1
    except _Preempted:
        # watchdog kill: fall through to checkpoint + progress flush
        # below, then exit cleanly inside the dispatcher's SIGTERM window
        pass
    finally:
        if prev_sigterm is not None:
            try:
                signal.signal(signal.SIGTERM, prev_sigterm)
            except ValueError:
                pass

    if lease_it is not None:
        save_checkpoint()
        if state["cumulative_steps"] >= target_steps and not lease_it.done:
            lease_it.complete()
        lease_it.write_progress()
        lease_it.close()
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    return state["cumulative_steps"]
