"""LeaseIterator — the lease-preemptible training iterator.

API- and protocol-compatible rebuild of the reference's GavelIterator
(scheduler/gavel_iterator.py:32-361):

* wraps any iterable data loader,
* asks the scheduler for an initial lease on construction and renews it at
  75% of the granted lease (``LEASE_UPDATE_FRACTION``),
* on lease expiry sets ``done``, barriers all ranks of a distributed job,
  and raises StopIteration — the cooperative preemption point,
* self-completes when projected runtime exceeds the deadline,
* writes the structured ``[time] [EVENT] [STATUS] msg`` log under
  ``checkpoint_dir/.gavel/round=R/worker=W.log`` that the dispatcher
  scrapes for (steps, duration),
* ``update_resource_requirement(big_bs, small_bs)`` reports an
  Accordion/GNS batch-size change and triggers checkpoint + restart.

Environment contract (same variable names as the reference so workloads
are interchangeable): GAVEL_JOB_ID, GAVEL_WORKER_ID, GAVEL_ROUND_ID,
GAVEL_SCHED_ADDR, GAVEL_SCHED_PORT.

The scheduler client is injected (``client=``) or constructed from the
environment; ``NullLeaseClient`` grants an infinite lease for standalone
(non-scheduled) runs.
"""

from __future__ import annotations

import atexit
import logging
import os
import time
from collections.abc import Iterable
import torch
from filelock import FileLock

from ..core.lease import Lease

INFINITY = 1e9
LEASE_UPDATE_FRACTION = 0.75
LOG_FORMAT = "[{asctime}] [{event}] [{status}] {message}"
DATE_FORMAT = "%Y-%m-%d %H:%M:%S"


class NullLeaseClient:
    """Infinite lease; used for standalone runs and unit tests."""

    def init(self):
        return INFINITY, INFINITY, 0, 0, INFINITY

    def update_lease(self, steps, duration, max_steps, max_duration):
        return INFINITY, INFINITY, 0, INFINITY

    def update_resource_requirement(self, big_bs, small_bs):
        pass


class LeaseIterator:
    def __init__(
        self,
        data_loader,
        checkpoint_dir,
        load_checkpoint_func,
        save_checkpoint_func,
        synthetic_data=False,
        write_on_close=True,
        verbose=True,
        client=None,
    ):
        if not isinstance(data_loader, Iterable):
            raise ValueError(f"data loader of uniterable type {type(data_loader)}")
        self._data_loader = data_loader
        self._write_on_close = write_on_close
        self._load_checkpoint_func = load_checkpoint_func
        self._save_checkpoint_func = save_checkpoint_func

        self._job_id = int(os.environ.get("GAVEL_JOB_ID", -1))
        self._worker_id = int(os.environ.get("GAVEL_WORKER_ID", 0))
        self._round_id = int(os.environ.get("GAVEL_ROUND_ID", 0))
        self._sched_addr = os.environ.get("GAVEL_SCHED_ADDR")
        self._sched_port = int(os.environ.get("GAVEL_SCHED_PORT", 0))

        os.makedirs(checkpoint_dir, exist_ok=True)
        self._lock = FileLock(os.path.join(checkpoint_dir, ".gavel.lock"))
        self._gavel_dir = os.path.join(checkpoint_dir, ".gavel")
        self._round_dir = os.path.join(
            self._gavel_dir, f"round={self._round_id}"
        )
        with self._lock:
            os.makedirs(self._round_dir, exist_ok=True)
        self._log_file = os.path.join(
            self._round_dir, f"worker={self._worker_id}.log"
        )
        self._init_logger()

        if client is not None:
            self._client = client
        elif self._job_id >= 0 and self._sched_addr:
            from ..rpc.iterator_client import IteratorRpcClient

            self._client = IteratorRpcClient(
                self._job_id, self._worker_id, self._sched_addr,
                self._sched_port, self._logger,
            )
        else:
            self._client = NullLeaseClient()

        atexit.register(self._close_file_handler)
        if self._write_on_close:
            atexit.register(self._write_info)

        self._steps = 0
        self._duration = 0.0
        self._synthetic_data = synthetic_data
        self._initial_val = None
        self._done = False
        self._lease = Lease(0, 0)
        self._steps_until_next_lease_update = INFINITY
        self._time_until_next_lease_update = INFINITY
        self._update_lease(init=True)
        self._write_info()
        # start the renewal countdown immediately: slow shared-storage reads
        # between construction and the first step must count against the
        # lease (reference gavel_iterator.py:96-106)
        self._prev_time = time.time()

    # -- iteration ----------------------------------------------------------

    def __iter__(self):
        self._iterator = iter(self._data_loader)
        return self

    def __len__(self):
        return len(self._data_loader)

    def __next__(self):
        cur_time = time.time()
        if self._prev_time is None:
            self._prev_time = cur_time
        elapsed = cur_time - self._prev_time
        self._duration += elapsed
        self._prev_time = cur_time

        if (
            self._steps_until_next_lease_update <= 0
            or self._time_until_next_lease_update <= 0
        ):
            self._update_lease()

        if (
            self._duration >= self._lease.max_duration
            or self._steps >= self._lease.max_steps
        ):
            self._done = True
            self._logger.info(
                "{0} / {1} steps, {2:.4f} / {3:.4f} seconds".format(
                    self._steps, self._lease.max_steps,
                    self._duration, self._lease.max_duration,
                ),
                extra={"event": "LEASE", "status": "EXPIRED"},
            )
            if torch.distributed.is_initialized():
                torch.distributed.barrier()
            raise StopIteration

        try:
            if self._synthetic_data and self._initial_val is not None:
                val = self._initial_val
            else:
                val = next(self._iterator)
                if self._synthetic_data and self._initial_val is None:
                    self._initial_val = val
            self._steps += 1
        except StopIteration:
            self._write_info()
            raise

        if self._synthetic_data and self._steps % len(self._data_loader) == 0:
            raise StopIteration

        self._steps_until_next_lease_update -= 1
        self._time_until_next_lease_update -= elapsed
        return val

    # -- public API ---------------------------------------------------------

    @property
    def done(self):
        return self._done

    @property
    def steps(self):
        return self._steps

    @property
    def duration(self):
        return self._duration

    def write_progress(self):
        """Flush a PROGRESS STEPS/DURATION pair to the round log (the
        dispatcher and profiling harness scrape these)."""
        self._write_info()
        self._file_handler.flush()

    def close(self):
        """Release the log handler and drop the atexit hooks (the hooks
        are crash insurance; a clean end must not leave one per lease
        piling up inside a long-lived warm runner)."""
        atexit.unregister(self._close_file_handler)
        if self._write_on_close:
            atexit.unregister(self._write_info)
        self._close_file_handler()

    def complete(self, timeout=False):
        self._done = True
        if not self._write_on_close:
            self._write_info()
        self._logger.info("", extra={"event": "LEASE", "status": "COMPLETE"})

    def update_resource_requirement(self, big_bs, small_bs):
        self._done = True  # checkpoint now; scheduler rescales + restarts
        self._client.update_resource_requirement(big_bs, small_bs)

    def load_checkpoint(self, *args, **kwargs):
        self._logger.info("", extra={"event": "LOAD CHECKPOINT", "status": "BEGIN"})
        ckpt = self._load_checkpoint_func(*args, **kwargs)
        self._logger.info("", extra={"event": "LOAD CHECKPOINT", "status": "END"})
        return ckpt

    def save_checkpoint(self, *args, **kwargs):
        self._logger.info("", extra={"event": "SAVE CHECKPOINT", "status": "BEGIN"})
        ret = self._save_checkpoint_func(*args, **kwargs)
        self._logger.info("", extra={"event": "SAVE CHECKPOINT", "status": "END"})
        return ret

    # -- internals ----------------------------------------------------------

    def _init_logger(self):
        logging.getLogger("filelock").setLevel(logging.CRITICAL)
        self._logger = logging.getLogger(
            f"lease_iterator.{self._job_id}.{self._worker_id}.{self._round_id}"
        )
        self._logger.propagate = False
        self._logger.setLevel(logging.DEBUG)
        for h in list(self._logger.handlers):
            self._logger.removeHandler(h)
        self._file_handler = logging.FileHandler(self._log_file)
        self._file_handler.setFormatter(
            logging.Formatter(LOG_FORMAT, datefmt=DATE_FORMAT, style="{")
        )
        self._logger.addHandler(self._file_handler)

    def _write_info(self):
        self._logger.info(
            str(self._steps), extra={"event": "PROGRESS", "status": "STEPS"}
        )
        self._logger.info(
            str(self._duration),
            extra={"event": "PROGRESS", "status": "DURATION"},
        )

    def _close_file_handler(self):
        self._logger.removeHandler(self._file_handler)
        self._file_handler.close()

    def _update_lease(self, init=False):
        if init:
            resp = self._client.init()
            # InitJob carries the full UpdateLeaseResponse (reference
            # iterator_to_scheduler.proto): run_time_so_far + deadline let
            # a restarted job self-complete immediately when it is already
            # over its deadline (legacy 3-tuple clients still accepted)
            if len(resp) == 5:
                max_steps, max_duration, extra_time, run_time, deadline = resp
                if deadline and run_time > deadline:
                    self._logger.info(
                        "run time already exceeds deadline at init",
                        extra={"event": "LEASE", "status": "DEADLINE"},
                    )
                    self.complete(timeout=True)
                    max_steps, max_duration = 0, 0
            else:
                max_steps, max_duration, extra_time = resp
        else:
            (max_steps, max_duration, run_time_so_far, deadline) = (
                self._client.update_lease(
                    self._steps, self._duration,
                    self._lease.max_steps, self._lease.max_duration,
                )
            )
            extra_time = 0
            if self._duration + run_time_so_far > deadline:
                self._logger.info(
                    "projected run time exceeds deadline",
                    extra={"event": "LEASE", "status": "DEADLINE"},
                )
                self.complete(timeout=True)
                raise StopIteration

        # schedule the next renewal at 75% of the NEW lease extension
        if max_steps == self._lease.max_steps:
            self._steps_until_next_lease_update = INFINITY
        else:
            additional = max_steps - self._lease.max_steps
            left = self._lease.max_steps - self._steps
            self._steps_until_next_lease_update = (
                left + additional * LEASE_UPDATE_FRACTION
            )
        if max_duration <= self._lease.max_duration:
            self._time_until_next_lease_update = INFINITY
        else:
            additional = max_duration - self._lease.max_duration
            left = self._lease.max_duration - self._duration
            self._time_until_next_lease_update = (
                left + additional * LEASE_UPDATE_FRACTION + extra_time
            )
        self._lease.max_steps = max_steps
        self._lease.max_duration = max_duration + extra_time


# Alias for drop-in use by reference-style workloads.
GavelIterator = LeaseIterator
