"""Physical-cluster scheduler: the head-node process.

Extends the RoundScheduler state machine with the live control plane
(reference scheduler.py physical path):

* gRPC server for worker registration / Done / lease RPCs
  (:3782-3878, 4341-4729, 3880-4200),
* the round loop thread — begin round, sleep to 50%, compute next round's
  schedule, extend leases for jobs keeping their workers, dispatch
  non-extended jobs for the next round, schedule watchdog completion
  events, wait for all round jobs, sleep out the round (:2382-2778),
* job dispatch with DP rendezvous args (master addr/port from the first
  worker; ports allocated from BASE_JOB_PORT) (:2494-2574),
* lease callbacks — init lease sized to the remaining round (+extra time
  for early dispatch), renewal with extension, multi-GPU first-requester
  max-steps computation with spin-wait (:3880-4200),
* watchdogs: kill + synthesized zero-step done callbacks for unresponsive
  jobs, extended-lease health checks (:4201-4339).
"""

from __future__ import annotations

import copy
import logging
import math
import sched as sched_module
import threading
import time
from collections import OrderedDict
from concurrent.futures import ThreadPoolExecutor

import numpy as np

from ..rpc.services import SchedulerRpcClient, serve_scheduler
from ..runtime.set_queue import SetQueue
from .scheduler import (
    BASE_JOB_PORT,
    EARLY_INIT_THRESHOLD,
    INFINITY,
    JOB_COMPLETION_BUFFER_TIME,
    MAX_FAILED_ATTEMPTS,
    MAX_PORT,
    SCHEDULE_RECOMPUTE_FRACTION,
    RoundScheduler,
)

logger = logging.getLogger("shockwave_amd.engine.physical")

# Rounds an extended-lease job may go without a renewal before it is
# declared dead and killed.  The reference kills after ONE silent round
# (:4283-4339), safe at its 360 s rounds; at short rounds (tests, small
# clusters) a renewal can land just past a round boundary — the 75%
# renewal cadence gives ~0.75x round between renewals, plus process
# startup — and killing on the first miss can livelock into a
# kill/redispatch cycle that never accumulates steps.  Two silent rounds
# cannot happen while a job is alive and renewing.
EXTENDED_LEASE_GRACE_ROUNDS = 2


class PhysicalScheduler(RoundScheduler):
    def __init__(
        self,
        policy,
        port: int = 50070,
        expected_num_workers: int = None,
        completion_buffer_s: float = JOB_COMPLETION_BUFFER_TIME,
        heartbeat_timeout_s: float = 90.0,
        **kwargs,
    ):
        super().__init__(policy, simulate=False, **kwargs)
        self._port = port
        self._expected_num_workers = expected_num_workers
        self._completion_buffer_s = completion_buffer_s
        # worker liveness: deregister after this long without a heartbeat
        # (workers beat every ~30 s); None disables the check
        self._heartbeat_timeout_s = heartbeat_timeout_s
        self._last_heartbeat = {}

        self._scheduler_lock = threading.RLock()
        self._scheduler_cv = threading.Condition(self._scheduler_lock)
        self._available_worker_ids = SetQueue()
        self._worker_connections = {}
        self._worker_addrs = {}
        self._port_offset = 0
        self._current_round_start_time = None
        self._next_worker_assignments = None
        self._redispatched_worker_assignments = OrderedDict()
        self._unresponsive_rounds = {}
        self._completed_jobs_in_current_round = set()
        self._completion_events = {}
        self._completion_event_scheduler = sched_module.scheduler(
            time.time, time.sleep
        )
        self._shutdown_event = threading.Event()

        # hang diagnosis: periodic all-thread stack dumps (reference
        # faulthandler.dump_traceback_later hook, scheduler.py:450-455)
        from ..utils.logging import enable_hang_diagnosis

        try:
            self._cancel_hang_diagnosis = enable_hang_diagnosis(
                ".stack_trace.log", interval_s=30.0
            )
        except OSError:
            self._cancel_hang_diagnosis = lambda: None

        self._server = serve_scheduler(
            port,
            {
                "RegisterWorker": self._register_worker_callback,
                "SendHeartbeat": self._heartbeat_callback,
                "Done": self._done_callback,
                "InitJob": self._init_job_callback,
                "UpdateLease": self._update_lease_callback,
                "UpdateResourceRequirement": self._update_resource_requirement_callback,
            },
        )
        self._mechanism_thread = threading.Thread(
            target=self._schedule_with_rounds, daemon=True
        )
        self._mechanism_thread.start()

    # ------------------------------------------------------------------
    # worker registration (reference :3782-3878)
    # ------------------------------------------------------------------

    def _register_worker_callback(self, worker_type, num_gpus, ip_addr, port):
        with self._scheduler_cv:
            rpc_client = SchedulerRpcClient(ip_addr, port)
            worker_ids = self.register_worker(worker_type, num_gpus=num_gpus)
            now = time.time()
            for worker_id in worker_ids:
                self._worker_connections[worker_id] = rpc_client
                self._worker_addrs[worker_id] = (ip_addr, port)
                self._last_heartbeat[worker_id] = now
            self._scheduler_cv.notify_all()
            return worker_ids, self._time_per_iteration

    # ------------------------------------------------------------------
    # worker liveness (exceeds the reference: its SendHeartbeat is a no-op
    # and a dead worker's jobs churn through watchdog kills forever)
    # ------------------------------------------------------------------

    def _heartbeat_callback(self, worker_ids):
        now = time.time()
        with self._scheduler_lock:
            for wid in worker_ids:
                if wid in self._worker_connections:
                    self._last_heartbeat[wid] = now

    def _check_worker_liveness(self):
        """Called at round start (lock held): deregister workers whose
        heartbeats stopped; their in-flight jobs fall to the watchdog
        (synthesized zero-step done) and are rescheduled elsewhere."""
        if self._heartbeat_timeout_s is None:
            return
        now = time.time()
        for wid in list(self._worker_connections):
            last = self._last_heartbeat.get(wid)
            if last is not None and now - last > self._heartbeat_timeout_s:
                self._deregister_worker(wid)

    def _deregister_worker(self, worker_id):
        logger.error(
            "worker %d missed heartbeats for >%ss; deregistering",
            worker_id, self._heartbeat_timeout_s,
        )
        if not self.deregister_worker(worker_id):
            return
        self._worker_connections.pop(worker_id, None)
        self._worker_addrs.pop(worker_id, None)
        self._last_heartbeat.pop(worker_id, None)
        try:
            self._available_worker_ids.get_nowait(item=worker_id)
        except Exception:
            pass
        if self.is_shockwave:
            self._shockwave_planner.ngpus = max(
                1, sum(self._cluster_spec.values())
            )

    def add_job(self, job, timestamp=None):
        with self._scheduler_cv:
            job_id = super().add_job(job, timestamp)
            self._scheduler_cv.notify_all()
            return job_id

    # ------------------------------------------------------------------
    # round loop (reference :2710-2778)
    # ------------------------------------------------------------------

    def _schedule_with_rounds(self):
        try:
            self._schedule_with_rounds_inner()
        except Exception:
            logger.exception("mechanism thread crashed")
            self._shutdown_event.set()

    def _schedule_with_rounds_inner(self):
        with self._scheduler_cv:
            while len(self._jobs) == 0 or (
                self._expected_num_workers is not None
                and len(self._worker_ids) < self._expected_num_workers
            ):
                self._scheduler_cv.wait(timeout=1)
                if self._shutdown_event.is_set():
                    return
            for worker_id in self._worker_ids:
                self._available_worker_ids.put(worker_id)
            if not self.is_shockwave:
                self._allocation = self._compute_allocation()
                self._need_to_update_allocation = False
            self._current_worker_assignments = self._schedule_jobs_on_workers()
            if self.is_shockwave:
                self._shockwave_planner.increment_round_ptr()
            for job_id, worker_ids in self._current_worker_assignments.items():
                self._try_dispatch_job(job_id, worker_ids)

        with ThreadPoolExecutor(max_workers=1) as pool:
            while not self._shutdown_event.is_set():
                is_final_round = (
                    self._max_rounds is not None
                    and self._num_completed_rounds + 1 == self._max_rounds
                )
                with self._scheduler_cv:
                    self._begin_round()
                time.sleep(
                    self._time_per_iteration * SCHEDULE_RECOMPUTE_FRACTION
                )
                with self._scheduler_cv:
                    self._mid_round(pool, is_final_round)
                    if self.is_shockwave:
                        jobs_with_extended_lease = copy.deepcopy(
                            self._jobs_with_extended_lease
                        )
                    self._end_round(is_final_round)
                    if self.is_shockwave:
                        self._update_shockwave_planner(
                            jobs_with_extended_lease=jobs_with_extended_lease
                        )
                if is_final_round or (
                    len(self._jobs) == 0 and self._job_id_counter > 0
                ):
                    break
        logger.info("round loop finished")
        self._shutdown_event.set()

    def _begin_round(self):
        self._check_worker_liveness()
        self._current_round_start_time = self.get_current_timestamp()
        for job_id in self._current_worker_assignments:
            for single in job_id.singletons():
                self._lease_update_requests[single] = []
                self._max_steps[single] = None
        for job_id, worker_ids in self._redispatched_worker_assignments.items():
            if any(x in self._jobs for x in job_id.singletons()):
                if job_id not in self._current_worker_assignments:
                    raise RuntimeError(
                        f"re-dispatching {job_id} but it is not scheduled"
                    )
                logger.info("re-dispatching early-completed job %s", job_id)
                self._try_dispatch_job(job_id, worker_ids)
        self._redispatched_worker_assignments = OrderedDict()
        logger.info("*** START ROUND %d ***", self._num_completed_rounds)

    def _mid_round(self, pool, is_final_round):
        if is_final_round:
            self._jobs_with_extended_lease = set()
            return
        round_end_time = (
            self._current_round_start_time + self._time_per_iteration
        )
        if not self.is_shockwave:
            self._update_priorities()
            self._allocation = self._compute_allocation()
        self._next_worker_assignments = self._schedule_jobs_on_workers()

        for job_id in self._current_worker_assignments:
            if any(x in self._jobs for x in job_id.singletons()):
                self._num_lease_extension_opportunities += 1

        for job_id in self._current_worker_assignments:
            current = set(self._current_worker_assignments[job_id])
            if (
                job_id in self._next_worker_assignments
                and job_id not in self._completed_jobs_in_current_round
            ):
                nxt = set(self._next_worker_assignments[job_id])
                if current == nxt:
                    self._jobs_with_extended_lease.add(job_id)
                    self._num_lease_extensions += 1
                else:
                    self._jobs_with_extended_lease.discard(job_id)
            else:
                self._jobs_with_extended_lease.discard(job_id)

        for job_id, worker_ids in self._next_worker_assignments.items():
            if not any(x in self._jobs for x in job_id.singletons()):
                continue
            if job_id not in self._jobs_with_extended_lease or (
                job_id in self._completed_jobs_in_current_round
            ):
                self._try_dispatch_job(job_id, worker_ids, next_round=True)

        self._schedule_completion_events(round_end_time, pool)

    def _end_round(self, is_final_round):
        current_round = self._num_completed_rounds
        jobs_to_complete = {
            job_id
            for job_id in self._current_worker_assignments
            if any(x in self._jobs for x in job_id.singletons())
        }
        while not jobs_to_complete.issubset(
            self._completed_jobs_in_current_round
        ):
            self._scheduler_cv.wait(timeout=5)
            if self._shutdown_event.is_set():
                return
            jobs_to_complete = {
                j for j in jobs_to_complete
                if any(x in self._jobs for x in j.singletons())
            }
        if self._completion_events:
            logger.warning(
                "completion events still pending at end of round: %s",
                list(self._completion_events),
            )
            self._completion_events.clear()

        for job_id in list(self._jobs_with_extended_lease):
            if job_id in self._jobs:
                for worker_id in self._current_worker_assignments[job_id]:
                    if worker_id in self._worker_connections:
                        self._available_worker_ids.put(worker_id)
            self._jobs_with_extended_lease.discard(job_id)

        if not is_final_round:
            if self._next_worker_assignments is None:
                raise RuntimeError("next worker assignments not computed")
            now = self.get_current_timestamp()
            round_end_time = (
                self._current_round_start_time + self._time_per_iteration
            )
            remaining = round_end_time - now
            if remaining > 0:
                self._scheduler_cv.release()
                try:
                    time.sleep(remaining)
                finally:
                    self._scheduler_cv.acquire()

        self._num_completed_rounds += 1
        self._completed_jobs_in_current_round = set()
        self._current_worker_assignments = (
            self._next_worker_assignments or OrderedDict()
        )
        self._next_worker_assignments = None
        self._scheduler_cv.notify_all()
        logger.info("*** END ROUND %d ***", current_round)

    # ------------------------------------------------------------------
    # dispatch (reference :2494-2606)
    # ------------------------------------------------------------------

    def _try_dispatch_job(self, job_id, worker_ids, next_round=False):
        if not next_round or job_id not in self._current_worker_assignments:
            self._in_progress_updates[job_id] = []
            for single in job_id.singletons():
                self._lease_update_requests[single] = []
                self._max_steps[single] = None

        scale_factor = len(worker_ids)
        if scale_factor > 1:
            master_addr = self._worker_addrs[worker_ids[0]][0]
            master_job_ports = []
            for _ in job_id.singletons():
                master_job_ports.append(BASE_JOB_PORT + self._port_offset)
                self._port_offset = (self._port_offset + 1) % (
                    MAX_PORT - BASE_JOB_PORT
                )

        current_round = self._num_completed_rounds + (1 if next_round else 0)
        for i, worker_id in enumerate(worker_ids):
            job_descriptions = []
            for j, single in enumerate(job_id.singletons()):
                job = self._jobs[single]
                command = job.command
                if scale_factor > 1:
                    command = (
                        f"{command} --master_addr {master_addr} "
                        f"--master_port {master_job_ports[j]} "
                        f"--world_size {scale_factor} --rank {i}"
                    )
                job_descriptions.append(
                    {
                        "job_id": single[0],
                        "job_type": job.job_type,
                        "command": command,
                        "working_directory": job.working_directory,
                        "needs_data_dir": job.needs_data_dir,
                        "num_steps_arg": job.num_steps_arg,
                        "num_steps": job.total_steps,
                        "mode": job.mode,
                        "mps_thread_percentage": job.mps_thread_percentage,
                    }
                )
            conn = self._worker_connections.get(worker_id)
            if conn is None:
                logger.warning(
                    "not dispatching %s to deregistered worker %d",
                    job_id, worker_id,
                )
                continue
            try:
                conn.run_job(job_descriptions, worker_id, current_round)
            except Exception:
                logger.warning(
                    "RunJob RPC to worker %d failed", worker_id,
                    exc_info=True,
                )
                continue
            if not next_round:
                try:
                    self._available_worker_ids.get_nowait(item=worker_id)
                except KeyError:
                    pass

    def _schedule_completion_events(self, round_end_time, pool):
        now = self.get_current_timestamp()
        for job_id in self._current_worker_assignments:
            if (
                not any(x in self._jobs for x in job_id.singletons())
                or job_id in self._completed_jobs_in_current_round
            ):
                continue
            delay = round_end_time - now
            if job_id not in self._jobs_with_extended_lease:
                delay += self._completion_buffer_s
                action = self._kill_job
            else:
                action = self._done_callback_extended_lease
            event = self._completion_event_scheduler.enter(
                delay=max(0.0, delay), priority=1, action=action,
                argument=(job_id,),
            )
            self._completion_events[job_id] = event
        try:
            pool.submit(self._completion_event_scheduler.run)
        except RuntimeError:
            # pool already closed: only happens while shutting down
            if not self._shutdown_event.is_set():
                raise

    # ------------------------------------------------------------------
    # lease callbacks (reference :3880-4200)
    # ------------------------------------------------------------------

    def _init_job_callback(self, job_id):
        with self._scheduler_cv:
            if job_id not in self._jobs:
                return (0, 0, 0, 0, 0)
            # wait while the job is dispatched for the NEXT round but its
            # workers are still running the current round
            while True:
                next_combination = None
                if self._next_worker_assignments is not None:
                    for combo in self._next_worker_assignments:
                        if job_id.overlaps_with(combo):
                            next_combination = combo
                            break
                currently_active = False
                if next_combination is not None:
                    for combo in self._current_worker_assignments:
                        for single in next_combination.singletons():
                            if single.overlaps_with(combo) and (
                                combo
                                not in self._completed_jobs_in_current_round
                            ):
                                currently_active = True
                                break
                        if currently_active:
                            break
                if currently_active and next_combination is not None:
                    self._scheduler_cv.wait(timeout=5)
                    if self._shutdown_event.is_set():
                        return (0, 0, 0, 0, 0)
                else:
                    break

            self._per_job_latest_timestamps[job_id] = (
                self.get_current_timestamp()
            )
            for single in job_id.singletons():
                self._running_jobs.add(single)

            scale_factor = self._jobs[job_id].scale_factor
            remaining_steps = int(
                math.ceil(self._get_remaining_steps(job_id) / scale_factor)
            )
            # real run_time_so_far + deadline in the InitJob response
            # (reference UpdateLeaseResponse; r1 stubbed these to 0)
            run_time_so_far = int(
                sum(self._cumulative_run_time.get(job_id, {}).values())
                / scale_factor
            )
            deadline = int(self._jobs[job_id].duration * 1.5)
            now = self.get_current_timestamp()
            if self._current_round_start_time is None:
                return (remaining_steps, self._time_per_iteration, 0,
                        run_time_so_far, deadline)
            round_end = (
                self._current_round_start_time + self._time_per_iteration
            )
            remaining_time = max(round_end - now, 0)

            if (
                self._next_worker_assignments is not None
                and next_combination is not None
            ):
                # early dispatch for next round: full round + extra time
                return (remaining_steps, self._time_per_iteration,
                        remaining_time, run_time_so_far, deadline)
            if remaining_time > 0:
                return (remaining_steps, remaining_time, 0,
                        run_time_so_far, deadline)
            return (
                remaining_steps,
                self._time_per_iteration - EARLY_INIT_THRESHOLD,
                remaining_time,
                run_time_so_far,
                deadline,
            )

    def _update_lease_callback(
        self, job_id, worker_id, steps, duration, max_steps, max_duration
    ):
        with self._scheduler_lock:
            if job_id not in self._jobs:
                return (max_steps, max_duration, 0, int(1e9))
            run_time_so_far = int(
                sum(self._cumulative_run_time[job_id].values())
                / self._jobs[job_id].scale_factor
            )
            deadline = int(self._jobs[job_id].duration * 1.5)
            self._lease_update_requests.setdefault(job_id, [])
            update_id = len(self._lease_update_requests[job_id])
            self._lease_update_requests[job_id].append(
                (steps, duration, max_steps, max_duration)
            )
            scale_factor = self._jobs[job_id].scale_factor
            remaining_steps = int(
                math.ceil(self._get_remaining_steps(job_id) / scale_factor)
            )
            now = self.get_current_timestamp()
            round_end = (
                self._current_round_start_time + self._time_per_iteration
            )
            remaining_time = max(0, round_end - now)

            # epoch-progress bookkeeping for the planner (aggregate steps)
            self._steps_run_in_current_lease[job_id] = (
                steps * scale_factor
            )

        if steps == 0 or duration == 0:
            return (remaining_steps, remaining_time, run_time_so_far, deadline)

        with self._scheduler_lock:
            for combo in self._jobs_with_extended_lease:
                if job_id.overlaps_with(combo):
                    return (
                        max_steps,
                        duration + remaining_time + self._time_per_iteration,
                        run_time_so_far,
                        deadline,
                    )

        if scale_factor == 1:
            return (
                max_steps,
                duration + remaining_time,
                run_time_so_far,
                deadline,
            )

        # multi-GPU: first requester computes the shared max_steps
        if update_id == 0:
            with self._scheduler_lock:
                throughput = steps / duration
                self._max_steps[job_id] = min(
                    remaining_steps,
                    steps + int(remaining_time * throughput),
                )
                return (
                    self._max_steps[job_id], INFINITY, run_time_so_far, deadline
                )
        deadline_t = time.time() + 60
        while time.time() < deadline_t:
            with self._scheduler_lock:
                ms = self._max_steps.get(job_id)
            if ms is not None:
                return (ms, INFINITY, run_time_so_far, deadline)
            time.sleep(0.5)
        return (max_steps, max_duration, run_time_so_far, deadline)

    def _update_resource_requirement_callback(
        self, job_id, worker_id, big_bs, small_bs
    ):
        with self._scheduler_cv:
            assert big_bs != small_bs
            if job_id not in self._bs_flags:
                return
            if big_bs:
                self._bs_flags[job_id]["big_bs"] = True
            else:
                self._bs_flags[job_id]["small_bs"] = True
            self._scheduler_cv.notify_all()

    # ------------------------------------------------------------------
    # watchdogs (reference :4201-4339)
    # ------------------------------------------------------------------

    def _kill_job(self, job_id):
        with self._scheduler_cv:
            if job_id not in self._current_worker_assignments:
                logger.warning("kill for inactive job %s ignored", job_id)
                return
            if job_id not in self._completion_events:
                if job_id in self._completed_jobs_in_current_round:
                    return
            logger.info("killing job %s", job_id)
            worker_ids = self._current_worker_assignments[job_id]
            servers = set()
            for worker_id in worker_ids:
                client = self._worker_connections.get(worker_id)
                if client is None:
                    continue  # worker deregistered (liveness)
                key = (client.addr, client.port)
                if key not in servers:
                    for single in job_id.singletons():
                        try:
                            client.kill_job(single[0])
                        except Exception:
                            logger.warning(
                                "kill RPC to worker %d failed (dead?)",
                                worker_id, exc_info=True,
                            )
                    servers.add(key)
            self._completion_events.pop(job_id, None)

            prev_round = self._num_completed_rounds
            self._scheduler_cv.wait(timeout=30)
            successful = (
                self._num_completed_rounds != prev_round
                or job_id in self._completed_jobs_in_current_round
            )
            if successful:
                return
            all_ids = set(self._current_worker_assignments[job_id])
            done_ids = {u[0] for u in self._in_progress_updates[job_id]}
            to_complete = all_ids - done_ids
        zeros = [0 for _ in job_id.singletons()]
        for worker_id in to_complete:
            self._done_callback(job_id, worker_id, zeros, zeros)

    def _done_callback_extended_lease(self, job_id):
        kill = False
        with self._scheduler_cv:
            if not any(x in self._jobs for x in job_id.singletons()):
                return
            scale_factor = self._jobs[job_id.singletons()[0]].scale_factor
            num_updates = [
                len(self._lease_update_requests.get(s, []))
                for s in job_id.singletons()
            ]
            if min(num_updates) < scale_factor:
                misses = self._unresponsive_rounds.get(job_id, 0) + 1
                self._unresponsive_rounds[job_id] = misses
                if misses >= EXTENDED_LEASE_GRACE_ROUNDS:
                    logger.error(
                        "job %s held an extended lease but was unresponsive "
                        "for %d consecutive rounds",
                        job_id, misses,
                    )
                    kill = True
                else:
                    logger.warning(
                        "job %s: no lease renewal this round (%d/%d before "
                        "kill)",
                        job_id, misses, EXTENDED_LEASE_GRACE_ROUNDS,
                    )
                    # tolerate this round: mark the job round-complete so
                    # _end_round can proceed
                    if job_id in self._completion_events:
                        self._completed_jobs_in_current_round.add(job_id)
                        del self._completion_events[job_id]
                    for single in job_id.singletons():
                        self._lease_update_requests[single] = []
                        self._max_steps[single] = None
            elif job_id in self._completion_events:
                self._unresponsive_rounds.pop(job_id, None)
                self._completed_jobs_in_current_round.add(job_id)
                del self._completion_events[job_id]
                for single in job_id.singletons():
                    self._lease_update_requests[single] = []
                    self._max_steps[single] = None
            if not kill:
                self._scheduler_cv.notify_all()
        if kill:
            self._unresponsive_rounds.pop(job_id, None)
            self._kill_job(job_id)

    # ------------------------------------------------------------------
    # done callback (physical wrapper around the shared accounting)
    # ------------------------------------------------------------------

    def _done_callback(
        self, job_id, worker_id, all_num_steps, all_execution_times,
        all_iterator_logs=None,
    ):
        with self._scheduler_cv:
            # wait when the notification arrives for a round that has not
            # started yet (job dispatched for round r+1, finished r early)
            deadline = time.time() + 60
            while (
                job_id not in self._current_worker_assignments
                or job_id in self._completed_jobs_in_current_round
            ):
                if job_id not in self._current_worker_assignments and (
                    self._next_worker_assignments is not None
                    and job_id not in self._next_worker_assignments
                ):
                    logger.warning(
                        "discarding completion for unscheduled job %s", job_id
                    )
                    return
                self._scheduler_cv.wait(timeout=5)
                if time.time() > deadline or self._shutdown_event.is_set():
                    logger.warning("timed out waiting to complete %s", job_id)
                    return

            is_active = {s: s in self._jobs for s in job_id.singletons()}
            if not any(is_active.values()):
                return

            # per-worker cumulative run time (reference :4374-4380)
            self._cumulative_run_time.setdefault(job_id, {})
            self._cumulative_run_time[job_id].setdefault(worker_id, 0.0)
            self._cumulative_run_time[job_id][worker_id] += float(
                np.max(all_execution_times)
            )

            if worker_id in self._worker_connections:
                self._available_worker_ids.put(worker_id)
            scale_factor = len(self._current_worker_assignments[job_id])
            self._in_progress_updates[job_id].append(
                (worker_id, all_num_steps, all_execution_times,
                 all_iterator_logs)
            )
            if len(self._in_progress_updates[job_id]) < scale_factor:
                return

            # all workers reported: cancel the watchdog
            if job_id in self._completion_events:
                event = self._completion_events.pop(job_id)
                try:
                    self._completion_event_scheduler.cancel(event)
                except ValueError:
                    pass
            self._completed_jobs_in_current_round.add(job_id)

            run_time_so_far = (
                sum(self._cumulative_run_time[job_id].values()) / scale_factor
            )
            lead = job_id.singletons()[0]
            is_over_deadline = (
                lead in self._jobs
                and run_time_so_far > int(self._jobs[lead].duration * 1.5)
            )

            self._in_progress_updates[job_id].sort(key=lambda x: x[0])
            micro_task_succeeded = True
            agg_steps = [0] * len(job_id.singletons())
            agg_times = [0.0] * len(job_id.singletons())
            for i, (wid, steps_, times_, logs_) in enumerate(
                self._in_progress_updates[job_id]
            ):
                for j, single in enumerate(job_id.singletons()):
                    if is_active[single] and steps_[j] <= 0 and times_[j] <= 0:
                        micro_task_succeeded = False
                    agg_steps[j] += steps_[j]
                    agg_times[j] = max(agg_times[j], times_[j])
                    if logs_ is not None and logs_[j]:
                        self._job_timelines[single][i].extend(
                            logs_[j].split("\n")
                        )
            all_worker_ids = sorted(
                u[0] for u in self._in_progress_updates[job_id]
            )
            self._in_progress_updates[job_id] = []
            for single in job_id.singletons():
                self._lease_update_requests[single] = []
                self._max_steps[single] = None
                if is_active[single]:
                    self._per_job_latest_timestamps[single] = (
                        self.get_current_timestamp()
                    )

            to_remove = []
            worker_type = self._worker_id_to_worker_type_mapping[worker_id]
            if not micro_task_succeeded:
                logger.info("[Micro-task failed] job %s", job_id)
                if not job_id.is_pair() and is_active[job_id]:
                    if is_over_deadline:
                        # zero-step report from a job already past its
                        # deadline (e.g. the iterator self-completed at
                        # init): force-complete instead of counting a
                        # failure (reference deadline handling :4392-4401)
                        to_remove.append(job_id)
                    else:
                        self._num_failures_per_job[job_id] += 1
                        if (
                            self._num_failures_per_job[job_id]
                            >= MAX_FAILED_ATTEMPTS
                        ):
                            to_remove.append(job_id)
                self._need_to_update_allocation = True
            else:
                self._num_failures_per_job[job_id] = 0
                for single, steps, exec_time in zip(
                    job_id.singletons(), agg_steps, agg_times
                ):
                    if not is_active[single]:
                        continue
                    if single in self._running_jobs:
                        self._running_jobs.remove(single)
                        self._steps_run_so_far[single][worker_type] += steps
                        self._total_steps_run[single] += steps
                        self._steps_run_in_current_lease[single] = 0
                        if (
                            self._get_remaining_steps(single) <= 0
                            or is_over_deadline
                        ):
                            to_remove.append(single)
                max_exec = float(np.max(agg_times))
                if job_id in self._job_time_so_far:
                    self._job_time_so_far[job_id][worker_type] += max_exec
                    self._worker_time_so_far[worker_type] += max_exec
                for wid in all_worker_ids:
                    self._cumulative_worker_time_so_far[wid] += max_exec
                self._accrue_cost(job_id, worker_type, max_exec)

            self._update_throughput(
                job_id, worker_type, agg_steps, agg_times
            )
            for single in job_id.singletons():
                self._scale_bs_and_iters(single)
            for single in to_remove:
                logger.info("[Job completed] %s", single)
                self._remove_job(single)
                if (
                    self.is_shockwave
                    and single[0] in self._shockwave_planner.metadata
                ):
                    self._shockwave_planner.remove_metadata(single[0])
                    self._shockwave_job_completed_flag = True

            # re-dispatch if the job holds an extended lease but completed
            if (
                any(x in self._jobs for x in job_id.singletons())
                and job_id in self._jobs_with_extended_lease
                and self._next_worker_assignments is not None
                and job_id in self._next_worker_assignments
            ):
                self._redispatched_worker_assignments[job_id] = (
                    self._next_worker_assignments[job_id]
                )

            for single in job_id.singletons():
                flags = self._bs_flags.get(single)
                if flags and (flags["big_bs"] or flags["small_bs"]):
                    self._need_to_update_allocation = True
                    flags["big_bs"] = False
                    flags["small_bs"] = False

            self._scheduler_cv.notify_all()

    # ------------------------------------------------------------------

    def is_done(self, jobs_to_complete=None):
        if jobs_to_complete is not None:
            return jobs_to_complete.issubset(self._completed_jobs)
        return self._shutdown_event.is_set() or (
            self._job_id_counter > 0 and len(self._jobs) == 0
        )

    def wait_until_done(self, poll_s: float = 2.0):
        while not self.is_done():
            time.sleep(poll_s)

    def shutdown(self, shutdown_workers: bool = True):
        self._shutdown_event.set()
        with self._scheduler_cv:
            self._scheduler_cv.notify_all()
        if shutdown_workers:
            seen = set()
            for worker_id, client in self._worker_connections.items():
                key = (client.addr, client.port)
                if key not in seen:
                    try:
                        client.shutdown()
                    except Exception:
                        pass
                    seen.add(key)
        self._server.stop(5)
        try:
            self._cancel_hang_diagnosis()
        except Exception:
            pass
