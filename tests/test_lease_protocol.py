"""Lease-protocol edge cases on the physical scheduler's callbacks
(SURVEY §7 'hard parts': early-init extra time, multi-GPU first-requester
lease computation, extended-lease renewal)."""

import time
from collections import OrderedDict

import pytest

from shockwave_amd.core.job import Job, JobIdPair
from shockwave_amd.core import trace as trace_mod
from shockwave_amd.engine.physical import PhysicalScheduler
from shockwave_amd.engine.scheduler import EARLY_INIT_THRESHOLD, INFINITY
from shockwave_amd.policies import get_policy
from tests.test_rpc_runtime import free_port


@pytest.fixture
def sched(throughputs):
    s = PhysicalScheduler(
        get_policy("max_min_fairness"),
        port=free_port(),
        expected_num_workers=99,  # round loop stays parked
        throughputs=throughputs,
        time_per_iteration=100,
        profiles=[],
        worker_type="mi355x",
    )
    yield s
    s.shutdown(shutdown_workers=False)


def add_job(s, throughputs, sf=1, steps=10000):
    job = Job(
        job_id=None,
        job_type="ResNet-18 (batch size 32)",
        command="python3 main.py --batch_size 32",
        working_directory="image_classification/cifar10",
        num_steps_arg="--num_steps",
        total_steps=steps,
        duration=3600,
        scale_factor=sf,
        mode="static",
    )
    return s.add_job(job), job


class TestInitLease:
    def test_unknown_job(self, sched):
        assert sched._init_job_callback(JobIdPair(42)) == (0, 0, 0, 0, 0)

    def test_mid_round_lease_is_remaining_time(self, sched, throughputs):
        jid, _ = add_job(sched, throughputs)
        sched._current_worker_assignments = OrderedDict({jid: (0,)})
        sched._current_round_start_time = sched.get_current_timestamp() - 40
        steps, duration, extra, rts, deadline = sched._init_job_callback(jid)
        assert steps == 10000
        assert rts == 0 and deadline == int(3600 * 1.5)
        assert 55 <= duration <= 60  # 100s round, 40s elapsed
        assert extra == 0

    def test_early_dispatch_gets_extra_time(self, sched, throughputs):
        """Job dispatched for the NEXT round while the current one still
        runs: full-round lease + the remaining current-round time as extra
        (reference :3966-4048)."""
        jid, _ = add_job(sched, throughputs)
        sched._current_round_start_time = sched.get_current_timestamp() - 70
        sched._next_worker_assignments = OrderedDict({jid: (0,)})
        sched._current_worker_assignments = OrderedDict()
        steps, duration, extra, _, _ = sched._init_job_callback(jid)
        assert duration == 100  # a full round
        assert 25 <= extra <= 30

    def test_between_rounds_threshold(self, sched, throughputs):
        jid, _ = add_job(sched, throughputs)
        sched._current_round_start_time = sched.get_current_timestamp() - 101
        sched._next_worker_assignments = None
        sched._current_worker_assignments = OrderedDict()
        steps, duration, extra, _, _ = sched._init_job_callback(jid)
        assert duration == 100 - EARLY_INIT_THRESHOLD

    def test_multi_gpu_steps_divided(self, sched, throughputs):
        jid, _ = add_job(sched, throughputs, sf=4, steps=1000)
        sched._current_worker_assignments = OrderedDict({jid: (0, 1, 2, 3)})
        sched._current_round_start_time = sched.get_current_timestamp()
        steps = sched._init_job_callback(jid)[0]
        assert steps == 250


class TestUpdateLease:
    def _prep(self, sched, throughputs, sf=1):
        jid, _ = add_job(sched, throughputs, sf=sf)
        sched._current_worker_assignments = OrderedDict(
            {jid: tuple(range(sf))}
        )
        sched._current_round_start_time = sched.get_current_timestamp() - 50
        return jid

    def test_zero_progress_gets_remaining_round(self, sched, throughputs):
        jid = self._prep(sched, throughputs)
        steps, duration, rts, deadline = sched._update_lease_callback(
            jid, 0, steps=0, duration=0, max_steps=100, max_duration=60
        )
        assert steps == 10000
        assert 45 <= duration <= 50
        assert deadline == int(3600 * 1.5)

    def test_single_gpu_extends_by_remaining(self, sched, throughputs):
        jid = self._prep(sched, throughputs)
        steps, duration, _, _ = sched._update_lease_callback(
            jid, 0, steps=50, duration=30.0, max_steps=100, max_duration=60
        )
        assert steps == 100
        assert 75 <= duration <= 80  # 30 used + ~50 remaining

    def test_extended_lease_adds_full_round(self, sched, throughputs):
        jid = self._prep(sched, throughputs)
        sched._jobs_with_extended_lease.add(jid)
        steps, duration, _, _ = sched._update_lease_callback(
            jid, 0, steps=50, duration=30.0, max_steps=100, max_duration=60
        )
        assert steps == 100
        assert 175 <= duration <= 180  # + the next full round

    def test_multi_gpu_first_requester_computes_shared_lease(
        self, sched, throughputs
    ):
        jid = self._prep(sched, throughputs, sf=2)
        s1 = sched._update_lease_callback(
            jid, 0, steps=40, duration=40.0, max_steps=100, max_duration=60
        )
        # first requester: throughput 1 step/s, ~50s left -> 40 + ~50 steps
        assert s1[1] == INFINITY
        assert 85 <= s1[0] <= 95
        # second worker receives the SAME max_steps
        s2 = sched._update_lease_callback(
            jid, 1, steps=38, duration=40.0, max_steps=100, max_duration=60
        )
        assert s2[0] == s1[0]
        assert s2[1] == INFINITY

    def test_lease_request_records_progress_for_planner(
        self, sched, throughputs
    ):
        jid = self._prep(sched, throughputs, sf=2)
        sched._update_lease_callback(
            jid, 0, steps=40, duration=40.0, max_steps=100, max_duration=60
        )
        assert sched._steps_run_in_current_lease[jid] == 80  # aggregated


class TestExtendedLeaseWatchdog:
    def test_responsive_job_completed(self, sched, throughputs):
        """An extended-lease job that requested renewals this round is
        marked complete by the health check (reference :4283-4339)."""
        jid, _ = add_job(sched, throughputs)
        sched._current_worker_assignments = OrderedDict({jid: (0,)})
        sched._jobs_with_extended_lease.add(jid)
        sched._lease_update_requests[jid] = [(10, 5.0, 100, 60.0)]
        sched._completion_events[jid] = object()
        sched._done_callback_extended_lease(jid)
        assert jid in sched._completed_jobs_in_current_round
        assert jid not in sched._completion_events

    def test_unresponsive_job_killed_after_grace(self, sched, throughputs):
        """No lease renewals -> one grace round (the round still completes
        for the job so _end_round can proceed), then killed on the second
        consecutive silent round.  Killing on the FIRST miss livelocks
        short-round configs: a renewal landing just past a round boundary
        triggered kill -> redispatch -> slow start -> miss again."""
        jid, _ = add_job(sched, throughputs)
        sched._current_worker_assignments = OrderedDict({jid: (0,)})
        sched._jobs_with_extended_lease.add(jid)
        sched._lease_update_requests[jid] = []
        sched._completion_events[jid] = object()

        killed = []
        sched._kill_job = lambda j: killed.append(j)
        sched._done_callback_extended_lease(jid)
        assert killed == []  # grace round
        assert jid in sched._completed_jobs_in_current_round

        # second consecutive silent round -> kill
        sched._completed_jobs_in_current_round = set()
        sched._lease_update_requests[jid] = []
        sched._done_callback_extended_lease(jid)
        assert killed == [jid]
        assert jid not in sched._completed_jobs_in_current_round

    def test_renewal_resets_grace_counter(self, sched, throughputs):
        """A renewal between two silent rounds resets the counter: only
        CONSECUTIVE misses kill."""
        jid, _ = add_job(sched, throughputs)
        sched._current_worker_assignments = OrderedDict({jid: (0,)})
        sched._jobs_with_extended_lease.add(jid)
        killed = []
        sched._kill_job = lambda j: killed.append(j)

        sched._lease_update_requests[jid] = []
        sched._completion_events[jid] = object()
        sched._done_callback_extended_lease(jid)  # miss 1
        # responsive round resets
        sched._lease_update_requests[jid] = [(10, 5.0, 100, 60.0)]
        sched._completion_events[jid] = object()
        sched._done_callback_extended_lease(jid)
        # next miss is miss 1 again, not 2
        sched._completed_jobs_in_current_round = set()
        sched._lease_update_requests[jid] = []
        sched._completion_events[jid] = object()
        sched._done_callback_extended_lease(jid)
        assert killed == []
