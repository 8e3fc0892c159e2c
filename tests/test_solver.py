"""Shockwave planner / EG MILP tests."""

import numpy as np
import pytest
from collections import OrderedDict

from shockwave_amd.core import trace
from shockwave_amd.core.metadata import JobMetadata
from shockwave_amd.solver import (
    MinMaxSumKSubarrays,
    PlannerJob,
    ShockwavePlanner,
    finish_time_momentumed_average,
    solve_eg_milp,
    solve_rank_milp,
)

BASES = [0.0, 0.2, 0.4, 0.6, 0.8, 1.0]
ORIGIN = {0.0: 1e-1}


def mk_planner_job(i, nworkers=1, epochs=10, progress=0, dur=100.0,
                   remaining=1000.0, bound=1e9, prio=1.0):
    return PlannerJob(
        job_id=i, nworkers=nworkers, epochs=epochs, epoch_progress=progress,
        epoch_duration_interp=dur, remaining_runtime=remaining,
        ftf_bound=bound, priority=prio,
    )


class TestSubarrays:
    def test_split(self):
        best, parts = MinMaxSumKSubarrays([1, 2, 3, 4, 5], 2).solve()
        assert best == pytest.approx(9.0, abs=1e-6)
        assert sum(len(p) for p in parts) == 5

    def test_k_equals_n(self):
        best, parts = MinMaxSumKSubarrays([5, 1, 3], 3).solve()
        assert best == pytest.approx(5.0, abs=1e-6)


class TestMomentumAverage:
    def test_single(self):
        assert finish_time_momentumed_average([(0, 100.0)], 5) == pytest.approx(100.0)

    def test_weighted(self):
        series = [(0, 100.0), (5, 200.0)]
        v = finish_time_momentumed_average(series, 10)
        # running avg = 0.5*100 + 0.5*200 = 150; momentum blend with last:
        assert v == pytest.approx(0.9 * 150 + 0.1 * 200)


class TestEGMilp:
    def test_capacity_respected(self):
        jobs = [mk_planner_job(i) for i in range(4)]
        sol = solve_eg_milp(jobs, ngpus=2, round_index=0, future_nrounds=5,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            timeout=30)
        assert sol is not None
        per_round = sol.schedule.sum(axis=0)
        assert (per_round <= 2).all()

    def test_all_fit_scheduled_every_round(self):
        jobs = [mk_planner_job(i) for i in range(2)]
        sol = solve_eg_milp(jobs, ngpus=4, round_index=0, future_nrounds=4,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            timeout=30)
        # plenty of GPUs: welfare maximized by scheduling everyone always
        assert sol.schedule.sum() == 8

    def test_ftf_infeasible_returns_none(self):
        jobs = [mk_planner_job(0, bound=1.0)]  # bound already passed
        sol = solve_eg_milp(jobs, ngpus=1, round_index=10, future_nrounds=5,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            timeout=30)
        assert sol is None

    def test_scale_factor_capacity(self):
        jobs = [mk_planner_job(0, nworkers=4), mk_planner_job(1, nworkers=4),
                mk_planner_job(2, nworkers=1)]
        sol = solve_eg_milp(jobs, ngpus=4, round_index=0, future_nrounds=4,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            timeout=30)
        used = (sol.schedule.T * np.array([4, 4, 1])).sum(axis=1)
        assert (used <= 4).all()

    def test_starved_job_gets_time(self):
        # job 1 tiny progress needs time to lift log utility from origin
        jobs = [
            mk_planner_job(0, epochs=10, progress=9, dur=50.0, remaining=50.0),
            mk_planner_job(1, epochs=10, progress=0, dur=50.0, remaining=500.0),
        ]
        sol = solve_eg_milp(jobs, ngpus=1, round_index=0, future_nrounds=6,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1e9,
                            timeout=30)
        assert sol.schedule[1].sum() >= sol.schedule[0].sum()


class TestRankMilp:
    def test_priority_goes_first(self):
        schedule = np.array([[1, 1, 0, 0], [0, 0, 1, 1]])
        ranked = solve_rank_milp(schedule, priorities=[1.0, 100.0],
                                 nworkers=[1, 1], ngpus=1)
        # high-priority job 1 should move to the earliest rounds
        assert ranked[1, :2].sum() == 2
        assert ranked.sum(axis=1).tolist() == [2, 2]
        assert (ranked.sum(axis=0) <= 1).all()


class TestPlanner:
    def _make_planner(self, throughputs, ngpus=2, njobs=4, sf=1):
        from tests.test_core import make_job

        md = OrderedDict()
        for i in range(njobs):
            j = make_job(mode="static", steps=1563 * 10, sf=sf)
            prof = trace.build_job_profile(j, throughputs)
            m = JobMetadata(i, prof)
            m.register_submit(0.0)
            md[i] = m
        return ShockwavePlanner(
            ngpus=ngpus, gram=288, init_metadata=md, future_nrounds=5,
            round_duration=120,
        )

    def test_round_schedule_capacity(self, throughputs):
        p = self._make_planner(throughputs)
        sched = p.round_schedule()
        assert len(sched) <= 2
        assert len(sched) >= 1

    def test_schedule_cached_until_resolve(self, throughputs):
        p = self._make_planner(throughputs)
        s0 = p.round_schedule()
        s1 = p.round_schedule()
        assert s0 == s1
        p.increment_round_ptr()
        s2 = p.round_schedule()  # still cached plan for round 1
        assert p.round_ptr == 1

    def test_work_conserving_fill(self, throughputs):
        # 4 GPUs, 2 single-GPU jobs: both always scheduled (fill)
        p = self._make_planner(throughputs, ngpus=4, njobs=2)
        sched = p.round_schedule()
        assert sorted(sched) == [0, 1]

    def test_remove_metadata_triggers_resolve(self, throughputs):
        p = self._make_planner(throughputs)
        p.round_schedule()
        p.remove_metadata(0)
        assert p.resolve
        sched = p.round_schedule()
        assert 0 not in sched


class TestRelaxedPriorities:
    def test_huge_rho_no_overflow(self, throughputs):
        """ratio**lambda with rho >> 1 must not overflow (hit on GPU when a
        tiny job's ftf bound was long passed)."""
        from tests.test_core import make_job
        from shockwave_amd.core import trace

        md = OrderedDict()
        j = make_job(mode="static", steps=1563 * 5)
        prof = trace.build_job_profile(j, throughputs)
        m = JobMetadata(0, prof)
        m.register_submit(0.0)
        md[0] = m
        p = ShockwavePlanner(
            ngpus=1, gram=288, init_metadata=md, future_nrounds=3,
            round_duration=30,
        )
        # fabricate a stale tiny bound so projected/bound explodes
        p.finish_time_uniform_share()
        p.share_series[0] = [(0, 1e-6)]
        p.round_ptr = 1000
        pri = p._relaxed_priorities([0])
        assert all(np.isfinite(pri))
        sched = p._greedy_schedule([0])
        assert sched.shape == (1, 3)


class TestEGMilpProperties:
    """Structural fairness/efficiency properties of the Nash-welfare MILP
    (reference shockwave.py dynamic_eisenberg_gale_scheduling :504-712)."""

    def test_symmetric_jobs_split_rounds_evenly(self):
        jobs = [mk_planner_job(i, remaining=2000.0) for i in range(3)]
        sol = solve_eg_milp(jobs, ngpus=1, round_index=0, future_nrounds=6,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            enable_ftf=False, timeout=30)
        counts = sol.schedule.sum(axis=1)
        assert counts.sum() <= 6
        # identical jobs must not differ by more than one round
        assert counts.max() - counts.min() <= 1

    def test_higher_priority_gets_no_fewer_rounds(self):
        jobs = [mk_planner_job(0, prio=5.0, remaining=2000.0),
                mk_planner_job(1, prio=1.0, remaining=2000.0)]
        sol = solve_eg_milp(jobs, ngpus=1, round_index=0, future_nrounds=4,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            enable_ftf=False, timeout=30)
        counts = sol.schedule.sum(axis=1)
        assert counts[0] >= counts[1]

    def test_nearly_done_job_not_overscheduled(self):
        """A job that finishes inside one round cannot usefully consume
        more rounds than it needs; the slack goes to the long job."""
        jobs = [mk_planner_job(0, epochs=10, progress=9, dur=10.0,
                               remaining=10.0),
                mk_planner_job(1, remaining=5000.0)]
        sol = solve_eg_milp(jobs, ngpus=1, round_index=0, future_nrounds=6,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            enable_ftf=False, timeout=30)
        counts = sol.schedule.sum(axis=1)
        assert counts[1] >= 4  # long job gets the bulk of the horizon

    def test_planned_progress_bounded(self):
        """planned_progress is epochs gained over the horizon: at most
        scheduled-time/epoch-duration and at most the remaining epochs."""
        jobs = [mk_planner_job(i, epochs=10, progress=0, dur=100.0,
                               remaining=900.0) for i in range(2)]
        sol = solve_eg_milp(jobs, ngpus=2, round_index=0, future_nrounds=5,
                            round_duration=120, logapx_bases=BASES,
                            logapx_origin=ORIGIN, k=1e-3, rhomax=1.0,
                            enable_ftf=False, timeout=30)
        counts = sol.schedule.sum(axis=1)
        for i in range(2):
            assert sol.planned_progress[i] >= -1e-6
            assert sol.planned_progress[i] <= min(
                counts[i] * 120.0 / 100.0, 10.0
            ) + 1e-6
