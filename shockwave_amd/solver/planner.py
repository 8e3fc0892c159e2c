"""ShockwavePlanner — the predictive-market round planner.

Rebuild of the reference's ShockwaveScheduler (shockwave.py:20-210) plus
its module-level pipeline (round_schedule -> dynamic EG MILP ->
relax-and-rerank fallback -> work-conserving schedule construction).

The planner holds per-job JobMetadata, estimates each job's uniform-share
finish time (the "market price" series), and every re-solve plans the next
``future_nrounds`` rounds by maximizing approximate Nash social welfare
subject to finish-time-fairness bounds; on infeasibility, FTF constraints
become priority boosts (rho^lambda) and a second MILP re-ranks rounds.
"""

from __future__ import annotations

import logging
from collections import OrderedDict
from typing import Dict, List

import numpy as np

from ..core.metadata import JobMetadata
from .eg import PlannerJob, solve_eg_milp, solve_rank_milp

logger = logging.getLogger("shockwave_amd.planner")


def finish_time_momentumed_average(series, round_index, momentum=0.9):
    """Windowed running average of the finish-time estimates, re-weighted by
    how long each estimate was current (shockwave.py:480-501)."""
    assert len(series) > 0
    irounds = [ir for ir, _ in series]
    assert max(irounds) <= round_index
    windows = np.diff(irounds + [round_index])
    if windows.size == 0 or windows.max() == 0:
        probs = [1.0] * len(series)
        probs = [p / len(probs) for p in probs]
    else:
        probs = (windows / windows.sum()).tolist()
    vals = [v for _, v in series]
    running = sum(p * v for p, v in zip(probs, vals))
    return momentum * running + (1.0 - momentum) * vals[-1]


class ShockwavePlanner:
    def __init__(
        self,
        ngpus: int,
        gram: int,
        init_metadata: OrderedDict,
        future_nrounds: int,
        round_duration: float,
        solver_rel_gap: float = 1e-3,
        solver_num_threads: int = 0,
        solver_timeout: float = 15.0,
        n_epoch_vars_max: int = 10,
        logapx_bases: List[float] = (0.0, 0.2, 0.4, 0.6, 0.8, 1.0),
        logapx_origin: Dict[float, float] = None,
        k: float = 1e-3,
        lam: float = 12.0,
        rhomax: float = 1.0,
    ):
        assert ngpus > 0 and future_nrounds > 0 and round_duration > 0
        self.ngpus = ngpus
        self.gram = gram
        self.future_nrounds = future_nrounds
        self.round_duration = round_duration
        self.solver_rel_gap = solver_rel_gap
        self.solver_timeout = solver_timeout
        self.logapx_bases = list(logapx_bases)
        self.logapx_origin = logapx_origin or {0.0: 1e-1}
        self.k = k
        self.lam = lam
        self.rhomax = rhomax

        self.metadata: OrderedDict = OrderedDict()
        for job_id, md in (init_metadata or OrderedDict()).items():
            self.add_metadata(job_id, md)
        self.completed_jobs: OrderedDict = OrderedDict()
        self.schedules: OrderedDict = OrderedDict()
        self.round_ptr = 0
        self.resolve = True
        self.reestimate_share = True
        self.share_series: Dict = {}

    # -- metadata lifecycle (shockwave.py:177-210) -------------------------

    def add_metadata(self, job_id, md: JobMetadata, share_update=True):
        assert job_id not in self.metadata
        self.metadata[job_id] = md
        self.set_resolve()
        if share_update:
            self.reestimate_share = True

    def remove_metadata(self, job_id, share_update=True):
        assert job_id not in self.completed_jobs
        self.completed_jobs[job_id] = self.metadata.pop(job_id)
        if share_update:
            self.reestimate_share = True
        self.set_resolve()

    def schedule_progress(self, job_id, epoch_progress):
        md = self.metadata[job_id]
        md.set_epoch_progress(epoch_progress)
        md.reset_waiting_delay()

    def deschedule_waiting_delay(self, job_id, delay):
        if job_id in self.metadata:
            self.metadata[job_id].add_waiting_delay(delay)

    def increment_round_ptr(self):
        self.round_ptr += 1

    def set_resolve(self):
        self.resolve = True

    def clear_resolve(self):
        self.resolve = False

    # -- market share estimation (shockwave.py:88-120) ---------------------

    def finish_time_uniform_share(self):
        njobs = len(self.metadata)
        if self.reestimate_share:
            for job_id, md in self.metadata.items():
                uniform_share = min(1.0, self.ngpus / njobs)
                assert uniform_share > 0
                md.calibrate()
                estimate = (
                    (md.timestamp_submit or 0.0)
                    + (
                        sum(md.epoch_duration[: md.epoch_progress])
                        + md.remaining_runtime(md.epoch_progress)
                    )
                    / uniform_share
                )
                self.share_series.setdefault(job_id, []).append(
                    (self.round_ptr, estimate)
                )
        self.reestimate_share = False

    # -- planning ----------------------------------------------------------

    def round_schedule(self) -> List:
        if not self.resolve and self.schedules:
            if self.round_ptr in self.schedules:
                return self.schedules[self.round_ptr]

        job_ids = list(self.metadata.keys())
        if not job_ids:
            return []
        self.finish_time_uniform_share()

        schedule = self._solve(job_ids)
        self.schedules = self._construct_schedules(schedule, job_ids)
        self.clear_resolve()
        return self.schedules[self.round_ptr]

    def _planner_jobs(self, job_ids, priorities=None) -> List[PlannerJob]:
        jobs = []
        for idx, job_id in enumerate(job_ids):
            md = self.metadata[job_id]
            ftf_bound = finish_time_momentumed_average(
                self.share_series[job_id], self.round_ptr
            )
            jobs.append(
                PlannerJob(
                    job_id=job_id,
                    nworkers=md.nworkers,
                    epochs=md.epochs,
                    epoch_progress=md.epoch_progress,
                    epoch_duration_interp=md.interpolated_epoch_duration(),
                    remaining_runtime=md.remaining_runtime(),
                    ftf_bound=ftf_bound,
                    priority=1.0 if priorities is None else priorities[idx],
                )
            )
        return jobs

    def _solve(self, job_ids) -> np.ndarray:
        jobs = self._planner_jobs(job_ids)
        sol = solve_eg_milp(
            jobs,
            self.ngpus,
            self.round_ptr,
            self.future_nrounds,
            self.round_duration,
            self.logapx_bases,
            self.logapx_origin,
            self.k,
            self.rhomax,
            enable_ftf=True,
            rel_gap=self.solver_rel_gap,
            timeout=self.solver_timeout,
        )
        if sol is not None:
            return sol.schedule

        # FTF-infeasible: relax constraints into utility priorities
        # (shockwave.py:630-706, 830-909)
        logger.info(
            "round %d: FTF constraints infeasible; relax + rerank", self.round_ptr
        )
        priorities = self._relaxed_priorities(job_ids)
        sol = solve_eg_milp(
            self._planner_jobs(job_ids, priorities),
            self.ngpus,
            self.round_ptr,
            self.future_nrounds,
            self.round_duration,
            self.logapx_bases,
            self.logapx_origin,
            self.k,
            self.rhomax,
            enable_ftf=False,
            rel_gap=self.solver_rel_gap,
            timeout=self.solver_timeout,
        )
        if sol is None:
            # solver failure: fall back to scheduling everything greedily
            logger.error("round %d: EG MILP failed twice; greedy fallback",
                         self.round_ptr)
            return self._greedy_schedule(job_ids)
        ranked = solve_rank_milp(
            sol.schedule,
            priorities,
            [self.metadata[j].nworkers for j in job_ids],
            self.ngpus,
            rel_gap=self.solver_rel_gap,
            timeout=self.solver_timeout,
        )
        return ranked

    def _relaxed_priorities(self, job_ids) -> List[float]:
        """rho^lambda priority boosts for jobs projected to violate FTF
        (shockwave.py:830-909)."""
        priority_M = 1e2
        round_time = self.round_duration * self.round_ptr
        njobs = len(self.metadata)
        future_share = min(1.0, self.ngpus / njobs)
        priorities = []
        for job_id in job_ids:
            md = self.metadata[job_id]
            md.calibrate()
            remaining = md.remaining_runtime()
            projected = round_time + remaining / future_share
            bound = finish_time_momentumed_average(
                self.share_series[job_id], self.round_ptr
            )
            ratio = projected / bound if bound > 0 else 1.0
            if ratio > self.rhomax:
                power = priority_M if remaining < self.round_duration else self.lam
                # ratio**power in log space: Python floats raise
                # OverflowError past 1e308 (large rho with power=100)
                import math

                log_p = power * math.log(max(ratio, 1e-9))
                priorities.append(math.exp(min(log_p, 700.0)))
            else:
                priorities.append(1.0)
        # clamp giant priorities to avoid numeric blowup in the MILP.
        # Clamp PER JOB (min), never rescale the whole vector: dividing
        # everything by one near-done job's astronomic ratio^100 crushed
        # moderate violators' boosts below the k*M term and starved them
        # for dozens of rounds (the r1 worst-rho tail: VERDICT item 3)
        arr = np.minimum(np.array(priorities), 1e6)
        return arr.tolist()

    def _greedy_schedule(self, job_ids) -> np.ndarray:
        """Capacity-respecting round-robin used only if HiGHS fails."""
        J, T = len(job_ids), self.future_nrounds
        out = np.zeros((J, T), dtype=int)
        order = sorted(
            range(J),
            key=lambda i: -self.metadata[job_ids[i]].remaining_runtime(),
        )
        for t in range(T):
            free = self.ngpus
            for i in order:
                w = self.metadata[job_ids[i]].nworkers
                if w <= free:
                    out[i, t] = 1
                    free -= w
                if free <= 0:
                    break
            order = order[1:] + order[:1]  # rotate for fairness
        return out

    def _construct_schedules(self, schedule: np.ndarray, job_ids) -> OrderedDict:
        """Solution matrix -> per-round job lists with work-conserving fill
        (shockwave.py:213-285): idle GPUs go to unscheduled jobs with the
        largest remaining runtime that still fit."""
        J, T = schedule.shape
        out = OrderedDict()
        for t in range(T):
            round_index = self.round_ptr + t
            ids = [job_ids[i] for i in range(J) if schedule[i, t] == 1]
            if not ids:
                logger.warning("no jobs scheduled in round %d", round_index)
            used = sum(self.metadata[j].nworkers for j in ids)
            idle = self.ngpus - used
            if idle > 0:
                rest = [j for j in job_ids if j not in ids]
                rest.sort(
                    key=lambda j: self.metadata[j].remaining_runtime(),
                    reverse=True,
                )
                for j in rest:
                    w = self.metadata[j].nworkers
                    if w <= idle:
                        idle -= w
                        ids.append(j)
                    if idle <= 0:
                        break
            out[round_index] = ids
        return out
