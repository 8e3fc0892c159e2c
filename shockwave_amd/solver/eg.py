"""Dynamic Eisenberg-Gale round-schedule MILP on scipy/HiGHS.

Rebuild of the reference's cvxpy+Gurobi formulation
(shockwave.py:288-711) as one sparse MILP solved by
``scipy.optimize.milp`` (HiGHS branch-and-bound):

Variables (per job i of J, round t of T, log-approximation base b of B):

* ``s[i,t]``  binary — job i scheduled in future round t
* ``p[i]``    >= 0   — planned epoch progress over the T-round window
* ``c[i,b]``  in [0,1] — piecewise-log interpolation cursor weights
* ``y[i,b]``  binary — active-base indicators (at most 2, adjacent)
* ``M``       >= 0   — max over jobs of unscheduled remaining runtime

Objective (maximized):  sum_i prio_i * sum_b c[i,b]*log(base_b) / (J*T)
                        - k * M
which is the first-order approximation of Nash social welfare over
normalized progress (shockwave.py:327-433, 565-568).

Constraints:
* per-round GPU capacity     sum_i w_i s[i,t] <= ngpus      (:297-319)
* progress <= scheduled time p_i d_i <= RD * sum_t s[i,t]   (:373-377)
* cursor interpolation       sum_b c[i,b] base_b = (e_i + p_i)/E_i,
                             sum_b c[i,b] = 1, c <= y, sum_b y <= 2,
                             y_l + y_r <= 1 for non-adjacent l,r (:384-419)
* regularizer                M >= D_i - p_i d_i             (:555-567)
* finish-time fairness       T_next + (D_i - p_i d_i)/share_i
                             <= rho_max * ftf_bound_i       (:573-597)
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional

import numpy as np
from scipy import sparse
from scipy.optimize import Bounds, LinearConstraint, milp


@dataclass
class PlannerJob:
    """The planner-facing view of one job."""

    job_id: object
    nworkers: int
    epochs: int
    epoch_progress: int
    epoch_duration_interp: float   # seconds per epoch (interpolated)
    remaining_runtime: float       # Dirichlet forecast D_i, seconds
    ftf_bound: float               # momentumed finish-time objective
    priority: float = 1.0          # utility multiplier (fallback pass)


@dataclass
class EGSolution:
    schedule: np.ndarray           # (J, T) 0/1
    planned_progress: np.ndarray   # (J,)
    status: int
    mip_gap: Optional[float] = None


def solve_eg_milp(
    jobs: List[PlannerJob],
    ngpus: int,
    round_index: int,
    future_nrounds: int,
    round_duration: float,
    logapx_bases: List[float],
    logapx_origin: dict,
    k: float,
    rhomax: float,
    enable_ftf: bool = True,
    rel_gap: float = 1e-3,
    timeout: float = 15.0,
) -> Optional[EGSolution]:
    J, T, B = len(jobs), future_nrounds, len(logapx_bases)
    assert J > 0
    assert logapx_bases[0] == 0.0
    base_vals = []
    for b in logapx_bases:
        assert 0.0 <= b <= 1.0
        base_vals.append(math.log(logapx_origin[0.0]) if b == 0.0 else math.log(b))
    assert all(a < b for a, b in zip(base_vals, base_vals[1:]))

    # variable layout
    S0 = 0                  # s[i,t] -> S0 + i*T + t
    P0 = S0 + J * T         # p[i]   -> P0 + i
    C0 = P0 + J             # c[i,b] -> C0 + i*B + b
    Y0 = C0 + J * B         # y[i,b] -> Y0 + i*B + b
    M0 = Y0 + J * B         # M
    NV = M0 + 1

    if enable_ftf:
        next_sched_time = round_duration * (round_index + future_nrounds)
        share = min(1.0, ngpus / J)
        for job in jobs:
            if job.ftf_bound * rhomax < next_sched_time:
                return None  # trivially infeasible; caller relaxes

    # objective: minimize -(sum prio*c*logval)/(J*T) + k*M
    c_obj = np.zeros(NV)
    for i, job in enumerate(jobs):
        for b in range(B):
            c_obj[C0 + i * B + b] = -job.priority * base_vals[b] / (J * T)
    c_obj[M0] = k

    rows, cols, vals, lb, ub = [], [], [], [], []
    r = 0

    def add(entries, lo, hi):
        nonlocal r
        for col, v in entries:
            rows.append(r)
            cols.append(col)
            vals.append(v)
        lb.append(lo)
        ub.append(hi)
        r += 1

    # per-round capacity
    for t in range(T):
        add([(S0 + i * T + t, jobs[i].nworkers) for i in range(J)], -np.inf, ngpus)

    for i, job in enumerate(jobs):
        d = job.epoch_duration_interp
        E = job.epochs
        e = job.epoch_progress
        # progress bound: p_i*d - RD*sum_t s[i,t] <= 0
        add(
            [(P0 + i, d)] + [(S0 + i * T + t, -round_duration) for t in range(T)],
            -np.inf,
            0.0,
        )
        # cursor base interpolation == (e + p)/E
        add(
            [(C0 + i * B + b, logapx_bases[b]) for b in range(B)]
            + [(P0 + i, -1.0 / E)],
            e / E,
            e / E,
        )
        # sum_b c = 1
        add([(C0 + i * B + b, 1.0) for b in range(B)], 1.0, 1.0)
        # c <= y
        for b in range(B):
            add([(C0 + i * B + b, 1.0), (Y0 + i * B + b, -1.0)], -np.inf, 0.0)
        # at most 2 active bases, adjacent
        add([(Y0 + i * B + b, 1.0) for b in range(B)], -np.inf, 2.0)
        for l in range(B - 2):
            for rr in range(l + 2, B):
                add([(Y0 + i * B + l, 1.0), (Y0 + i * B + rr, 1.0)], -np.inf, 1.0)
        # M >= D_i - p_i*d  <=>  -M - p_i*d <= -D_i
        add([(M0, -1.0), (P0 + i, -d)], -np.inf, -job.remaining_runtime)
        # finish-time fairness
        if enable_ftf:
            # (D_i - p_i*d)/share <= rhomax*bound - next_sched
            rhs = rhomax * job.ftf_bound - next_sched_time - job.remaining_runtime / share
            add([(P0 + i, -d / share)], -np.inf, rhs)

    A = sparse.csc_matrix((vals, (rows, cols)), shape=(r, NV))
    constraints = LinearConstraint(A, np.array(lb), np.array(ub))

    integrality = np.zeros(NV)
    integrality[S0 : S0 + J * T] = 1
    integrality[Y0 : Y0 + J * B] = 1

    var_lb = np.zeros(NV)
    var_ub = np.full(NV, np.inf)
    var_ub[S0 : S0 + J * T] = 1
    var_ub[C0 : C0 + J * B] = 1
    var_ub[Y0 : Y0 + J * B] = 1

    res = milp(
        c=c_obj,
        constraints=constraints,
        integrality=integrality,
        bounds=Bounds(var_lb, var_ub),
        options={"mip_rel_gap": rel_gap, "time_limit": timeout, "disp": False},
    )
    if res.x is None:
        return None
    x = res.x
    schedule = np.round(
        x[S0 : S0 + J * T].reshape(J, T)
    ).astype(int)
    return EGSolution(
        schedule=schedule,
        planned_progress=x[P0 : P0 + J],
        status=res.status,
        mip_gap=getattr(res, "mip_gap", None),
    )


def solve_rank_milp(
    schedule: np.ndarray,
    priorities: List[float],
    nworkers: List[int],
    ngpus: int,
    rel_gap: float = 1e-3,
    timeout: float = 15.0,
) -> np.ndarray:
    """Reorder each job's scheduled rounds to put high-priority jobs early
    (reference rank_in_schedule_jobs, shockwave.py:714-793): minimize
    sum_i prio_i * mean(t : z[i,t]=1) subject to per-job round counts and
    per-round capacity."""
    J, T = schedule.shape
    counts = schedule.sum(axis=1)
    NV = J * T
    c_obj = np.zeros(NV)
    for i in range(J):
        if counts[i] > 0:
            for t in range(T):
                c_obj[i * T + t] = priorities[i] * t / counts[i]

    rows, cols, vals, lb, ub = [], [], [], [], []
    r = 0
    for i in range(J):
        for t in range(T):
            rows.append(r)
            cols.append(i * T + t)
            vals.append(1.0)
        lb.append(counts[i])
        ub.append(counts[i])
        r += 1
    for t in range(T):
        for i in range(J):
            rows.append(r)
            cols.append(i * T + t)
            vals.append(nworkers[i])
        lb.append(-np.inf)
        ub.append(ngpus)
        r += 1
    A = sparse.csc_matrix((vals, (rows, cols)), shape=(r, NV))
    res = milp(
        c=c_obj,
        constraints=LinearConstraint(A, np.array(lb), np.array(ub)),
        integrality=np.ones(NV),
        bounds=Bounds(np.zeros(NV), np.ones(NV)),
        options={"mip_rel_gap": rel_gap, "time_limit": timeout, "disp": False},
    )
    if res.x is None:
        return schedule
    return np.round(res.x.reshape(J, T)).astype(int)
