"""Finish-time fairness (Themis) policy.

Reference: policies/finish_time_fairness.py:57-279.  Minimize the maximum
expected finish-time ratio

    rho_i = (t_elapsed_i + R_i / (tput_i . x_i)) / T_isolated_i

The reference solves this directly with cvxpy's ``inv_pos`` (convex); here
we bisect on rho: for fixed rho the constraint becomes linear,

    tput_i . x_i >= R_i / (rho * T_iso_i - t_elapsed_i),

so each probe is an LP feasibility check (HiGHS).  ~40 probes land within
1e-3 relative, cheaper than the reference's conic solve at this size.

Stateful bookkeeping matches the reference: cumulative isolated time per
job accrues (steps run last round) / (isolated throughput last round).
"""

from __future__ import annotations

import copy

import numpy as np

from .base import Policy, PolicyWithPacking
from .simple import IsolatedPolicy


class FinishTimeFairnessPolicyWithPerf(Policy):
    name = "FinishTimeFairness_Perf"

    def __init__(self):
        self._isolated = IsolatedPolicy()
        self._cumulative_isolated_time = {}
        self._isolated_throughputs_prev = {}
        self._num_steps_remaining_prev = {}

    def _feasible(self, rho, m, n, weights, rhs_steps, t_elapsed, t_iso, sfa):
        A_ub, b_ub = self.base_constraints(m, n, sfa)
        rows, rhs = [], []
        for i in range(m):
            budget = rho * t_iso[i] - t_elapsed[i]
            if budget <= 0:
                if rhs_steps[i] > 0:
                    return None  # cannot meet rho for job i at all
                continue
            row = np.zeros(m * n)
            row[i * n : (i + 1) * n] = -weights[i]
            rows.append(row)
            rhs.append(-rhs_steps[i] / budget)
        if rows:
            A = np.vstack([A_ub, np.array(rows)])
            b = np.concatenate([b_ub, np.array(rhs)])
        else:
            A, b = A_ub, b_ub
        res = self.solve_lp(np.zeros(m * n), A, b)
        return res.x[: m * n].reshape((m, n)) if res.success else None

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        times_since_start,
        num_steps_remaining,
        cluster_spec,
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            self._isolated_throughputs_prev = {}
            self._num_steps_remaining_prev = {}
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        isolated_tputs = self._isolated.get_throughputs(
            throughputs, index, scale_factors, cluster_spec
        ).reshape(-1)

        t_elapsed = np.zeros(m)
        t_iso = np.zeros(m)
        rhs_steps = np.zeros(m)
        for i, jid in enumerate(job_ids):
            self._cumulative_isolated_time.setdefault(jid, 0.0)
            if jid in self._num_steps_remaining_prev:
                steps_run = (
                    self._num_steps_remaining_prev[jid] - num_steps_remaining[jid]
                )
                self._cumulative_isolated_time[jid] += (
                    steps_run / self._isolated_throughputs_prev[jid]
                )
            t_elapsed[i] = times_since_start[jid]
            rhs_steps[i] = max(0.0, num_steps_remaining[jid])
            t_iso[i] = self._cumulative_isolated_time[jid] + (
                num_steps_remaining[jid] / max(isolated_tputs[i], 1e-10)
            )
            t_iso[i] = max(t_iso[i], 1e-6)

        # bisection on rho
        lo, hi = 0.0, 2.0
        x_best = None
        for _ in range(60):
            x = self._feasible(hi, m, n, throughputs, rhs_steps, t_elapsed, t_iso, sfa)
            if x is not None:
                x_best = x
                break
            lo, hi = hi, hi * 2.0
            if hi > 1e9:
                break
        if x_best is None:
            # fall back to isolated shares
            x = self._isolated._allocation(m, n, sfa, worker_types, cluster_spec)
            return self.unflatten(x, index)
        for _ in range(40):
            mid = 0.5 * (lo + hi)
            x = self._feasible(mid, m, n, throughputs, rhs_steps, t_elapsed, t_iso, sfa)
            if x is not None:
                x_best, hi = x, mid
            else:
                lo = mid
            if hi - lo <= 1e-3 * max(1.0, hi):
                break

        self._num_steps_remaining_prev = copy.copy(num_steps_remaining)
        self._isolated_throughputs_prev = {
            jid: isolated_tputs[i] for i, jid in enumerate(job_ids)
        }
        return self.unflatten(self.clip_allocation(x_best), index)


class FinishTimeFairnessPolicy(Policy):
    """Wrapper that collapses throughputs to the canonical worker type
    (finish_time_fairness.py:36-46 hardcodes v100; we use whichever single
    type the cluster runs)."""

    name = "FinishTimeFairness"

    def __init__(self):
        self._perf = FinishTimeFairnessPolicyWithPerf()

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        times_since_start,
        num_steps_remaining,
        cluster_spec,
    ):
        canonical = None
        for wt in ("mi355x", "v100"):
            sample = next(iter(unflattened_throughputs.values()), {})
            if wt in sample:
                canonical = wt
                break
        new_tputs = {}
        for jid, per_wt in unflattened_throughputs.items():
            ref = per_wt[canonical] if canonical else max(per_wt.values())
            new_tputs[jid] = {wt: ref for wt in per_wt}
        return self._perf.get_allocation(
            new_tputs,
            scale_factors,
            unflattened_priority_weights,
            times_since_start,
            num_steps_remaining,
            cluster_spec,
        )


class FinishTimeFairnessPolicyWithPacking(PolicyWithPacking):
    """Packed Themis: rho bisection where each SINGLE job's rate sums its
    effective throughput over every combination row involving it
    (reference finish_time_fairness.py:131-279).  Stateful bookkeeping
    (cumulative isolated time) is keyed by single job, as in the perf
    variant."""

    name = "FinishTimeFairness_Packing"

    def __init__(self):
        self._isolated = IsolatedPolicy()
        self._cumulative_isolated_time = {}
        self._isolated_throughputs_prev = {}
        self._num_steps_remaining_prev = {}

    def _feasible(self, rho, m, n, w, rhs_steps, t_elapsed, t_iso,
                  base_A, base_b):
        rows, rhs = [], []
        for k in range(len(rhs_steps)):
            budget = rho * t_iso[k] - t_elapsed[k]
            if budget <= 0:
                if rhs_steps[k] > 0:
                    return None
                continue
            rows.append(-w[k])
            rhs.append(-rhs_steps[k] / budget)
        if rows:
            A = np.vstack([base_A, np.array(rows)])
            b = np.concatenate([base_b, np.array(rhs)])
        else:
            A, b = base_A, base_b
        res = self.solve_lp(np.zeros(m * n), A, b)
        return res.x[: m * n].reshape((m, n)) if res.success else None

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        times_since_start,
        num_steps_remaining,
        cluster_spec,
    ):
        all_tputs, index, singles = self.flatten_packed(
            unflattened_throughputs, cluster_spec
        )
        if all_tputs is None:
            self._isolated_throughputs_prev = {}
            self._num_steps_remaining_prev = {}
            return None
        job_ids, worker_types = index
        m, n = all_tputs[0].shape
        K = len(singles)
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        w = np.array([t.reshape(-1) for t in all_tputs])

        iso = self.isolated_single_throughputs(all_tputs, singles)
        sf_singles = {s: scale_factors[s] for s in singles}
        isolated_tputs = self._isolated.get_throughputs(
            iso, (singles, worker_types), sf_singles, cluster_spec
        ).reshape(-1)

        t_elapsed = np.zeros(K)
        t_iso = np.zeros(K)
        rhs_steps = np.zeros(K)
        for k, s in enumerate(singles):
            self._cumulative_isolated_time.setdefault(s, 0.0)
            if s in self._num_steps_remaining_prev:
                steps_run = (
                    self._num_steps_remaining_prev[s] - num_steps_remaining[s]
                )
                self._cumulative_isolated_time[s] += (
                    steps_run / self._isolated_throughputs_prev[s]
                )
            t_elapsed[k] = times_since_start[s]
            rhs_steps[k] = max(0.0, num_steps_remaining[s])
            t_iso[k] = self._cumulative_isolated_time[s] + (
                num_steps_remaining[s] / max(isolated_tputs[k], 1e-10)
            )
            t_iso[k] = max(t_iso[k], 1e-6)

        base_A, base_b = self.packed_constraints(
            m, n, sfa, job_ids, singles
        )
        lo, hi = 0.0, 2.0
        x_best = None
        for _ in range(60):
            x = self._feasible(hi, m, n, w, rhs_steps, t_elapsed, t_iso,
                               base_A, base_b)
            if x is not None:
                x_best = x
                break
            lo, hi = hi, hi * 2.0
            if hi > 1e9:
                break
        if x_best is None:
            return None
        for _ in range(40):
            mid = 0.5 * (lo + hi)
            x = self._feasible(mid, m, n, w, rhs_steps, t_elapsed, t_iso,
                               base_A, base_b)
            if x is not None:
                x_best, hi = x, mid
            else:
                lo = mid
            if hi - lo <= 1e-3 * max(1.0, hi):
                break

        self._num_steps_remaining_prev = copy.copy(num_steps_remaining)
        self._isolated_throughputs_prev = {
            s: isolated_tputs[k] for k, s in enumerate(singles)
        }
        return self.unflatten(self.clip_allocation(x_best), index)
