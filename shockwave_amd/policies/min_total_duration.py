"""Min-total-duration (OSSP) policy.

Reference: policies/min_total_duration.py:50-135.  Binary-search the
smallest horizon T such that every job can finish its remaining steps:
feasibility LP with constraint ``tput_i . x_i >= steps_i / T`` under the
base constraints.
"""

from __future__ import annotations

import numpy as np

from .base import Policy, PolicyWithPacking


class MinTotalDurationPolicyWithPerf(Policy):
    name = "MinTotalDuration_Perf"

    def _feasible(self, T, m, n, throughputs, steps, sfa):
        A_ub, b_ub = self.base_constraints(m, n, sfa)
        rows = np.zeros((m, m * n))
        for i in range(m):
            rows[i, i * n : (i + 1) * n] = -throughputs[i]
        A = np.vstack([A_ub, rows])
        b = np.concatenate([b_ub, -steps / T])
        res = self.solve_lp(np.zeros(m * n), A, b)
        return res.x[: m * n].reshape((m, n)) if res.success else None

    def get_allocation(
        self, unflattened_throughputs, scale_factors, num_steps_remaining, cluster_spec
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        steps = np.array(
            [max(0.0, num_steps_remaining[jid]) for jid in job_ids]
        )

        max_T, min_T = 1e6, 100.0
        last_max_T = max_T
        x_best = None
        while x_best is None:
            while 1.05 * min_T < max_T:
                T = 0.5 * (min_T + max_T)
                x = self._feasible(T, m, n, throughputs, steps, sfa)
                if x is not None:
                    x_best, max_T = x, T
                else:
                    min_T = T
            if x_best is not None:
                break
            min_T, max_T = last_max_T, last_max_T * 10.0
            last_max_T *= 10.0
            if last_max_T > 1e14:
                return None
        return self.unflatten(self.clip_allocation(x_best), index)


class MinTotalDurationPolicy(Policy):
    """Collapses throughputs to the canonical worker type before solving
    (min_total_duration.py:31-41)."""

    name = "MinTotalDuration"

    def __init__(self):
        self._perf = MinTotalDurationPolicyWithPerf()

    def get_allocation(
        self, unflattened_throughputs, scale_factors, num_steps_remaining, cluster_spec
    ):
        canonical = None
        sample = next(iter(unflattened_throughputs.values()), {})
        for wt in ("mi355x", "v100"):
            if wt in sample:
                canonical = wt
                break
        new_tputs = {
            jid: {wt: (per_wt[canonical] if canonical else max(per_wt.values()))
                  for wt in per_wt}
            for jid, per_wt in unflattened_throughputs.items()
        }
        return self._perf.get_allocation(
            new_tputs, scale_factors, num_steps_remaining, cluster_spec
        )


class MinTotalDurationPolicyWithPacking(PolicyWithPacking):
    """Packed OSSP: bisection on horizon T where the feasibility LP ranges
    over singles AND pairs, and a single job's finish constraint sums its
    effective throughput over every combination row involving it
    (reference min_total_duration.py:135-234)."""

    name = "MinTotalDuration_Packing"

    def _feasible(self, T, m, n, w, steps, sfa, job_ids, singles):
        A_ub, b_ub = self.packed_constraints(m, n, sfa, job_ids, singles)
        rows = np.zeros((len(singles), m * n))
        for k in range(len(singles)):
            rows[k] = -w[k]
        A = np.vstack([A_ub, rows])
        b = np.concatenate([b_ub, -steps / T])
        res = self.solve_lp(np.zeros(m * n), A, b)
        return res.x[: m * n].reshape((m, n)) if res.success else None

    def get_allocation(
        self, unflattened_throughputs, scale_factors, num_steps_remaining,
        cluster_spec
    ):
        all_tputs, index, singles = self.flatten_packed(
            unflattened_throughputs, cluster_spec
        )
        if all_tputs is None:
            return None
        job_ids, worker_types = index
        m, n = all_tputs[0].shape
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        w = np.array([t.reshape(-1) for t in all_tputs])
        steps = np.array(
            [max(0.0, num_steps_remaining[s]) for s in singles]
        )

        max_T, min_T = 1e6, 100.0
        last_max_T = max_T
        x_best = None
        while x_best is None:
            while 1.05 * min_T < max_T:
                T = 0.5 * (min_T + max_T)
                x = self._feasible(T, m, n, w, steps, sfa, job_ids, singles)
                if x is not None:
                    x_best, max_T = x, T
                else:
                    min_T = T
            if x_best is not None:
                break
            min_T, max_T = last_max_T, last_max_T * 10.0
            last_max_T *= 10.0
            if last_max_T > 1e14:
                return None
        return self.unflatten(self.clip_allocation(x_best), index)
