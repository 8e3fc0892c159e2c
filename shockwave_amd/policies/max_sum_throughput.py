"""Max-sum-throughput (MST) policies.

Reference: policies/max_sum_throughput.py:1-198.  Maximize total cluster
throughput, optionally cost-normalized and with per-job SLO floor
constraints.  Pure LP.
"""

from __future__ import annotations

import numpy as np

from .base import Policy, PolicyWithPacking


class ThroughputNormalizedByCostSumWithPerfSLOs(Policy):
    name = "ThroughputNormalizedByCostSum_PerfSLOs"

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        cluster_spec,
        instance_costs=None,
        SLOs=None,
        num_steps_remaining=None,
    ):
        SLOs = SLOs or {}
        num_steps_remaining = num_steps_remaining or {}
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        costs = np.ones(n)
        if instance_costs is not None:
            costs = np.array([instance_costs[wt] for wt in worker_types])

        c = -(throughputs / costs).reshape(-1)  # maximize sum
        A_ub, b_ub = self.base_constraints(m, n, sfa)

        slo_rows, slo_rhs = [], []
        for jid, slo in SLOs.items():
            i = job_ids.index(jid)
            row = np.zeros(m * n)
            row[i * n : (i + 1) * n] = -throughputs[i]
            slo_rows.append(row)
            slo_rhs.append(-num_steps_remaining[jid] / slo)
        if slo_rows:
            A = np.vstack([A_ub, np.array(slo_rows)])
            b = np.concatenate([b_ub, np.array(slo_rhs)])
        else:
            A, b = A_ub, b_ub
        res = self.solve_lp(c, A, b)
        if not res.success and slo_rows:
            res = self.solve_lp(c, A_ub, b_ub)  # drop SLOs if infeasible
        if not res.success:
            return None
        return self.unflatten(
            self.clip_allocation(res.x[: m * n].reshape((m, n))), index
        )


class ThroughputSumWithPerf(Policy):
    name = "ThroughputSumWithPerf"

    def __init__(self):
        self._policy = ThroughputNormalizedByCostSumWithPerfSLOs()

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        return self._policy.get_allocation(
            unflattened_throughputs, scale_factors, cluster_spec
        )


class ThroughputNormalizedByCostSumWithPerf(Policy):
    name = "ThroughputNormalizedByCostSum_Perf"

    def __init__(self):
        self._policy = ThroughputNormalizedByCostSumWithPerfSLOs()

    def get_allocation(
        self, unflattened_throughputs, scale_factors, cluster_spec, instance_costs
    ):
        return self._policy.get_allocation(
            unflattened_throughputs,
            scale_factors,
            cluster_spec,
            instance_costs=instance_costs,
        )


class ThroughputNormalizedByCostSumWithPackingSLOs(PolicyWithPacking):
    """Packed MST with SLOs: maximize the sum over SINGLE jobs of
    cost-normalized effective throughput (summed across every combination
    row involving the single), subject to packed base constraints and,
    for SLO jobs, a finish-by-deadline floor
    (reference max_sum_throughput.py:118-198)."""

    name = "ThroughputNormalizedByCostSum_PackingSLOs"

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        cluster_spec,
        instance_costs=None,
        SLOs=None,
        num_steps_remaining=None,
    ):
        SLOs = SLOs or {}
        num_steps_remaining = num_steps_remaining or {}
        all_tputs, index, singles = self.flatten_packed(
            unflattened_throughputs, cluster_spec
        )
        if all_tputs is None:
            return None
        job_ids, worker_types = index
        m, n = all_tputs[0].shape
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        costs = np.ones(n)
        if instance_costs is not None:
            costs = np.array([instance_costs[wt] for wt in worker_types])

        # objective: sum over singles of their (cost-normalized) rate
        c = np.zeros(m * n)
        for t in all_tputs:
            c -= (t / costs).reshape(-1)

        A_ub, b_ub = self.packed_constraints(m, n, sfa, job_ids, singles)
        slo_rows, slo_rhs = [], []
        for s, slo in SLOs.items():
            if s not in singles:
                continue
            k = singles.index(s)
            slo_rows.append(-all_tputs[k].reshape(-1))
            slo_rhs.append(-num_steps_remaining[s] / slo)
        if slo_rows:
            A = np.vstack([A_ub, np.array(slo_rows)])
            b = np.concatenate([b_ub, np.array(slo_rhs)])
        else:
            A, b = A_ub, b_ub
        res = self.solve_lp(c, A, b)
        if not res.success and slo_rows:
            res = self.solve_lp(c, A_ub, b_ub)  # drop SLOs if infeasible
        if not res.success:
            return None
        return self.unflatten(
            self.clip_allocation(res.x[: m * n].reshape((m, n))), index
        )
