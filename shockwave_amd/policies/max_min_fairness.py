"""Max-min fairness (Gavel's LAS) policies via LP.

Reference: policies/max_min_fairness.py:1-410.  Maximize the minimum
(priority- and share-normalized) effective throughput across jobs:

    max min_i sum_j (w_ij * x_ij)     w_ij = tput_ij * prio_i * sf_i

LAS variant replaces throughputs with 1.0 so allocations equalize *time*
share rather than throughput (max_min_fairness.py:33-44).
"""

from __future__ import annotations

import numpy as np

from .base import Policy, PolicyWithPacking
from .simple import ProportionalPolicy


class MaxMinFairnessPolicyWithPerf(Policy):
    name = "MaxMinFairness_Perf"

    def __init__(self):
        self._proportional = ProportionalPolicy()

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        priority = np.array(
            [1.0 / unflattened_priority_weights[jid] for jid in job_ids]
        )
        proportional_tputs = self._proportional.get_throughputs(
            throughputs, index, cluster_spec
        )
        priority = priority.reshape((m, 1)) / proportional_tputs.reshape((m, 1))

        weights = throughputs * priority * sfa

        # LP vars: [x flattened (m*n), t]; maximize t
        nv = m * n + 1
        A_ub, b_ub = self.base_constraints(m, n, sfa, extra_vars=1)
        rows = []
        for i in range(m):
            row = np.zeros(nv)
            row[i * n : (i + 1) * n] = -weights[i]
            row[-1] = 1.0  # t - w_i . x_i <= 0
            rows.append(row)
        A_ub = np.vstack([A_ub, np.array(rows)])
        b_ub = np.concatenate([b_ub, np.zeros(m)])
        c = np.zeros(nv)
        c[-1] = -1.0
        res = self.solve_lp(c, A_ub, b_ub)
        if not res.success:
            return None
        x = self.clip_allocation(res.x[: m * n].reshape((m, n)))
        return self.unflatten(x, index)


class MaxMinFairnessPolicy(Policy):
    """LAS: unit throughputs (time-fair, not throughput-fair)."""

    name = "MaxMinFairness"

    def __init__(self):
        self._perf = MaxMinFairnessPolicyWithPerf()

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        priority_weights,
        cluster_spec,
    ):
        ones = {
            jid: {wt: 1.0 for wt in per_wt}
            for jid, per_wt in unflattened_throughputs.items()
        }
        return self._perf.get_allocation(
            ones, scale_factors, priority_weights, cluster_spec
        )


class MaxMinFairnessPolicyWithPacking(PolicyWithPacking):
    """Packed max-min fairness: allocation rows span singles AND colocated
    pairs; maximize the minimum normalized effective rate over SINGLE jobs,
    where a single's rate sums its throughput contribution from every
    combination row it appears in (reference max_min_fairness.py:220-410).
    """

    name = "MaxMinFairness_Packing"

    def __init__(self):
        self._proportional = ProportionalPolicy()

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
    ):
        all_tputs, index, singles = self.flatten_packed(
            unflattened_throughputs, cluster_spec
        )
        if all_tputs is None:
            return None
        job_ids, worker_types = index
        m, n = all_tputs[0].shape
        K = len(singles)
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        iso = self.isolated_single_throughputs(all_tputs, singles)
        prop = self._proportional.get_throughputs(
            iso, (singles, worker_types), cluster_spec
        ).reshape(-1)
        prio = np.array(
            [1.0 / unflattened_priority_weights[s] for s in singles]
        )
        sf_s = np.array([scale_factors[s] for s in singles], dtype=float)
        scale = prio * sf_s / np.maximum(prop, 1e-10)

        nv = m * n + 1
        A_ub, b_ub = self.packed_constraints(
            m, n, sfa, job_ids, singles, extra_vars=1
        )
        rows = []
        for k in range(K):
            row = np.zeros(nv)
            row[: m * n] = -(all_tputs[k] * scale[k]).reshape(-1)
            row[-1] = 1.0  # t - rate_k <= 0
            rows.append(row)
        A = np.vstack([A_ub, np.array(rows)])
        b = np.concatenate([b_ub, np.zeros(K)])
        c = np.zeros(nv)
        c[-1] = -1.0
        res = self.solve_lp(c, A, b)
        if not res.success:
            return None
        x = self.clip_allocation(res.x[: m * n].reshape((m, n)))
        return self.unflatten(x, index)


class MaxMinFairnessWaterFillingPolicyWithPerf(Policy):
    """Water-filling max-min fairness: iteratively maximize the minimum
    normalized rate, freeze saturated jobs at the achieved level, recurse on
    the rest (reference max_min_fairness_water_filling.py:475-569; this
    is the standard algorithm, not a translation)."""

    name = "MaxMinFairnessWaterFilling_Perf"

    def __init__(self, priority_reweighting_policies=None, max_iterations=64):
        self._max_iterations = max_iterations

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
        **kwargs,
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        priority = np.array(
            [1.0 / unflattened_priority_weights[jid] for jid in job_ids]
        ).reshape((m, 1))
        weights = throughputs * priority * sfa

        frozen_level = {}  # i -> minimum rate locked in
        x_final = np.zeros((m, n))
        for _ in range(self._max_iterations):
            free = [i for i in range(m) if i not in frozen_level]
            if not free:
                break
            nv = m * n + 1
            A_ub, b_ub = self.base_constraints(m, n, sfa, extra_vars=1)
            rows, rhs = [], []
            for i in free:
                row = np.zeros(nv)
                row[i * n : (i + 1) * n] = -weights[i]
                row[-1] = 1.0
                rows.append(row)
                rhs.append(0.0)
            for i, level in frozen_level.items():
                row = np.zeros(nv)
                row[i * n : (i + 1) * n] = -weights[i]
                rows.append(row)
                rhs.append(-level)
            A_ub = np.vstack([A_ub, np.array(rows)])
            b_ub = np.concatenate([b_ub, np.array(rhs)])
            c = np.zeros(nv)
            c[-1] = -1.0
            res = self.solve_lp(c, A_ub, b_ub)
            if not res.success:
                break
            t_star = res.x[-1]
            x_star = res.x[: m * n].reshape((m, n))
            x_final = x_star
            # freeze jobs that can't exceed t_star: test each free job by
            # checking if its rate can be raised with others fixed >= t_star
            newly_frozen = False
            for i in free:
                c2 = np.zeros(nv)
                c2[i * n : (i + 1) * n] = -weights[i]
                rows2 = []
                rhs2 = []
                for k in free:
                    if k == i:
                        continue
                    row = np.zeros(nv)
                    row[k * n : (k + 1) * n] = -weights[k]
                    rows2.append(row)
                    rhs2.append(-t_star + 1e-9)
                for k, level in frozen_level.items():
                    row = np.zeros(nv)
                    row[k * n : (k + 1) * n] = -weights[k]
                    rows2.append(row)
                    rhs2.append(-level)
                A2 = np.vstack([A_ub[: n + m], np.array(rows2)]) if rows2 else A_ub[: n + m]
                b2 = (
                    np.concatenate([b_ub[: n + m], np.array(rhs2)])
                    if rows2
                    else b_ub[: n + m]
                )
                res2 = self.solve_lp(c2, A2, b2)
                best = -res2.fun if res2.success else t_star
                if best <= t_star * (1 + 1e-6) + 1e-9:
                    frozen_level[i] = t_star
                    newly_frozen = True
            if not newly_frozen:
                for i in free:
                    frozen_level[i] = t_star
        return self.unflatten(self.clip_allocation(x_final), index)


class MaxMinFairnessWaterFillingPolicy(Policy):
    """LAS water filling: unit throughputs (time-fair), delegated to the
    perf water filler (reference max_min_fairness_water_filling.py:416-473
    substitutes throughput 1.0 for every (job, worker_type))."""

    name = "MaxMinFairnessWaterFilling"

    def __init__(self, priority_reweighting_policies=None, max_iterations=64):
        self._perf = MaxMinFairnessWaterFillingPolicyWithPerf(
            priority_reweighting_policies, max_iterations
        )

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
        **kwargs,
    ):
        ones = {
            jid: {wt: 1.0 for wt in per_wt}
            for jid, per_wt in unflattened_throughputs.items()
        }
        return self._perf.get_allocation(
            ones, scale_factors, unflattened_priority_weights, cluster_spec
        )


class MaxMinFairnessWaterFillingPolicyWithPacking(PolicyWithPacking):
    """Water filling over packed rows: each round of filling maximizes the
    minimum normalized rate over FREE single jobs (rates summed over every
    combination row involving the single), freezes saturated singles at
    the achieved level, and recurses (reference
    max_min_fairness_water_filling.py:569-718)."""

    name = "MaxMinFairnessWaterFilling_Packing"

    def __init__(self, priority_reweighting_policies=None, max_iterations=64):
        self._max_iterations = max_iterations
        self._proportional = ProportionalPolicy()

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
        **kwargs,
    ):
        all_tputs, index, singles = self.flatten_packed(
            unflattened_throughputs, cluster_spec
        )
        if all_tputs is None:
            return None
        job_ids, worker_types = index
        m, n = all_tputs[0].shape
        K = len(singles)
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        iso = self.isolated_single_throughputs(all_tputs, singles)
        prop = self._proportional.get_throughputs(
            iso, (singles, worker_types), cluster_spec
        ).reshape(-1)
        prio = np.array(
            [1.0 / unflattened_priority_weights[s] for s in singles]
        )
        sf_s = np.array([scale_factors[s] for s in singles], dtype=float)
        scale = prio * sf_s / np.maximum(prop, 1e-10)
        # per-single weight row over the flattened x
        w = np.array(
            [(all_tputs[k] * scale[k]).reshape(-1) for k in range(K)]
        )

        base_A, base_b = self.packed_constraints(
            m, n, sfa, job_ids, singles, extra_vars=1
        )
        nv = m * n + 1
        frozen_level = {}
        x_final = np.zeros(m * n)
        for _ in range(self._max_iterations):
            free = [k for k in range(K) if k not in frozen_level]
            if not free:
                break
            rows, rhs = [], []
            for k in free:
                row = np.zeros(nv)
                row[: m * n] = -w[k]
                row[-1] = 1.0
                rows.append(row)
                rhs.append(0.0)
            for k, level in frozen_level.items():
                row = np.zeros(nv)
                row[: m * n] = -w[k]
                rows.append(row)
                rhs.append(-level)
            A = np.vstack([base_A, np.array(rows)])
            b = np.concatenate([base_b, np.array(rhs)])
            c = np.zeros(nv)
            c[-1] = -1.0
            res = self.solve_lp(c, A, b)
            if not res.success:
                break
            t_star = res.x[-1]
            x_final = res.x[: m * n]
            # freeze free singles whose rate cannot be raised above t_star
            newly_frozen = False
            for k in free:
                c2 = np.zeros(nv)
                c2[: m * n] = -w[k]
                rows2, rhs2 = [], []
                for o in free:
                    if o == k:
                        continue
                    row = np.zeros(nv)
                    row[: m * n] = -w[o]
                    rows2.append(row)
                    rhs2.append(-t_star + 1e-9)
                for o, level in frozen_level.items():
                    row = np.zeros(nv)
                    row[: m * n] = -w[o]
                    rows2.append(row)
                    rhs2.append(-level)
                A2 = (
                    np.vstack([base_A, np.array(rows2)]) if rows2 else base_A
                )
                b2 = (
                    np.concatenate([base_b, np.array(rhs2)])
                    if rows2
                    else base_b
                )
                res2 = self.solve_lp(c2, A2, b2)
                best = -res2.fun if res2.success else t_star
                if best <= t_star * (1 + 1e-6) + 1e-9:
                    frozen_level[k] = t_star
                    newly_frozen = True
            if not newly_frozen:
                for k in free:
                    frozen_level[k] = t_star
        x = self.clip_allocation(x_final.reshape((m, n)))
        return self.unflatten(x, index)


class MaxMinFairnessStrategyProofPolicy(MaxMinFairnessPolicy):
    """Strategy-proof LAS: identical to MaxMinFairness with unit
    throughputs — exactly what the reference's non-perf variant computes
    (max_min_fairness_strategy_proof.py:13-45 replaces every throughput
    with 1.0 before delegating)."""

    name = "MaxMinFairness_StrategyProof"


class MaxMinFairnessStrategyProofPolicyWithPerf(Policy):
    """Nash-bargaining allocation with VCG-style leave-one-out discounts
    (reference max_min_fairness_strategy_proof.py:47-155): maximize the
    geometric mean of weighted effective rates, then scale each job's
    allocation by the product of externalities it imposes on the others
    (rate-with-me / rate-without-me), which removes the incentive to
    misreport throughputs.

    The reference solves the geo-mean program with cvxpy; on a homogeneous
    cluster the KKT conditions give the closed form
    ``x_i = min(1, nu / sf_i)`` with nu found by bisection on the capacity
    constraint (rates w_i x_i; the log objective makes x independent of
    w beyond the weighting of the discounts)."""

    name = "MaxMinFairness_Perf_StrategyProof"

    def __init__(self):
        self._proportional = ProportionalPolicy()

    def _nash_x(self, sfa_col, num_workers):
        lo, hi = 0.0, max(num_workers, sfa_col.max()) + 1.0
        for _ in range(60):
            nu = 0.5 * (lo + hi)
            used = np.minimum(1.0, nu / sfa_col) * sfa_col
            if used.sum() > num_workers:
                hi = nu
            else:
                lo = nu
        return np.minimum(1.0, lo / sfa_col)

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        unflattened_priority_weights,
        cluster_spec,
        recurse_deeper=True,
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        m, n = throughputs.shape
        job_ids, worker_types = index
        assert n == 1, "strategy-proof perf assumes a homogeneous cluster"
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)

        if recurse_deeper:
            rates_minus_job = []
            for jid in job_ids:
                minus = dict(unflattened_throughputs)
                del minus[jid]
                # with a single job the leave-one-out economy is empty:
                # no externality, no discount
                rates = self.get_allocation(
                    minus, scale_factors, unflattened_priority_weights,
                    cluster_spec, recurse_deeper=False,
                )
                rates_minus_job.append(rates if rates is not None else {})

        priority = np.array(
            [1.0 / unflattened_priority_weights[jid] for jid in job_ids]
        ).reshape((m, 1))
        proportional_tputs = self._proportional.get_throughputs(
            throughputs, index, cluster_spec
        )
        weights = throughputs * (priority / proportional_tputs) * sfa

        x = self._nash_x(sfa[:, 0], self._num_workers[0]).reshape((m, 1))
        rates = (weights * x).sum(axis=1)
        rates_dict = {jid: rates[i] for i, jid in enumerate(job_ids)}
        if not recurse_deeper:
            return rates_dict

        discounts = np.ones(m)
        for i, jid in enumerate(job_ids):
            for other, rate_without in rates_minus_job[i].items():
                if rate_without > 0:
                    discounts[i] *= rates_dict[other] / rate_without
        discounted = np.clip(x * discounts.reshape((m, 1)), 0.0, 1.0)
        return self.unflatten(discounted, index), discounts
