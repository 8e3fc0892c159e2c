"""AlloX policy — Hungarian assignment of jobs to (worker, order) slots.

Reference: policies/allox.py:13-188.  Each unallocated job may be placed at
position k on some worker; serving a job at position k delays it by k times
its own processing time, so the cost matrix is
``q[i, (k-1)*n + j] = k * steps_i/tput_ij + t_elapsed_i`` and a min-cost
perfect matching (scipy ``linear_sum_assignment``) picks the schedule.
Only the first job per worker is allocated this round; running jobs keep
their previous allocation.
"""

from __future__ import annotations

import copy

import numpy as np
from scipy.optimize import linear_sum_assignment

from .base import Policy


class AlloXPolicy(Policy):
    name = "AlloX_Perf"

    def __init__(self, alpha=1.0):
        self._alpha = alpha
        self._prev_allocation = {}

    def get_allocation(
        self,
        unflattened_throughputs,
        scale_factors,
        times_since_start,
        num_steps_remaining,
        per_round_schedule,
        cluster_spec,
    ):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        job_ids, worker_types = index

        unallocated, already_allocated = [], []
        for jid in unflattened_throughputs:
            prev = self._prev_allocation.get(jid)
            if prev is not None and sum(prev.values()) == 1.0:
                already_allocated.append(jid)
            else:
                unallocated.append(jid)

        # expand worker types into individual unoccupied workers
        worker_id_to_type = {}
        n = 0
        for wt in worker_types:
            num = cluster_spec[wt]
            for jid in already_allocated:
                if self._prev_allocation[jid][wt] == 1.0:
                    num -= 1
            for _ in range(max(0, num)):
                worker_id_to_type[n] = wt
                n += 1

        unallocated.sort(key=lambda j: -times_since_start[j])
        m = len(unallocated)
        unallocated = unallocated[: max(int(self._alpha * m), n)]
        m = len(unallocated)

        allocation = {
            jid: {wt: 0.0 for wt in cluster_spec} for jid in unflattened_throughputs
        }
        for jid in unflattened_throughputs:
            if jid in self._prev_allocation:
                allocation[jid] = copy.copy(self._prev_allocation[jid])

        if m > 0 and n > 0:
            q_base = np.zeros((m, n))
            d_base = np.zeros((m, n))
            for i, jid in enumerate(unallocated):
                for j in range(n):
                    tput = unflattened_throughputs[jid][worker_id_to_type[j]]
                    q_base[i, j] = num_steps_remaining[jid] / max(tput, 1e-10)
                    d_base[i, j] = times_since_start[jid]
            q = np.concatenate([k * q_base + d_base for k in range(1, m + 1)], axis=1)

            rows, cols = linear_sum_assignment(q)
            per_worker = {j: [] for j in range(n)}
            for r, c in zip(rows, cols):
                per_worker[c % n].append((unallocated[r], c // n))
            for j in range(n):
                entries = [
                    (jid, len(per_worker[j]) - 1 - order)
                    for jid, order in per_worker[j]
                ]
                entries.sort(key=lambda e: e[1])
                if entries:
                    jid = entries[0][0]
                    allocation[jid][worker_id_to_type[j]] = (
                        1.0 / scale_factors[jid]
                    )

        self._prev_allocation = copy.copy(allocation)
        return allocation
