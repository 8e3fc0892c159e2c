"""FIFO policies.

Reference: policies/fifo.py:1-219.  Base mode: jobs are granted whole
workers in arrival (job-id) order and keep them until completion; perf mode
recomputes each round placing jobs on their fastest worker type; packing
mode additionally merges a queued job onto a scheduled job when the
combined normalized throughput clears a threshold.
"""

from __future__ import annotations

import copy
import random

from ..core.job import JobIdPair
from .base import Policy, PolicyWithPacking


class FIFOPolicy(Policy):
    name = "FIFO"

    def __init__(self, mode="base", seed=None, packing_threshold=1.5):
        self._mode = mode
        self._allocation = {}  # job_id -> worker_type
        self._rng = random.Random(seed)
        self._packing_threshold = packing_threshold

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        throughputs = unflattened_throughputs
        if not throughputs:
            return None
        worker_types = sorted(next(iter(throughputs.values())).keys())
        available = copy.deepcopy(cluster_spec)

        if self._mode != "base":
            self._allocation = {}

        # drop allocations of completed jobs
        self._allocation = {
            jid: wt for jid, wt in self._allocation.items() if jid in throughputs
        }
        for jid, wt in self._allocation.items():
            available[wt] -= scale_factors[jid]

        queue = [
            jid
            for jid in sorted(throughputs.keys())
            if jid not in self._allocation and not (isinstance(jid, JobIdPair) and jid.is_pair())
        ]

        for jid in queue:
            sf = scale_factors[jid]
            if self._mode == "base":
                candidates = [wt for wt in worker_types if available[wt] >= sf]
                if not candidates:
                    continue
                wt = self._rng.choice(candidates)
            else:
                # perf: fastest worker type with capacity
                candidates = sorted(
                    (wt for wt in worker_types if available[wt] >= sf),
                    key=lambda w: -throughputs[jid][w],
                )
                if not candidates:
                    continue
                wt = candidates[0]
            self._allocation[jid] = wt
            available[wt] -= sf

        allocation = {}
        for jid in throughputs:
            allocation[jid] = {wt: 0.0 for wt in worker_types}
            if jid in self._allocation:
                allocation[jid][self._allocation[jid]] = 1.0
        return allocation


class FIFOPolicyWithPerf(FIFOPolicy):
    name = "FIFO_Perf"

    def __init__(self):
        super().__init__(mode="perf")


class FIFOPolicyWithPacking(PolicyWithPacking):
    name = "FIFO_Packing"

    def __init__(self, packing_threshold=1.5):
        self._packing_threshold = packing_threshold
        self._base = FIFOPolicy(mode="perf")

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        # separate singles; run perf-FIFO; then merge queued jobs into pairs
        singles = {
            jid: v
            for jid, v in unflattened_throughputs.items()
            if not jid.is_pair()
        }
        sf_single = {jid: scale_factors[jid] for jid in singles}
        alloc = self._base.get_allocation(singles, sf_single, cluster_spec)
        if alloc is None:
            return None
        worker_types = sorted(next(iter(singles.values())).keys())
        scheduled = {
            jid for jid, a in alloc.items() if any(v > 0 for v in a.values())
        }
        queue = [jid for jid in sorted(singles) if jid not in scheduled]
        for jid in queue:
            best, best_score = None, self._packing_threshold
            for sched in sorted(scheduled):
                if sched.is_pair():
                    # at most two jobs share a GPU; an existing pair is
                    # not a packing candidate (and has no scale_factors
                    # entry of its own)
                    continue
                if scale_factors[sched] != scale_factors[jid]:
                    continue
                pair = JobIdPair(sched[0], jid[0])
                if pair not in unflattened_throughputs:
                    continue
                wt = max(worker_types, key=lambda w: alloc[sched][w])
                pair_tput = unflattened_throughputs[pair][wt]
                score = 0.0
                for k, single in enumerate(pair.singletons()):
                    iso = unflattened_throughputs[single][wt]
                    if iso > 0 and pair_tput[k] > 0:
                        score += pair_tput[k] / iso
                if score > best_score:
                    best, best_score, best_wt = sched, score, wt
            if best is not None:
                pair = JobIdPair(best[0], jid[0])
                alloc[pair] = {w: 0.0 for w in worker_types}
                alloc[pair][best_wt] = 1.0
                alloc[best] = {w: 0.0 for w in worker_types}
                scheduled.discard(best)
                scheduled.add(pair)
        # fill in zero rows for any pair keys present in input
        for jid in unflattened_throughputs:
            if jid not in alloc:
                alloc[jid] = {w: 0.0 for w in worker_types}
        return alloc
