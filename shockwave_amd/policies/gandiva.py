"""Gandiva policy — packing heuristic with random pair exploration.

Reference: policies/gandiva.py:1-170.  When the cluster is under-subscribed
every single job gets a proportional share; when over-subscribed, unassigned
jobs are randomly paired, pairs whose combined normalized throughput >= 1
are kept (time-shared), and the resulting combinations get proportional
shares.
"""

from __future__ import annotations

import random

import numpy as np

from ..core.job import JobIdPair
from .base import PolicyWithPacking


class GandivaPolicy(PolicyWithPacking):
    name = "Gandiva_Packing"

    def __init__(self, seed=None):
        self._assigned_combinations = {}  # single_id -> (combination, other_id)
        self._rng = random.Random(seed)

    def _normalized_throughput(self, combo, throughputs, worker_types):
        if not combo.is_pair():
            return 0.0
        total = 0.0
        for wt in worker_types:
            packed = throughputs[combo][wt]
            for k, single in enumerate(combo.singletons()):
                if packed[k] <= 0.0:
                    return 0.0
                total += packed[k] / throughputs[single][wt]
        return total

    def _proportional_rows(self, combos, all_ids, worker_types, scale_factors, cluster_spec):
        m = len(combos)
        x = np.zeros((len(all_ids), len(worker_types)))
        if m == 0:
            return x
        idx = {jid: i for i, jid in enumerate(all_ids)}
        for combo in combos:
            i = idx[combo]
            sfs = {scale_factors[s] for s in combo.singletons()}
            sf = sfs.pop() if len(sfs) == 1 else max(sfs, default=1)
            row = np.array(
                [cluster_spec[wt] / m for wt in worker_types], dtype=float
            ) / max(sf, 1)
            x[i] = row
        row_sum = np.maximum(x.sum(axis=1), 1.0)
        return x / row_sum[:, None]

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        if not unflattened_throughputs:
            return None
        all_ids = sorted(unflattened_throughputs.keys())
        worker_types = sorted(next(iter(unflattened_throughputs.values())).keys())
        single_ids = sorted(jid for jid in all_ids if not jid.is_pair())

        # drop stale / no-longer-beneficial combinations
        to_delete = []
        for jid, (combo, other) in list(self._assigned_combinations.items()):
            if jid not in all_ids or (other is not None and other not in all_ids):
                to_delete += [jid, other]
                continue
            if combo.is_pair() and self._normalized_throughput(
                combo, unflattened_throughputs, worker_types
            ) < 1.0:
                to_delete += [jid, other]
        for jid in to_delete:
            if jid is not None:
                self._assigned_combinations.pop(jid, None)

        requested = sum(scale_factors[jid] for jid in single_ids)
        available = sum(cluster_spec[wt] for wt in worker_types)

        if requested <= available:
            combos = single_ids
        else:
            unassigned = [
                jid for jid in single_ids if jid not in self._assigned_combinations
            ]
            self._rng.shuffle(unassigned)
            i = 0
            while i + 1 < len(unassigned):
                a, b = unassigned[i], unassigned[i + 1]
                combo = JobIdPair(a[0], b[0])
                if (
                    combo in unflattened_throughputs
                    and scale_factors[a] == scale_factors[b]
                    and self._normalized_throughput(
                        combo, unflattened_throughputs, worker_types
                    )
                    >= 1.0
                ):
                    self._assigned_combinations[a] = (combo, b)
                    self._assigned_combinations[b] = (combo, a)
                    i += 2
                else:
                    self._assigned_combinations[a] = (JobIdPair(a[0]), None)
                    i += 1
            for jid in unassigned:
                if jid not in self._assigned_combinations:
                    self._assigned_combinations[jid] = (JobIdPair(jid[0]), None)
            combos = sorted(
                {c for c, _ in self._assigned_combinations.values() if c in all_ids
                 or not c.is_pair()},
            )
            combos = [c for c in combos if c in all_ids]

        x = self._proportional_rows(
            combos, all_ids, worker_types, scale_factors, cluster_spec
        )
        return self.unflatten(x, (all_ids, worker_types))
