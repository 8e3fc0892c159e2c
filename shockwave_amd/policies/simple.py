"""Closed-form policies: Isolated, Proportional, Gandiva-Fair.

References: policies/isolated.py:1-76, proportional.py:1-56,
gandiva_fair_proportional.py:1-58.
"""

from __future__ import annotations

import numpy as np

from .base import Policy


class IsolatedPolicy(Policy):
    """Equal cluster split per job, scale-factor-aware: each job gets
    num_workers/m GPU-fraction, divided by its scale factor, row-normalized
    to <= 1 (isolated.py:35-56)."""

    name = "Isolated"

    def _allocation(self, m, n, scale_factors_array, worker_types, cluster_spec):
        x = np.array(
            [[cluster_spec[wt] / m for wt in worker_types] for _ in range(m)],
            dtype=float,
        )
        x = x / scale_factors_array
        row_sum = np.maximum(x.sum(axis=1), 1.0)
        return x / row_sum[:, None]

    def get_throughputs(self, throughputs, index, scale_factors, cluster_spec):
        if throughputs is None:
            return None
        job_ids, worker_types = index
        m, n = throughputs.shape
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        x = self._allocation(m, n, sfa, worker_types, cluster_spec)
        return np.sum(throughputs * x, axis=1).reshape((m, 1))

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        job_ids, worker_types = index
        m, n = throughputs.shape
        sfa = self.scale_factors_array(scale_factors, job_ids, m, n)
        x = self._allocation(m, n, sfa, worker_types, cluster_spec)
        return self.unflatten(x, index)


class IsolatedPlusPolicy(IsolatedPolicy):
    """Like Isolated but a job's share is NOT divided by its scale
    factor — an sf-GPU job receives a full equal share per GPU
    (isolated_plus.py:35-56; the only diff from isolated.py is the
    removed scale-factor division at :50)."""

    name = "Isolated_Plus"

    def _allocation(self, m, n, scale_factors_array, worker_types,
                    cluster_spec):
        x = np.array(
            [[cluster_spec[wt] / m for wt in worker_types] for _ in range(m)],
            dtype=float,
        )
        row_sum = np.maximum(x.sum(axis=1), 1.0)
        return x / row_sum[:, None]


class ProportionalPolicy(Policy):
    """Cluster split proportional to worker counts, normalized by the max
    row sum (proportional.py:26-43)."""

    name = "Proportional"

    def _allocation(self, m, worker_types, cluster_spec):
        x = np.array(
            [[cluster_spec[wt] / m for wt in worker_types] for _ in range(m)],
            dtype=float,
        )
        return x / x.sum(axis=1).max()

    def get_throughputs(self, throughputs, index, cluster_spec):
        if throughputs is None:
            return None
        job_ids, worker_types = index
        m, _ = throughputs.shape
        x = self._allocation(m, worker_types, cluster_spec)
        return np.sum(throughputs * x, axis=1).reshape((m, 1))

    def get_allocation(self, unflattened_throughputs, cluster_spec):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        _, worker_types = index
        m, _ = throughputs.shape
        return self.unflatten(self._allocation(m, worker_types, cluster_spec), index)


class GandivaFairPolicy(Policy):
    """Gandiva-Fair baseline: proportional share with per-row normalization
    (gandiva_fair_proportional.py:26-41)."""

    name = "GandivaFairProportional"

    def get_allocation(self, unflattened_throughputs, scale_factors, cluster_spec):
        throughputs, index = self.flatten(unflattened_throughputs, cluster_spec)
        if throughputs is None:
            return None
        _, worker_types = index
        m, _ = throughputs.shape
        x = np.array(
            [[cluster_spec[wt] / m for wt in worker_types] for _ in range(m)],
            dtype=float,
        )
        row_sum = np.maximum(x.sum(axis=1), 1.0)
        x = x / row_sum[:, None]
        return self.unflatten(x, index)
