"""Throughput estimator — match unseen jobs to profiled reference types.

Rebuild of the reference's ThroughputEstimator
(scheduler/throughput_estimator.py:17-204), used by packing policies to
price colocation for jobs without full pairwise profiles:

1. the oracle's pairwise colocated throughputs are normalized by isolated
   throughput into a (job_type x worker*job_type) matrix in [0, 1],
2. a new job is profiled against a random subset
   (``profiling_percentage``) of the reference job types,
3. the missing entries are filled by low-rank matrix completion — the
   reference uses the ``matrix_completion`` package's PMF solver, which is
   not available here, so we use an equivalent alternating-least-squares
   factorization (rank k=10, ridge mu=1e-2, same defaults),
4. the completed row is cosine-matched to the nearest reference job type.
"""

from __future__ import annotations

import random
from typing import Dict, List

import numpy as np

DEFAULT_K = 10
DEFAULT_MU = 1e-2


def cosine_distance(a: np.ndarray, b: np.ndarray) -> float:
    denom = np.linalg.norm(a) * np.linalg.norm(b)
    if denom == 0:
        return 1.0
    return 1.0 - float(np.dot(a, b) / denom)


def als_matrix_completion(
    matrix: np.ndarray, mask: np.ndarray, k: int = DEFAULT_K,
    mu: float = DEFAULT_MU, iters: int = 50, seed: int = 0,
) -> np.ndarray:
    """Fill unobserved entries (mask==0) with a rank-k ALS factorization."""
    rng = np.random.RandomState(seed)
    n, m = matrix.shape
    k = min(k, n, m)
    U = rng.randn(n, k) * 0.1
    V = rng.randn(m, k) * 0.1
    eye = mu * np.eye(k)
    for _ in range(iters):
        for i in range(n):
            idx = mask[i] > 0
            if not idx.any():
                continue
            Vi = V[idx]
            U[i] = np.linalg.solve(Vi.T @ Vi + eye, Vi.T @ matrix[i, idx])
        for j in range(m):
            idx = mask[:, j] > 0
            if not idx.any():
                continue
            Uj = U[idx]
            V[j] = np.linalg.solve(Uj.T @ Uj + eye, Uj.T @ matrix[idx, j])
    completed = U @ V.T
    out = matrix.copy()
    out[mask == 0] = completed[mask == 0]
    return np.clip(out, 0.0, 1.0)


class ThroughputEstimator:
    def __init__(
        self,
        oracle_throughputs: Dict,
        worker_types: List[str],
        job_types: List,
        num_reference_job_types: int,
        profiling_percentage: float,
        seed: int = 0,
    ):
        self._rng = random.Random(seed)
        self._oracle = oracle_throughputs
        self._worker_types = worker_types
        self._job_types = list(job_types)
        self._m = len(worker_types)
        self._n = len(self._job_types)
        self._profiling_percentage = profiling_percentage
        self._build_normalized()
        self._pick_reference(num_reference_job_types)

    def _build_normalized(self):
        m, n = self._m, self._n
        self._normalized = np.zeros((n, m * n), dtype=np.float32)
        for i, jt in enumerate(self._job_types):
            for j, wt in enumerate(self._worker_types):
                per_wt = self._oracle[wt][jt]
                iso = per_wt["null"]
                for k, other in enumerate(self._job_types):
                    pair = per_wt.get(other)
                    if pair is not None and iso > 0:
                        self._normalized[i, j * n + k] = min(
                            1.0, max(0.0, pair[0] / iso)
                        )

    def _pick_reference(self, num_reference: int):
        idx = sorted(
            self._rng.sample(range(self._n), min(num_reference, self._n))
        )
        self._reference_job_types = [self._job_types[i] for i in idx]
        cols = []
        for i in range(self._m):
            cols += [x + i * self._n for x in idx]
        self._reference_throughputs = self._normalized[idx][:, cols]
        self._reference_cols = cols

    def get_reference_throughputs(self):
        return self._reference_job_types, self._reference_throughputs

    def _profile_job(self, true_job_type):
        """Sampled profiling observations of the new job vs reference
        types (the simulator reads them from the oracle)."""
        i = self._job_types.index(true_job_type)
        row = np.zeros(len(self._reference_cols), dtype=np.float32)
        mask = np.zeros(len(self._reference_cols), dtype=np.float32)
        for c, col in enumerate(self._reference_cols):
            if self._rng.uniform(0, 1) <= self._profiling_percentage:
                row[c] = self._normalized[i, col]
                mask[c] = 1.0
        return row, mask

    def match_job_to_reference_job(self, true_job_type):
        row, mask = self._profile_job(true_job_type)
        if mask.sum() == 0:
            return self._reference_job_types[0]
        if (mask == 0).any():
            stacked = np.vstack([self._reference_throughputs, row])
            stacked_mask = np.vstack(
                [np.ones_like(self._reference_throughputs), mask]
            )
            completed = als_matrix_completion(stacked, stacked_mask)
            row = completed[-1]
        dists = [
            cosine_distance(row, ref) for ref in self._reference_throughputs
        ]
        return self._reference_job_types[int(np.argmin(dists))]
