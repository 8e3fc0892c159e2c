"""Policy framework: allocation matrices + LP assembly over scipy/HiGHS.

Rebuild of the reference's Policy/PolicyWithPacking (policies/policy.py:
11-260) with scipy.optimize.linprog as the LP engine instead of cvxpy/ECOS.

A policy maps job→worker-type throughputs to an allocation
``{job_id: {worker_type: fraction-of-time}}`` subject to the base
constraints (policy.py:59-65):

* ``x >= 0``
* per worker type j: ``sum_i scale_factor_i * x[i,j] <= num_workers[j]``
* per job i: ``sum_j x[i,j] <= 1``
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import numpy as np
from scipy.optimize import linprog

from ..core.job import JobIdPair


class Policy:
    name = "Policy"

    def flatten(self, d: Dict, cluster_spec: Dict[str, int]):
        """2-level dict -> (m x n array, (job_ids, worker_types))."""
        job_ids = sorted(d.keys())
        if not job_ids:
            return None, None
        worker_types = sorted(d[job_ids[0]].keys())
        if not worker_types:
            return None, None
        self._num_workers = np.array(
            [cluster_spec[wt] for wt in worker_types], dtype=float
        )
        mat = np.array(
            [[d[job_id][wt] for wt in worker_types] for job_id in job_ids],
            dtype=float,
        )
        return mat, (job_ids, worker_types)

    def unflatten(self, m: np.ndarray, index) -> Dict:
        job_ids, worker_types = index
        return {
            job_ids[i]: {worker_types[j]: m[i][j] for j in range(len(worker_types))}
            for i in range(len(job_ids))
        }

    def scale_factors_array(self, scale_factors, job_ids, m, n) -> np.ndarray:
        out = np.zeros((m, n))
        for i, job_id in enumerate(job_ids):
            if isinstance(job_id, JobIdPair) and job_id.is_pair():
                sfs = {scale_factors[s] for s in job_id.singletons()}
                sf = sfs.pop() if len(sfs) == 1 else 0
            else:
                sf = scale_factors[job_id]
            out[i, :] = sf
        return out

    # -- LP assembly --------------------------------------------------------

    def base_constraints(
        self, m: int, n: int, scale_factors_array: np.ndarray, extra_vars: int = 0
    ) -> Tuple[np.ndarray, np.ndarray]:
        """A_ub, b_ub rows for the base constraints over variables
        [x(0,0)..x(m-1,n-1), extras...] (x flattened row-major)."""
        nv = m * n + extra_vars
        rows, rhs = [], []
        for j in range(n):
            row = np.zeros(nv)
            for i in range(m):
                row[i * n + j] = scale_factors_array[i, j]
            rows.append(row)
            rhs.append(self._num_workers[j])
        for i in range(m):
            row = np.zeros(nv)
            row[i * n : (i + 1) * n] = 1.0
            rows.append(row)
            rhs.append(1.0)
        return np.array(rows), np.array(rhs)

    def solve_lp(
        self,
        c: np.ndarray,
        A_ub: Optional[np.ndarray],
        b_ub,
        A_eq=None,
        b_eq=None,
        bounds=None,
    ):
        res = linprog(
            c,
            A_ub=A_ub,
            b_ub=b_ub,
            A_eq=A_eq,
            b_eq=b_eq,
            bounds=bounds if bounds is not None else (0, None),
            method="highs",
        )
        return res

    @staticmethod
    def clip_allocation(x: np.ndarray) -> np.ndarray:
        return np.clip(x, 0.0, 1.0)


class PolicyWithPacking(Policy):
    """Adds pairwise-colocation handling: throughput dict keys are
    JobIdPairs; pairs carry per-singleton throughput tuples."""

    def flatten_packed(self, d: Dict, cluster_spec: Dict[str, int]):
        """Returns (per-single-job throughput matrices, index, singles).

        all_throughputs[k] is an (m x n) matrix giving single job k's
        throughput in each row-combination (0 where the combination doesn't
        involve job k); mirrors PolicyWithPacking.flatten
        (policy.py:87-200)."""
        job_ids = sorted(d.keys())
        if not job_ids:
            return None, None, None
        worker_types = sorted(d[job_ids[0]].keys())
        self._num_workers = np.array(
            [cluster_spec[wt] for wt in worker_types], dtype=float
        )
        single_job_ids = sorted(
            {s for jid in job_ids for s in jid.singletons()}
        )
        m, n = len(job_ids), len(worker_types)
        self._row_of = {jid: i for i, jid in enumerate(job_ids)}
        all_throughputs = []
        for single in single_job_ids:
            mat = np.zeros((m, n))
            for i, jid in enumerate(job_ids):
                if not jid.overlaps_with(single):
                    continue
                for j, wt in enumerate(worker_types):
                    v = d[jid][wt]
                    if jid.is_pair():
                        k = 0 if jid.singletons()[0] == single else 1
                        mat[i, j] = v[k]
                    else:
                        mat[i, j] = v
            all_throughputs.append(mat)
        return all_throughputs, (job_ids, worker_types), single_job_ids

    def packed_constraints(
        self,
        m: int,
        n: int,
        scale_factors_array: np.ndarray,
        job_ids,
        single_job_ids,
        extra_vars: int = 0,
    ) -> Tuple[np.ndarray, np.ndarray]:
        """A_ub, b_ub for the packed base constraints (policy.py:202-236):
        per-worker-type capacity over ALL rows (singles and pairs), plus,
        per SINGLE job, total time fraction across every combination row
        involving it <= 1 (replaces the per-row constraint of
        ``base_constraints`` — a job colocated in two pairs still only has
        one unit of wall-clock)."""
        nv = m * n + extra_vars
        rows, rhs = [], []
        for j in range(n):
            row = np.zeros(nv)
            for i in range(m):
                row[i * n + j] = scale_factors_array[i, j]
            rows.append(row)
            rhs.append(self._num_workers[j])
        for single in single_job_ids:
            row = np.zeros(nv)
            for i, jid in enumerate(job_ids):
                if jid.overlaps_with(single):
                    row[i * n : (i + 1) * n] = 1.0
            rows.append(row)
            rhs.append(1.0)
        return np.array(rows), np.array(rhs)

    def isolated_single_throughputs(self, all_throughputs, single_job_ids):
        """(K x n) matrix of each single job's ISOLATED throughput — its
        own singleton row of the packed tensor (every live job has one:
        the engine registers singleton keys before any pair)."""
        K = len(single_job_ids)
        n = all_throughputs[0].shape[1]
        iso = np.zeros((K, n))
        for k, single in enumerate(single_job_ids):
            i = self._row_of.get(single)
            if i is not None:
                iso[k] = all_throughputs[k][i]
            else:  # defensive: best over combinations
                iso[k] = np.max(all_throughputs[k], axis=0)
        return iso
