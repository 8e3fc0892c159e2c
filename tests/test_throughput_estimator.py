"""Throughput-estimator (matrix completion) tests."""

import numpy as np
import pytest

from shockwave_amd.core.throughput_estimator import (
    ThroughputEstimator,
    als_matrix_completion,
    cosine_distance,
)
from shockwave_amd.core.throughputs import read_throughputs


class TestALS:
    def test_recovers_low_rank(self):
        rng = np.random.RandomState(0)
        U = rng.rand(12, 3)
        V = rng.rand(15, 3)
        M = np.clip(U @ V.T, 0, 1)
        mask = (rng.rand(*M.shape) < 0.7).astype(float)
        out = als_matrix_completion(M * mask, mask, k=3)
        err = np.abs(out - M)[mask == 0].mean()
        assert err < 0.15

    def test_observed_entries_preserved(self):
        M = np.random.RandomState(1).rand(5, 5)
        mask = np.ones_like(M)
        mask[0, 0] = 0
        out = als_matrix_completion(M, mask)
        assert np.allclose(out[mask == 1], M[mask == 1])


class TestEstimator:
    @pytest.fixture(scope="class")
    def estimator(self, oracle_path=None):
        import os

        path = os.path.join(
            os.path.dirname(__file__), "..", "traces", "mi355x_throughputs.json"
        )
        oracle = read_throughputs(path)
        wt = "mi355x"
        job_types = [
            k for k in oracle[wt] if k[1] == 1
        ][:12]
        return ThroughputEstimator(
            oracle, [wt], job_types,
            num_reference_job_types=8, profiling_percentage=0.6, seed=0,
        ), job_types

    def test_full_profile_matches_self(self):
        import os

        path = os.path.join(
            os.path.dirname(__file__), "..", "traces", "mi355x_throughputs.json"
        )
        oracle = read_throughputs(path)
        wt = "mi355x"
        job_types = [k for k in oracle[wt] if k[1] == 1][:10]
        est = ThroughputEstimator(
            oracle, [wt], job_types,
            num_reference_job_types=len(job_types),
            profiling_percentage=1.0, seed=0,
        )
        for jt in job_types:
            assert est.match_job_to_reference_job(jt) == jt

    def test_partial_profile_returns_reference_type(self, estimator):
        est, job_types = estimator
        for jt in job_types[:5]:
            match = est.match_job_to_reference_job(jt)
            assert match in est.get_reference_throughputs()[0]

    def test_cosine_distance(self):
        a = np.array([1.0, 0.0])
        assert cosine_distance(a, a) == pytest.approx(0.0)
        assert cosine_distance(a, np.array([0.0, 1.0])) == pytest.approx(1.0)
