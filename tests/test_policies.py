"""Policy allocation tests: LP policies checked against closed-form /
brute-force expectations on small instances."""

import numpy as np
import pytest

from shockwave_amd.core.job import JobIdPair
from shockwave_amd.policies import (
    AlloXPolicy,
    FIFOPolicy,
    FinishTimeFairnessPolicy,
    GandivaFairPolicy,
    GandivaPolicy,
    IsolatedPolicy,
    MaxMinFairnessPolicy,
    MaxMinFairnessPolicyWithPerf,
    MaxMinFairnessWaterFillingPolicy,
    MinTotalDurationPolicy,
    ProportionalPolicy,
    ThroughputSumWithPerf,
    get_policy,
)

WT = "mi355x"


def mk_tputs(vals):
    return {JobIdPair(i): {WT: v} for i, v in enumerate(vals)}


def mk_sf(n, sf=1):
    return {JobIdPair(i): sf for i in range(n)}


def mk_prio(n):
    return {JobIdPair(i): 1.0 for i in range(n)}


def total_workers_used(alloc, sf):
    return sum(alloc[j][WT] * sf[j] for j in alloc)


class TestIsolatedProportional:
    def test_isolated_equal_split(self):
        alloc = IsolatedPolicy().get_allocation(mk_tputs([1, 1]), mk_sf(2), {WT: 4})
        # 2 jobs, 4 GPUs -> each gets min(1, 2) = 1.0
        for j in alloc:
            assert alloc[j][WT] == pytest.approx(1.0)

    def test_isolated_oversubscribed(self):
        alloc = IsolatedPolicy().get_allocation(mk_tputs([1] * 8), mk_sf(8), {WT: 4})
        for j in alloc:
            assert alloc[j][WT] == pytest.approx(0.5)

    def test_proportional(self):
        # reference semantics (proportional.py:38-41): rows normalized by the
        # max row sum, so each row sums to exactly 1 with a single worker type
        alloc = ProportionalPolicy().get_allocation(mk_tputs([1] * 4), {WT: 2})
        for j in alloc:
            assert alloc[j][WT] == pytest.approx(1.0)

    def test_gandiva_fair(self):
        alloc = GandivaFairPolicy().get_allocation(
            mk_tputs([1] * 4), mk_sf(4), {WT: 2}
        )
        for j in alloc:
            assert alloc[j][WT] == pytest.approx(0.5)


class TestMaxMinFairness:
    def test_las_equalizes_time(self):
        # LAS ignores throughputs: equal time share
        alloc = MaxMinFairnessPolicy().get_allocation(
            mk_tputs([10.0, 1.0]), mk_sf(2), mk_prio(2), {WT: 1}
        )
        assert alloc[JobIdPair(0)][WT] == pytest.approx(0.5, abs=1e-4)
        assert alloc[JobIdPair(1)][WT] == pytest.approx(0.5, abs=1e-4)

    def test_perf_equalizes_throughput_ratio(self):
        # with perf, allocations weight toward the slow job to equalize
        # normalized effective throughput
        alloc = MaxMinFairnessPolicyWithPerf().get_allocation(
            mk_tputs([4.0, 1.0]), mk_sf(2), mk_prio(2), {WT: 1}
        )
        x0, x1 = alloc[JobIdPair(0)][WT], alloc[JobIdPair(1)][WT]
        assert x0 + x1 <= 1.0 + 1e-6
        # normalized rates equal: (tput0*x0)/prop0 == (tput1*x1)/prop1
        # prop_i = tput_i * 0.5 -> x0 == x1 == 0.5
        assert x0 == pytest.approx(0.5, abs=1e-3)

    def test_scale_factor_capacity(self):
        sf = {JobIdPair(0): 4, JobIdPair(1): 1}
        alloc = MaxMinFairnessPolicy().get_allocation(
            mk_tputs([1.0, 1.0]), sf, mk_prio(2), {WT: 4}
        )
        used = alloc[JobIdPair(0)][WT] * 4 + alloc[JobIdPair(1)][WT] * 1
        assert used <= 4 + 1e-6

    def test_water_filling_matches_maxmin_on_symmetric(self):
        alloc = MaxMinFairnessWaterFillingPolicy().get_allocation(
            mk_tputs([1.0, 1.0]), mk_sf(2), mk_prio(2), {WT: 1}
        )
        assert alloc[JobIdPair(0)][WT] == pytest.approx(0.5, abs=1e-3)

    def test_water_filling_uses_slack(self):
        # 3 jobs, 2 GPUs: max-min level is 2/3, but water-filling should
        # not leave capacity stranded
        alloc = MaxMinFairnessWaterFillingPolicy().get_allocation(
            mk_tputs([1.0, 1.0, 1.0]), mk_sf(3), mk_prio(3), {WT: 2}
        )
        used = sum(alloc[j][WT] for j in alloc)
        assert used == pytest.approx(2.0, abs=1e-2)


class TestThemis:
    def test_equal_jobs_equal_rho(self):
        p = FinishTimeFairnessPolicy()
        n = 2
        alloc = p.get_allocation(
            mk_tputs([2.0, 2.0]),
            mk_sf(n),
            mk_prio(n),
            {JobIdPair(i): 100.0 for i in range(n)},
            {JobIdPair(i): 1000.0 for i in range(n)},
            {WT: 1},
        )
        assert alloc[JobIdPair(0)][WT] == pytest.approx(0.5, abs=0.05)

    def test_lagging_job_favored(self):
        p = FinishTimeFairnessPolicy()
        times = {JobIdPair(0): 5000.0, JobIdPair(1): 100.0}
        steps = {JobIdPair(0): 1000.0, JobIdPair(1): 1000.0}
        alloc = p.get_allocation(
            mk_tputs([2.0, 2.0]), mk_sf(2), mk_prio(2), times, steps, {WT: 1}
        )
        # job 0 has waited much longer -> needs more share to hit same rho
        assert alloc[JobIdPair(0)][WT] > alloc[JobIdPair(1)][WT]


class TestMST:
    def test_prefers_fast_job(self):
        alloc = ThroughputSumWithPerf().get_allocation(
            mk_tputs([10.0, 1.0]), mk_sf(2), {WT: 1}
        )
        assert alloc[JobIdPair(0)][WT] == pytest.approx(1.0, abs=1e-4)

    def test_capacity_respected(self):
        alloc = ThroughputSumWithPerf().get_allocation(
            mk_tputs([5.0, 4.0, 3.0]), mk_sf(3), {WT: 2}
        )
        assert total_workers_used(alloc, mk_sf(3)) <= 2 + 1e-6


class TestOSSP:
    def test_binary_search_feasible(self):
        steps = {JobIdPair(0): 1000.0, JobIdPair(1): 4000.0}
        alloc = MinTotalDurationPolicy().get_allocation(
            mk_tputs([2.0, 2.0]), mk_sf(2), steps, {WT: 1}
        )
        # both jobs must fit in the same horizon T*:
        # T* = (1000+4000)/2 = 2500s; job1 needs 4x job0's share
        x0, x1 = alloc[JobIdPair(0)][WT], alloc[JobIdPair(1)][WT]
        assert x1 / max(x0, 1e-9) == pytest.approx(4.0, rel=0.15)


class TestFIFO:
    def test_arrival_order(self):
        p = FIFOPolicy(seed=0)
        alloc = p.get_allocation(mk_tputs([1, 1, 1]), mk_sf(3), {WT: 2})
        assert alloc[JobIdPair(0)][WT] == 1.0
        assert alloc[JobIdPair(1)][WT] == 1.0
        assert alloc[JobIdPair(2)][WT] == 0.0

    def test_sticky_until_done(self):
        p = FIFOPolicy(seed=0)
        p.get_allocation(mk_tputs([1, 1, 1]), mk_sf(3), {WT: 2})
        # job 0 completes; job 2 takes its slot
        tputs = {JobIdPair(1): {WT: 1}, JobIdPair(2): {WT: 1}}
        sf = {JobIdPair(1): 1, JobIdPair(2): 1}
        alloc = p.get_allocation(tputs, sf, {WT: 2})
        assert alloc[JobIdPair(1)][WT] == 1.0
        assert alloc[JobIdPair(2)][WT] == 1.0


class TestAllox:
    def test_assigns_workers(self):
        p = AlloXPolicy()
        n = 3
        alloc = p.get_allocation(
            mk_tputs([2.0, 1.0, 1.5]),
            mk_sf(n),
            {JobIdPair(i): float(i) for i in range(n)},
            {JobIdPair(i): 100.0 for i in range(n)},
            [],
            {WT: 2},
        )
        placed = [j for j in alloc if alloc[j][WT] > 0]
        assert len(placed) == 2


class TestGandiva:
    def test_undersubscribed_proportional(self):
        p = GandivaPolicy(seed=0)
        tputs = {JobIdPair(0): {WT: 1.0}, JobIdPair(1): {WT: 1.0}}
        sf = {JobIdPair(0): 1, JobIdPair(1): 1}
        alloc = p.get_allocation(tputs, sf, {WT: 4})
        assert alloc[JobIdPair(0)][WT] > 0


class TestFactory:
    @pytest.mark.parametrize(
        "name",
        [
            "fifo", "fifo_perf", "fifo_packed", "finish_time_fairness",
            "gandiva", "gandiva_fair", "isolated", "max_min_fairness",
            "max_min_fairness_perf", "max_min_fairness_water_filling",
            "max_sum_throughput_perf", "min_total_duration", "shockwave",
            "allox",
        ],
    )
    def test_factory(self, name):
        assert get_policy(name) is not None

    def test_unknown_raises(self):
        with pytest.raises(ValueError):
            get_policy("nope")


class TestStrategyProofPerf:
    def test_symmetric_jobs_no_discount(self):
        from shockwave_amd.policies import MaxMinFairnessStrategyProofPolicyWithPerf

        p = MaxMinFairnessStrategyProofPolicyWithPerf()
        alloc, discounts = p.get_allocation(
            mk_tputs([2.0, 2.0]), mk_sf(2), mk_prio(2), {WT: 1}
        )
        # identical jobs impose identical externalities
        assert discounts[0] == pytest.approx(discounts[1], rel=1e-6)
        assert alloc[JobIdPair(0)][WT] == pytest.approx(
            alloc[JobIdPair(1)][WT], rel=1e-6
        )

    def test_nash_capacity_respected(self):
        from shockwave_amd.policies import MaxMinFairnessStrategyProofPolicyWithPerf

        p = MaxMinFairnessStrategyProofPolicyWithPerf()
        sf = {JobIdPair(0): 2, JobIdPair(1): 1, JobIdPair(2): 1}
        alloc, _ = p.get_allocation(
            mk_tputs([1.0, 2.0, 3.0]), sf, mk_prio(3), {WT: 2}
        )
        used = sum(alloc[j][WT] * sf[j] for j in alloc)
        assert used <= 2 + 1e-6


class TestFifoPacking:
    def test_pair_not_repacked(self):
        """A formed pair must not become a packing candidate itself.

        Regression: with one GPU and three queued same-scale jobs whose
        pairwise packed throughput clears the threshold, job 1 packs onto
        job 0 forming a pair; job 2 then iterates the scheduled set which
        contains that pair — looking up scale_factors[pair] raised
        KeyError (pairs have no scale_factors entry), and Gavel packs at
        most two jobs per GPU anyway.
        """
        from shockwave_amd.policies.fifo import FIFOPolicyWithPacking

        p = FIFOPolicyWithPacking(packing_threshold=1.5)
        singles = [JobIdPair(i) for i in range(3)]
        tputs = {j: {WT: 1.0} for j in singles}
        # every pair packs at 0.9+0.9 = 1.8 > 1.5 threshold
        for a in range(3):
            for b in range(a + 1, 3):
                tputs[JobIdPair(a, b)] = {WT: (0.9, 0.9)}
        sf = {j: 1 for j in singles}
        alloc = p.get_allocation(tputs, sf, {WT: 1})
        pair01 = JobIdPair(0, 1)
        assert alloc[pair01][WT] == 1.0
        # job 2 stays queued: the GPU already holds a full pair
        assert all(v == 0.0 for v in alloc[JobIdPair(2)].values())


def mk_packed(n, iso=2.0, pair=1.4):
    """n singles, all pairs colocatable at per-job throughput ``pair``."""
    tputs = {JobIdPair(i): {WT: iso} for i in range(n)}
    for a in range(n):
        for b in range(a + 1, n):
            tputs[JobIdPair(a, b)] = {WT: (pair, pair)}
    return tputs


class TestPackedLPPolicies:
    """The reference's *_packed LP policy family (utils.py:603-686 names),
    over our packed flatten/constraint machinery (policy.py:87-236)."""

    def test_max_min_fairness_packed_prefers_pairs(self):
        from shockwave_amd.policies import MaxMinFairnessPolicyWithPacking

        # 3 jobs on 1 GPU: time-slicing singles gives each rate 2/3;
        # rotating pairs gives each 2*(1/3)*1.4 = 0.93 — LP must pack
        alloc = MaxMinFairnessPolicyWithPacking().get_allocation(
            mk_packed(3), mk_sf(3), mk_prio(3), {WT: 1}
        )
        pair_time = sum(
            alloc[j][WT] for j in alloc if j.is_pair()
        )
        assert pair_time == pytest.approx(1.0, abs=1e-3)

    def test_max_min_fairness_packed_skips_bad_pairs(self):
        from shockwave_amd.policies import MaxMinFairnessPolicyWithPacking

        # pair throughput 0.4 each (total 0.8 < 1.0 isolated-normalized):
        # packing hurts, LP should time-slice singles instead
        alloc = MaxMinFairnessPolicyWithPacking().get_allocation(
            mk_packed(2, iso=2.0, pair=0.4), mk_sf(2), mk_prio(2), {WT: 1}
        )
        assert alloc[JobIdPair(0, 1)][WT] == pytest.approx(0.0, abs=1e-3)
        assert alloc[JobIdPair(0)][WT] == pytest.approx(0.5, abs=1e-2)

    def test_min_total_duration_packed(self):
        from shockwave_amd.policies import MinTotalDurationPolicyWithPacking

        nsr = {JobIdPair(i): 1000.0 for i in range(3)}
        alloc = MinTotalDurationPolicyWithPacking().get_allocation(
            mk_packed(3), mk_sf(3), nsr, {WT: 1}
        )
        # total GPU time fraction across rows <= 1
        used = sum(alloc[j][WT] for j in alloc)
        assert used <= 1.0 + 1e-6
        assert any(alloc[j][WT] > 0 for j in alloc if j.is_pair())

    def test_finish_time_fairness_packed(self):
        from shockwave_amd.policies import FinishTimeFairnessPolicyWithPacking

        p = FinishTimeFairnessPolicyWithPacking()
        tss = {JobIdPair(i): 100.0 for i in range(3)}
        nsr = {JobIdPair(i): 500.0 for i in range(3)}
        alloc = p.get_allocation(
            mk_packed(3), mk_sf(3), mk_prio(3), tss, nsr, {WT: 1}
        )
        rates = []
        for i in range(3):
            r = alloc[JobIdPair(i)][WT] * 2.0
            for j in alloc:
                if j.is_pair() and j.overlaps_with(JobIdPair(i)):
                    r += alloc[j][WT] * 1.4
            rates.append(r)
        # symmetric jobs -> symmetric packed rates
        assert max(rates) == pytest.approx(min(rates), abs=0.05)

    def test_mst_packed_slos(self):
        from shockwave_amd.policies import (
            ThroughputNormalizedByCostSumWithPackingSLOs,
        )

        p = ThroughputNormalizedByCostSumWithPackingSLOs()
        alloc = p.get_allocation(mk_packed(2), mk_sf(2), {WT: 1})
        # sum objective picks the pair (2.8 total > 2.0 single)
        assert alloc[JobIdPair(0, 1)][WT] == pytest.approx(1.0, abs=1e-3)
        # an SLO forcing job 0 to run alone beats the pair
        alloc = p.get_allocation(
            mk_packed(2), mk_sf(2), {WT: 1},
            SLOs={JobIdPair(0): 100.0},
            num_steps_remaining={JobIdPair(0): 190.0, JobIdPair(1): 190.0},
        )
        # job 0 needs rate 1.9 > 1.4 packed: solo time must cover the gap
        rate0 = (
            alloc[JobIdPair(0)][WT] * 2.0 + alloc[JobIdPair(0, 1)][WT] * 1.4
        )
        assert rate0 >= 1.9 - 1e-6

    def test_water_filling_packed(self):
        from shockwave_amd.policies import (
            MaxMinFairnessWaterFillingPolicyWithPacking,
        )

        alloc = MaxMinFairnessWaterFillingPolicyWithPacking().get_allocation(
            mk_packed(3), mk_sf(3), mk_prio(3), {WT: 1}
        )
        pair_time = sum(alloc[j][WT] for j in alloc if j.is_pair())
        assert pair_time == pytest.approx(1.0, abs=1e-2)

    def test_water_filling_base_is_las(self):
        """Base water filling uses unit throughputs (reference
        max_min_fairness_water_filling.py:443-447): a slow job gets the
        same TIME share as a fast one."""
        alloc = MaxMinFairnessWaterFillingPolicy().get_allocation(
            mk_tputs([10.0, 1.0]), mk_sf(2), mk_prio(2), {WT: 1}
        )
        assert alloc[JobIdPair(0)][WT] == pytest.approx(
            alloc[JobIdPair(1)][WT], abs=1e-3
        )

    def test_factory_packed_names(self):
        for name in (
            "max_min_fairness_packed",
            "finish_time_fairness_packed",
            "min_total_duration_packed",
            "max_sum_throughput_normalized_by_cost_packed_SLOs",
            "max_min_fairness_water_filling_perf",
            "max_min_fairness_water_filling_packed",
        ):
            p = get_policy(name)
            assert "Packing" in p.name or "Perf" in p.name

    def test_allox_alpha_parsing(self):
        assert get_policy("allox_alpha=0.7")._alpha == pytest.approx(0.7)
        assert get_policy("allox")._alpha == pytest.approx(0.2)


class TestIsolatedPlus:
    def test_no_scale_factor_division(self):
        """isolated_plus gives a 4-GPU job the same TIME share as a
        1-GPU job (isolated divides it by the scale factor)."""
        from shockwave_amd.policies import (
            IsolatedPolicy, IsolatedPlusPolicy,
        )

        tputs = mk_tputs([1.0, 1.0])
        sf = {JobIdPair(0): 4, JobIdPair(1): 1}
        plus = IsolatedPlusPolicy().get_allocation(tputs, sf, {WT: 2})
        base = IsolatedPolicy().get_allocation(tputs, sf, {WT: 2})
        assert plus[JobIdPair(0)][WT] == pytest.approx(
            plus[JobIdPair(1)][WT]
        )
        assert base[JobIdPair(0)][WT] < base[JobIdPair(1)][WT]
