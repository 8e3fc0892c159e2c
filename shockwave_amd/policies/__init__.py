"""Policy registry (reference: utils.get_policy, utils.py:603-686)."""

from .base import Policy, PolicyWithPacking
from .simple import (
    IsolatedPolicy,
    IsolatedPlusPolicy,
    ProportionalPolicy,
    GandivaFairPolicy,
)
from .fifo import FIFOPolicy, FIFOPolicyWithPerf, FIFOPolicyWithPacking
from .max_min_fairness import (
    MaxMinFairnessPolicy,
    MaxMinFairnessPolicyWithPerf,
    MaxMinFairnessPolicyWithPacking,
    MaxMinFairnessWaterFillingPolicy,
    MaxMinFairnessWaterFillingPolicyWithPerf,
    MaxMinFairnessWaterFillingPolicyWithPacking,
    MaxMinFairnessStrategyProofPolicy,
    MaxMinFairnessStrategyProofPolicyWithPerf,
)
from .finish_time_fairness import (
    FinishTimeFairnessPolicy,
    FinishTimeFairnessPolicyWithPerf,
    FinishTimeFairnessPolicyWithPacking,
)
from .max_sum_throughput import (
    ThroughputSumWithPerf,
    ThroughputNormalizedByCostSumWithPerf,
    ThroughputNormalizedByCostSumWithPerfSLOs,
    ThroughputNormalizedByCostSumWithPackingSLOs,
)
from .min_total_duration import (
    MinTotalDurationPolicy,
    MinTotalDurationPolicyWithPerf,
    MinTotalDurationPolicyWithPacking,
)
from .allox import AlloXPolicy
from .gandiva import GandivaPolicy


class ShockwavePolicyStub(Policy):
    """Name marker only: the Shockwave planner drives the round schedule
    directly (engine + solver/planner.py), matching the reference where
    policies/shockwave.py is a 10-line stub."""

    name = "Shockwave"


def get_policy(policy_name: str, seed=None, solver=None,
               priority_reweighting_policies=None):
    # reference accepts "allox_alpha=<float>" spellings (utils.py:606-611)
    if policy_name.startswith("allox"):
        alpha = (
            float(policy_name.split("allox_alpha=")[1])
            if "alpha=" in policy_name
            else 0.2
        )
        return AlloXPolicy(alpha=alpha)
    table = {
        "fifo": lambda: FIFOPolicy(seed=seed),
        "fifo_perf": FIFOPolicyWithPerf,
        "fifo_packed": FIFOPolicyWithPacking,
        "finish_time_fairness": FinishTimeFairnessPolicy,
        "finish_time_fairness_perf": FinishTimeFairnessPolicyWithPerf,
        "finish_time_fairness_packed": FinishTimeFairnessPolicyWithPacking,
        "gandiva": lambda: GandivaPolicy(seed=seed),
        "gandiva_fair": GandivaFairPolicy,
        "isolated": IsolatedPolicy,
        "isolated_plus": IsolatedPlusPolicy,
        "max_min_fairness": MaxMinFairnessPolicy,
        "max_min_fairness_perf": MaxMinFairnessPolicyWithPerf,
        "max_min_fairness_packed": MaxMinFairnessPolicyWithPacking,
        "max_min_fairness_water_filling": lambda: MaxMinFairnessWaterFillingPolicy(
            priority_reweighting_policies=priority_reweighting_policies
        ),
        "max_min_fairness_water_filling_perf": lambda: MaxMinFairnessWaterFillingPolicyWithPerf(
            priority_reweighting_policies=priority_reweighting_policies
        ),
        "max_min_fairness_water_filling_packed": lambda: MaxMinFairnessWaterFillingPolicyWithPacking(
            priority_reweighting_policies=priority_reweighting_policies
        ),
        "max_min_fairness_strategy_proof": MaxMinFairnessStrategyProofPolicy,
        "max_min_fairness_strategy_proof_perf": MaxMinFairnessStrategyProofPolicyWithPerf,
        "max_sum_throughput_perf": ThroughputSumWithPerf,
        "max_sum_throughput_normalized_by_cost_perf": ThroughputNormalizedByCostSumWithPerf,
        "max_sum_throughput_normalized_by_cost_perf_SLOs": ThroughputNormalizedByCostSumWithPerfSLOs,
        "max_sum_throughput_normalized_by_cost_packed_SLOs": ThroughputNormalizedByCostSumWithPackingSLOs,
        "min_total_duration": MinTotalDurationPolicy,
        "min_total_duration_perf": MinTotalDurationPolicyWithPerf,
        "min_total_duration_packed": MinTotalDurationPolicyWithPacking,
        "shockwave": ShockwavePolicyStub,
    }
    if policy_name not in table:
        raise ValueError(f"unknown policy {policy_name!r}")
    return table[policy_name]()
