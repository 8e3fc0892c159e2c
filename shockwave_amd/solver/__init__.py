from .planner import ShockwavePlanner, finish_time_momentumed_average
from .eg import PlannerJob, solve_eg_milp, solve_rank_milp
from .subarrays import MinMaxSumKSubarrays, min_max_sum_k_subarrays

__all__ = [
    "ShockwavePlanner",
    "finish_time_momentumed_average",
    "PlannerJob",
    "solve_eg_milp",
    "solve_rank_milp",
    "MinMaxSumKSubarrays",
    "min_max_sum_k_subarrays",
]
