"""Min-max-sum split of an array into k contiguous subarrays.

Reference: scheduler/shockwave_helper.py:1-74 (binary search over the
feasible max-sum).  Returns (best_max_sum, boundaries) where boundaries are
the k subarray slices.
"""

from __future__ import annotations

from typing import List, Sequence, Tuple


def min_max_sum_k_subarrays(arr: Sequence[float], k: int) -> Tuple[float, List[List[float]]]:
    assert k >= 1 and len(arr) >= k
    lo, hi = max(arr), sum(arr)

    def pieces_needed(cap: float) -> int:
        count, acc = 1, 0.0
        for v in arr:
            if acc + v > cap:
                count += 1
                acc = v
            else:
                acc += v
        return count

    while hi - lo > 1e-9 * max(1.0, hi):
        mid = 0.5 * (lo + hi)
        if pieces_needed(mid) <= k:
            hi = mid
        else:
            lo = mid

    # materialize the split at capacity hi
    out, acc = [], []
    acc_sum = 0.0
    for v in arr:
        if acc and acc_sum + v > hi:
            out.append(acc)
            acc, acc_sum = [v], v
        else:
            acc.append(v)
            acc_sum += v
    out.append(acc)
    while len(out) < k:  # pad with empty splits if fewer pieces used
        out.append([])
    return hi, out


class MinMaxSumKSubarrays:
    """Class-shaped API mirroring the reference helper."""

    def __init__(self, arr: Sequence[float], k: int):
        self.arr = list(arr)
        self.k = k

    def solve(self):
        return min_max_sum_k_subarrays(self.arr, self.k)
