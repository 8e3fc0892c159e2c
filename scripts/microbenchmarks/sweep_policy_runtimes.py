#!/usr/bin/env python3
"""Policy/solver latency vs job count (scalability regression).

Reference: scripts/microbenchmarks/sweep_policy_runtimes.py — measures how
long each policy's allocation solve takes as the active job count grows,
plus the Shockwave EG MILP at planning scale.
"""

import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from shockwave_amd.core.job import JobIdPair
from shockwave_amd.policies import get_policy
from shockwave_amd.solver import PlannerJob, solve_eg_milp

WT = "mi355x"


def bench_policy(name, njobs, seed=0):
    rng = random.Random(seed)
    tputs = {
        JobIdPair(i): {WT: rng.uniform(1, 100)} for i in range(njobs)
    }
    sf = {JobIdPair(i): rng.choice([1] * 7 + [2, 2, 4]) for i in range(njobs)}
    prio = {JobIdPair(i): 1.0 for i in range(njobs)}
    times = {JobIdPair(i): rng.uniform(0, 1e4) for i in range(njobs)}
    steps = {JobIdPair(i): rng.uniform(1e3, 1e6) for i in range(njobs)}
    cluster = {WT: max(4, njobs // 4)}

    policy = get_policy(name, seed=seed)
    from shockwave_amd.policies import PolicyWithPacking

    if isinstance(policy, PolicyWithPacking):
        # packed LPs range over singles AND same-scale pairs, like the
        # engine's pair registration — include them so the measured LP
        # size is the real one
        singles = list(tputs.keys())
        for a in range(njobs):
            for b in range(a + 1, njobs):
                ja, jb = singles[a], singles[b]
                if sf[ja] != sf[jb]:
                    continue
                t1, t2 = tputs[ja][WT] * 0.7, tputs[jb][WT] * 0.7
                tputs[JobIdPair(ja[0], jb[0])] = {WT: (t1, t2)}
    t0 = time.time()
    if name == "allox":
        policy.get_allocation(tputs, sf, times, steps, [], cluster)
    elif name.startswith("finish_time_fairness"):
        policy.get_allocation(tputs, sf, prio, times, steps, cluster)
    elif name.startswith("min_total_duration"):
        policy.get_allocation(tputs, sf, steps, cluster)
    elif name.startswith("max_min"):
        policy.get_allocation(tputs, sf, prio, cluster)
    else:
        policy.get_allocation(tputs, sf, cluster)
    return time.time() - t0


def bench_shockwave(njobs, ngpus, future_rounds=20, seed=0):
    rng = random.Random(seed)
    jobs = []
    for i in range(njobs):
        E = rng.randint(10, 200)
        prog = rng.randint(0, E - 1)
        d = rng.uniform(20, 400)
        jobs.append(
            PlannerJob(
                i, rng.choice([1] * 7 + [2, 2, 4]), E, prog, d,
                (E - prog) * d, ftf_bound=1e9, priority=1.0,
            )
        )
    t0 = time.time()
    solve_eg_milp(
        jobs, ngpus, 0, future_rounds, 120,
        [0.0, 0.2, 0.4, 0.6, 0.8, 1.0], {0.0: 1e-1}, 1e-3, 1.0,
        enable_ftf=False, timeout=60,
    )
    return time.time() - t0


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--policies", nargs="*", default=[
        "max_min_fairness", "finish_time_fairness", "min_total_duration",
        "max_sum_throughput_perf", "allox", "gandiva_fair",
    ])
    p.add_argument("--job_counts", nargs="*", type=int,
                   default=[16, 64, 128, 256, 512])
    args = p.parse_args()

    print(f"{'policy':28s} " + " ".join(f"{n:>9d}" for n in args.job_counts))
    for name in args.policies:
        times = [bench_policy(name, n) for n in args.job_counts]
        print(f"{name:28s} " + " ".join(f"{t:8.3f}s" for t in times))
    times = [bench_shockwave(n, max(4, n // 4)) for n in args.job_counts]
    print(f"{'shockwave (EG MILP)':28s} " + " ".join(f"{t:8.3f}s" for t in times))


if __name__ == "__main__":
    main()
