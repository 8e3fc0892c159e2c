#!/usr/bin/env python3
"""Policy sweep harness: grid of (policy x load x seed) simulations.

MI355X rebuild of the reference's scripts/sweeps/run_sweep_continuous.py
:146-330 and run_sweep_static.py: sweep a grid of input throughputs
(jobs/hr -> lambda = 3600/throughput exponential interarrivals), policies
and seeds, run each simulation in a worker process with a per-experiment
timeout, and record average JCT + utilization over a measurement window
of job ids (continuous mode) or makespan over a fixed batch (static
mode).

Continuous mode generates window_end + margin jobs and stops each
simulation as soon as the window jobs [window_start, window_end) have
completed (engine ``jobs_to_complete``).

Results land in <log_dir>/sweep_results.jsonl, one JSON object per
experiment, plus a stdout table.
"""

import argparse
import copy
import json
import multiprocessing
import os
import signal
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from shockwave_amd.core import generator, trace as trace_mod
from shockwave_amd.core.job import JobIdPair
from shockwave_amd.core.throughputs import read_throughputs
from shockwave_amd.engine import RoundScheduler
from shockwave_amd.policies import get_policy


class _Timeout(Exception):
    pass


def _alarm(signum, frame):
    raise _Timeout()


def run_experiment(spec):
    """One simulation; returns a result dict (inf metrics on timeout)."""
    import logging

    logging.disable(logging.WARNING)
    signal.signal(signal.SIGALRM, _alarm)
    if spec["timeout"]:
        signal.alarm(int(spec["timeout"]))

    tputs = read_throughputs(spec["throughputs_file"])
    worker_type = trace_mod.canonical_worker_type(tputs)
    num_jobs = spec["num_jobs"]
    jobs, arrivals = generator.generate_trace(
        tputs,
        worker_type,
        num_jobs,
        lam_s=spec["lam"],
        min_duration_s=spec["min_duration_s"],
        max_duration_s=spec["max_duration_s"],
        seed=spec["seed"],
    )
    if spec["mode"] == "static":
        arrivals = [0.0] * len(arrivals)
    profiles = [trace_mod.build_job_profile(j, tputs) for j in jobs]
    for j, p in zip(jobs, profiles):
        j.duration = sum(p["duration_every_epoch"])

    shockwave_config = None
    if spec["policy"] == "shockwave":
        shockwave_config = {
            "future_rounds": 8, "k": 1e-3, "lambda": 12.0, "rhomax": 1.0,
            "time_per_iteration": spec["interval"],
            "num_gpus": spec["num_gpus"],
        }
    sched = RoundScheduler(
        get_policy(spec["policy"], seed=spec["seed"]),
        simulate=True,
        throughputs=tputs,
        seed=spec["seed"],
        time_per_iteration=spec["interval"],
        profiles=profiles,
        shockwave_config=shockwave_config,
        worker_type=worker_type,
    )
    window = None
    if spec["mode"] == "continuous":
        window = {
            JobIdPair(i)
            for i in range(spec["window_start"], spec["window_end"])
        }
    t0 = time.time()
    result = dict(spec)
    result.pop("timeout", None)
    try:
        makespan = sched.simulate(
            {worker_type: spec["num_gpus"]},
            list(arrivals),
            copy.deepcopy(jobs),
            jobs_to_complete=window,
            ideal=spec.get("ideal", False),
        )
        window_ids = sorted(window) if window else None
        avg_jct, geo_jct, _, _ = sched.get_average_jct(window_ids)
        util, _ = sched.get_cluster_utilization()
        ftf, _ = sched.get_finish_time_fairness()
        result.update(
            status="ok",
            makespan_s=makespan,
            avg_jct_s=avg_jct,
            geo_jct_s=geo_jct,
            cluster_util=util,
            worst_ftf_rho=max(ftf) if ftf else None,
            wall_s=round(time.time() - t0, 1),
        )
    except _Timeout:
        result.update(status="timeout", avg_jct_s=float("inf"),
                      makespan_s=float("inf"), cluster_util=None,
                      wall_s=round(time.time() - t0, 1))
    except Exception as e:  # report, don't kill the pool
        result.update(status=f"error: {e!r}"[:300], avg_jct_s=float("inf"),
                      makespan_s=float("inf"), cluster_util=None,
                      wall_s=round(time.time() - t0, 1))
    finally:
        signal.alarm(0)
    return result


def build_grid(args):
    specs = []
    if args.mode == "continuous":
        tputs_grid = [
            args.throughput_lower_bound
            + i
            * (args.throughput_upper_bound - args.throughput_lower_bound)
            / max(1, args.num_data_points - 1)
            for i in range(args.num_data_points)
        ]
        lams = [3600.0 / t for t in tputs_grid if t > 0]
    else:
        lams = [args.lam]
    for policy in args.policies:
        for lam in lams:
            for seed in args.seeds:
                specs.append(
                    {
                        "mode": args.mode,
                        "policy": policy,
                        "lam": lam,
                        "jobs_per_hr": round(3600.0 / lam, 3),
                        "seed": seed,
                        "num_gpus": args.cluster_spec,
                        "interval": args.interval,
                        "num_jobs": (
                            args.window_end + args.margin_jobs
                            if args.mode == "continuous"
                            else args.num_jobs
                        ),
                        "window_start": args.window_start,
                        "window_end": args.window_end,
                        "min_duration_s": args.min_duration,
                        "max_duration_s": args.max_duration,
                        "throughputs_file": args.throughputs_file,
                        "ideal": args.ideal,
                        "timeout": args.timeout,
                    }
                )
    # run high-load (small lambda) experiments last: sort by decreasing
    # lambda so quick ones drain first (reference run_sweep_continuous
    # sorts the same way)
    specs.sort(key=lambda s: -s["lam"])
    return specs


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--mode", choices=("static", "continuous"),
                   default="continuous")
    p.add_argument("--policies", nargs="+", required=True)
    p.add_argument("--seeds", nargs="+", type=int, default=[0])
    p.add_argument("--throughputs_file",
                   default="traces/mi355x_throughputs.json")
    p.add_argument("-c", "--cluster_spec", type=int, default=32,
                   help="number of GPUs (single worker type)")
    p.add_argument("--interval", type=int, default=120,
                   help="round duration (s)")
    p.add_argument("-a", "--throughput_lower_bound", type=float, default=1.0,
                   help="continuous: min input load, jobs/hr")
    p.add_argument("-b", "--throughput_upper_bound", type=float, default=6.0,
                   help="continuous: max input load, jobs/hr")
    p.add_argument("-n", "--num_data_points", type=int, default=3)
    p.add_argument("-s", "--window_start", type=int, default=20)
    p.add_argument("-e", "--window_end", type=int, default=60)
    p.add_argument("--margin_jobs", type=int, default=20,
                   help="extra jobs generated past the window to keep the "
                        "cluster loaded while window jobs finish")
    p.add_argument("--lam", type=float, default=60.0,
                   help="static: mean interarrival (s)")
    p.add_argument("--num_jobs", type=int, default=60, help="static mode")
    p.add_argument("--min_duration", type=float, default=300.0)
    p.add_argument("--max_duration", type=float, default=8000.0)
    p.add_argument("-p", "--processes", type=int,
                   default=max(1, (os.cpu_count() or 2) // 2))
    p.add_argument("--ideal", action="store_true",
                   help="round-free fractional-allocation upper bound")
    p.add_argument("--timeout", type=float, default=1200.0,
                   help="per-experiment timeout (s)")
    p.add_argument("-l", "--log_dir", default="results/sweep")
    args = p.parse_args()

    specs = build_grid(args)
    os.makedirs(args.log_dir, exist_ok=True)
    out_path = os.path.join(args.log_dir, "sweep_results.jsonl")
    print(f"running {len(specs)} experiment(s) on {args.processes} "
          f"process(es) -> {out_path}")
    t0 = time.time()
    with multiprocessing.Pool(args.processes) as pool, \
            open(out_path, "w") as out:
        for r in pool.imap_unordered(run_experiment, specs):
            out.write(json.dumps(r) + "\n")
            out.flush()
            print(
                f"[{time.time() - t0:7.1f}s] {r['policy']:<34} "
                f"load={r['jobs_per_hr']:<7} seed={r['seed']} "
                f"{r['status']}: avg JCT "
                f"{r['avg_jct_s'] / 3600.0:.3f} h, util "
                f"{r['cluster_util']}"
            )
    print(f"done in {time.time() - t0:.0f}s")


if __name__ == "__main__":
    main()
