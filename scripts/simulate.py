#!/usr/bin/env python3
"""Simulation driver: run a trace through the round scheduler.

MI355X rebuild of the reference's
scripts/drivers/simulate_scheduler_with_trace.py:21-260: builds the
epoch-profile pickle, constructs the scheduler with the chosen policy,
simulates, and dumps a results pickle.
"""

import argparse
import json
import os
import pickle
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from shockwave_amd.core import trace as trace_mod
from shockwave_amd.core.throughputs import read_throughputs
from shockwave_amd.engine import RoundScheduler
from shockwave_amd.policies import get_policy


def run_simulation(
    trace_file,
    throughputs_file,
    policy_name,
    num_gpus=32,
    time_per_iteration=120,
    seed=0,
    shockwave_config_file=None,
    results_dir=None,
    log_level="WARNING",
    preemption_overhead_s=20.0,
    warm_overhead_s=None,
    midround_staleness=False,
    fixed_rounds=False,
    startup_table=None,
    ideal=False,
    world_throughputs_file=None,
):
    import logging

    logging.basicConfig(level=getattr(logging, log_level))
    if results_dir:
        # mirror console output per policy (reference scheduler.py:126-133)
        os.makedirs(results_dir, exist_ok=True)
        fh = logging.FileHandler(
            os.path.join(results_dir, f"console_output_{policy_name}.txt")
        )
        fh.setLevel(getattr(logging, log_level))
        logging.getLogger().addHandler(fh)

    throughputs = read_throughputs(throughputs_file)
    worker_type = trace_mod.canonical_worker_type(throughputs)
    jobs, arrival_times, profiles = trace_mod.generate_profiles(
        trace_file, throughputs
    )
    # use post-adaptation duration as each job's reference duration
    for job, prof in zip(jobs, profiles):
        job.duration = sum(prof["duration_every_epoch"])

    policy = get_policy(policy_name, seed=seed)

    shockwave_config = None
    if policy_name == "shockwave":
        if shockwave_config_file:
            shockwave_config = json.load(open(shockwave_config_file))
        else:
            shockwave_config = {}
        shockwave_config.setdefault("future_rounds", 20)
        shockwave_config.setdefault("k", 1e-3)
        shockwave_config.setdefault("lambda", 12.0)
        shockwave_config.setdefault("rhomax", 1.0)
        shockwave_config["time_per_iteration"] = time_per_iteration
        shockwave_config["num_gpus"] = num_gpus

    sched = RoundScheduler(
        policy,
        simulate=True,
        throughputs=throughputs,
        seed=seed,
        time_per_iteration=time_per_iteration,
        profiles=profiles,
        shockwave_config=shockwave_config,
        worker_type=worker_type,
        preemption_overhead_s=preemption_overhead_s,
        warm_preemption_overhead_s=warm_overhead_s,
        midround_staleness=midround_staleness,
        fixed_rounds=fixed_rounds,
        startup_table=startup_table,
        world_throughputs=(read_throughputs(world_throughputs_file)
                           if world_throughputs_file else None),
    )

    start = time.time()
    makespan = sched.simulate({worker_type: num_gpus}, arrival_times, jobs,
                              ideal=ideal)
    wall = time.time() - start

    avg_jct, geo_jct, har_jct, jct_list = sched.get_average_jct()
    ftf_static, ftf_themis = sched.get_finish_time_fairness()
    util, util_list = sched.get_cluster_utilization()
    ext_pct, next_, nopp = sched.get_num_lease_extensions()

    unfair_pct = (
        100.0 * sum(1 for r in ftf_static if r > 1.05) / len(ftf_static)
        if ftf_static
        else 0.0
    )
    results = {
        "policy": policy_name,
        "trace": os.path.basename(trace_file),
        "num_gpus": num_gpus,
        "round_duration": time_per_iteration,
        "seed": seed,
        "makespan_s": makespan,
        "makespan_h": makespan / 3600.0,
        "avg_jct_s": avg_jct,
        "avg_jct_h": avg_jct / 3600.0,
        "geo_jct_s": geo_jct,
        "harmonic_jct_s": har_jct,
        "jct_list": jct_list,
        "ftf_rho_list": ftf_static,
        "ftf_rho_themis_list": ftf_themis,
        "worst_ftf_rho": max(ftf_static) if ftf_static else None,
        "unfair_fraction_pct": unfair_pct,
        "cluster_util": util,
        "utilization_list": util_list,
        "lease_extension_pct": ext_pct,
        "per_round_schedule": sched.get_per_round_schedule(),
        "sim_wall_time_s": wall,
    }
    if results_dir:
        os.makedirs(results_dir, exist_ok=True)
        name = os.path.splitext(os.path.basename(trace_file))[0]
        out = os.path.join(results_dir, f"{policy_name}_{name}_simulation.pickle")
        with open(out, "wb") as f:
            pickle.dump(results, f)
        print(f"results -> {out}")
    return results


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("-t", "--trace_file", default=None)
    p.add_argument("--ideal", action="store_true",
                   help="round-free fractional-allocation upper bound "
                        "(reference simulate ideal=True)")
    p.add_argument("--generate_jobs", type=int, default=None,
                   help="generate N jobs on the fly instead of a trace "
                        "(reference simulate_scheduler_with_generated_jobs)")
    p.add_argument("--lam", type=float, default=60.0,
                   help="mean interarrival seconds for --generate_jobs")
    p.add_argument("--throughputs_file", default="traces/mi355x_throughputs.json")
    p.add_argument("-p", "--policy", default="shockwave")
    p.add_argument("-c", "--config", default=None, help="shockwave config json")
    p.add_argument("-n", "--num_gpus", type=int, default=32)
    p.add_argument("--time_per_iteration", type=int, default=120)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--results_dir", default="results")
    p.add_argument("--log_level", default="WARNING")
    p.add_argument("--preemption_overhead", type=float, default=20.0,
                   help="simulated checkpoint/restart cost per migration (s)")
    p.add_argument("--midround_staleness", action="store_true",
                   help="model the physical planner's mid-round decision "
                        "point (half of the current round's service "
                        "unobserved) for non-shockwave policies")
    p.add_argument("--world_throughputs_file", default=None,
                   help="fidelity: rates the simulated WORLD runs at, "
                        "while --throughputs_file stays the rates the "
                        "policy believes (the physical scheduler's own "
                        "oracle)")
    p.add_argument("--startup_table", default=None,
                   help="JSON of per-job-type first-dispatch startup "
                        "seconds (scripts/calibrate_sim.py)")
    p.add_argument("--fixed_rounds", action="store_true",
                   help="wall-clock round boundaries (round-tail idle "
                        "after early completions), as the physical "
                        "mechanism")
    p.add_argument("--warm_overhead", type=float, default=None,
                   help="migration cost once the job type has run before "
                        "(MIOpen find-db / warm-runner session hot); "
                        "default: flat model")
    args = p.parse_args()

    if args.generate_jobs:
        import tempfile

        from shockwave_amd.core import generator
        from shockwave_amd.core.throughputs import read_throughputs as _rt
        from shockwave_amd.core.trace import (
            canonical_worker_type as _cwt,
            write_trace as _wt,
        )

        tputs = _rt(args.throughputs_file)
        jobs, arrivals = generator.generate_trace(
            tputs, _cwt(tputs), args.generate_jobs, lam_s=args.lam,
            seed=args.seed,
        )
        fd, args.trace_file = tempfile.mkstemp(suffix=".trace")
        os.close(fd)
        _wt(jobs, arrivals, args.trace_file)
        print(f"generated {len(jobs)} jobs -> {args.trace_file}")
    assert args.trace_file, "need --trace_file or --generate_jobs"

    r = run_simulation(
        args.trace_file,
        args.throughputs_file,
        args.policy,
        num_gpus=args.num_gpus,
        time_per_iteration=args.time_per_iteration,
        seed=args.seed,
        shockwave_config_file=args.config,
        results_dir=args.results_dir,
        log_level=args.log_level,
        preemption_overhead_s=args.preemption_overhead,
        warm_overhead_s=args.warm_overhead,
        midround_staleness=args.midround_staleness,
        fixed_rounds=args.fixed_rounds,
        startup_table=(json.load(open(args.startup_table))
                       if args.startup_table else None),
        ideal=args.ideal,
        world_throughputs_file=args.world_throughputs_file,
    )
    print(
        json.dumps(
            {
                k: r[k]
                for k in (
                    "policy",
                    "makespan_h",
                    "avg_jct_h",
                    "worst_ftf_rho",
                    "unfair_fraction_pct",
                    "cluster_util",
                    "sim_wall_time_s",
                )
            },
            indent=1,
        )
    )


if __name__ == "__main__":
    main()
