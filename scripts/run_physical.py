#!/usr/bin/env python3
"""Physical-cluster driver: run a trace against live workers.

Rebuild of the reference's scripts/drivers/run_scheduler_with_trace.py:
starts the head process (PhysicalScheduler), paces job submission by the
trace's arrival times, waits for completion, dumps results.

Workers are started separately (``python -m shockwave_amd.runtime.worker``)
or in-process with --inprocess_workers N (single-node convenience used by
tests and single-box benchmarks).
"""

import argparse
import json
import os
import pickle
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from shockwave_amd.core import trace as trace_mod
from shockwave_amd.core.throughputs import read_throughputs
from shockwave_amd.engine.physical import PhysicalScheduler
from shockwave_amd.policies import get_policy


def launch_inprocess_worker(args, num_gpus, worker_port):
    from shockwave_amd.runtime.worker import Worker

    repo = os.path.join(os.path.dirname(__file__), "..")
    worker = Worker(
        worker_type=args.worker_type,
        sched_addr="127.0.0.1",
        sched_port=args.port,
        worker_port=worker_port,
        num_gpus=num_gpus,
        ip_addr="127.0.0.1",
        run_dir=os.path.join(repo, "workloads", "pytorch"),
        static_run_dir=os.path.join(repo, "workloads", "pytorch"),
        accordion_run_dir=os.path.join(repo, "workloads", "accordion"),
        gns_run_dir=os.path.join(repo, "workloads", "gns"),
        data_dir=args.data_dir,
        checkpoint_dir=args.checkpoint_dir,
    )
    return worker


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("-t", "--trace_file", required=True)
    p.add_argument("--throughputs_file", default="traces/mi355x_throughputs.json")
    p.add_argument("-p", "--policy", default="shockwave")
    p.add_argument("-c", "--config", default=None)
    p.add_argument("--port", type=int, default=50070)
    p.add_argument("--time_per_iteration", type=int, default=120)
    p.add_argument("--expected_num_workers", type=int, default=None)
    p.add_argument("--inprocess_workers", type=int, default=0,
                   help="start N local GPU workers in this process")
    p.add_argument("--worker_type", default="mi355x")
    p.add_argument("--data_dir", default=None)
    p.add_argument("--checkpoint_dir", default="/tmp/swq_checkpoints")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--timeout", type=float, default=None)
    p.add_argument("--arrival_time_scale", type=float, default=1.0)
    p.add_argument("--results_file", default=None)
    p.add_argument("--max_rounds", type=int, default=None)
    p.add_argument("--log_level", default="INFO")
    args = p.parse_args(argv)

    import logging

    logging.basicConfig(level=getattr(logging, args.log_level))

    throughputs = read_throughputs(args.throughputs_file)
    jobs, arrival_times, profiles = trace_mod.generate_profiles(
        args.trace_file, throughputs
    )
    for job, prof in zip(jobs, profiles):
        job.duration = sum(prof["duration_every_epoch"])

    policy = get_policy(args.policy, seed=args.seed)
    shockwave_config = None
    if args.policy == "shockwave":
        shockwave_config = json.load(open(args.config)) if args.config else {}
        shockwave_config.setdefault("future_rounds", 10)
        shockwave_config["time_per_iteration"] = args.time_per_iteration
        shockwave_config.setdefault(
            "num_gpus",
            args.expected_num_workers or args.inprocess_workers or 1,
        )

    sched = PhysicalScheduler(
        policy,
        port=args.port,
        expected_num_workers=(
            args.expected_num_workers or args.inprocess_workers or None
        ),
        throughputs=throughputs,
        seed=args.seed,
        time_per_iteration=args.time_per_iteration,
        profiles=profiles,
        shockwave_config=shockwave_config,
        worker_type=args.worker_type,
        max_rounds=args.max_rounds,
    )

    workers = []
    if args.inprocess_workers:
        workers.append(
            launch_inprocess_worker(args, args.inprocess_workers, 50061)
        )

    # paced submission (reference run_scheduler_with_trace.py:38-120)
    job_types = {}

    def submit():
        start = time.time()
        for job, at in zip(jobs, arrival_times):
            delay = at * args.arrival_time_scale - (time.time() - start)
            if delay > 0:
                time.sleep(delay)
            jid = sched.add_job(job)
            job_types[str(jid[0] if hasattr(jid, "__getitem__") else jid)] = (
                job.job_type
            )

    submitter = threading.Thread(target=submit, daemon=True)
    submitter.start()

    start = time.time()
    deadline = start + args.timeout if args.timeout else None
    while not sched.is_done():
        time.sleep(2)
        if deadline and time.time() > deadline:
            print("TIMEOUT: shutting down", flush=True)
            break
    makespan = time.time() - start

    jct = sched.get_average_jct()
    ftf = sched.get_finish_time_fairness()
    util = sched.get_cluster_utilization()
    results = {
        "policy": args.policy,
        "makespan_s": makespan,
        "avg_jct_s": jct[0] if jct else None,
        "jct_list": jct[3] if jct else [],
        "ftf_rho_list": ftf[0] if ftf else [],
        "worst_ftf_rho": max(ftf[0]) if ftf and ftf[0] else None,
        "cluster_util": util[0],
        "num_completed": len(sched.get_job_completion_times()),
        # raw per-job evidence for scripts/analyze_jobs.py: iterator
        # event logs per (job, worker slot), per-round schedules, and the
        # per-round (throughput, bs) timeline
        "job_timelines": {
            str(jid): logs for jid, logs in sched._job_timelines.items()
        },
        "per_round_schedule": [
            {str(j): list(w) for j, w in rnd.items()}
            for rnd in sched._per_round_schedule
        ],
        "throughput_timeline": {
            int(j): dict(t) for j, t in sched._throughput_timeline.items()
        },
        "job_types": job_types,
        "job_total_steps": {
            str(jid): job.total_steps for jid, job in sched._jobs.items()
        },
        "completion_times": {
            str(j): t for j, t in sched.get_job_completion_times().items()
        },
    }
    sched.shutdown()
    print(json.dumps(results))
    if args.results_file:
        with open(args.results_file, "wb") as f:
            pickle.dump(results, f)
    return results


if __name__ == "__main__":
    main()
