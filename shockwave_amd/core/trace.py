"""Trace file parsing/writing and epoch-profile generation.

Trace format (12 tab-separated fields per line, reference
utils.py:1446-1498):

    job_type  command  working_directory  num_steps_arg  needs_data_dir
    total_steps  scale_factor  mode  priority_weight  SLO  duration
    arrival_time

``generate_profiles`` converts a trace + throughput oracle into the per-job
epoch-level profile list the Shockwave planner consumes (reference
``generate_pickle_file`` utils.py:1331-1444): for each job, the oracle
batch-size schedule per epoch, per-epoch memory/utilization from profiling
tables, and per-epoch duration = (dataset_len / bs) / isolated_throughput.
"""

from __future__ import annotations

import math
import os
import pickle
from typing import Dict, List, Tuple

from . import bs_patterns, datasets
from .job import Job
from .throughputs import isolated_throughput


def parse_trace(trace_file: str) -> Tuple[List[Job], List[float]]:
    jobs, arrival_times = [], []
    with open(trace_file) as f:
        for line in f:
            if not line.strip():
                continue
            (
                job_type,
                command,
                working_directory,
                num_steps_arg,
                needs_data_dir,
                total_steps,
                scale_factor,
                mode,
                priority_weight,
                slo,
                duration,
                arrival_time,
            ) = line.rstrip("\n").split("\t")
            assert int(scale_factor) >= 1
            jobs.append(
                Job(
                    job_id=None,
                    job_type=job_type,
                    command=command,
                    working_directory=working_directory,
                    needs_data_dir=bool(int(needs_data_dir)),
                    num_steps_arg=num_steps_arg,
                    total_steps=int(total_steps),
                    duration=duration,
                    scale_factor=int(scale_factor),
                    mode=mode,
                    priority_weight=float(priority_weight),
                    SLO=float(slo),
                )
            )
            arrival_times.append(float(arrival_time))
    return jobs, arrival_times


def write_trace(jobs: List[Job], arrival_times: List[float], path: str) -> None:
    with open(path, "w") as f:
        for job, at in zip(jobs, arrival_times):
            f.write("%s\t%d\n" % (str(job), at))


def job_num_epochs(job: Job) -> int:
    spe = datasets.steps_per_epoch(job.model, job.batch_size)
    return math.ceil(job.total_steps / spe)


def canonical_worker_type(parsed_throughputs: Dict) -> str:
    """The single homogeneous accelerator type the profiles are quoted on.

    The reference hard-codes "v100" (scheduler.py:1128, 2300); we prefer
    "mi355x" when an MI355X-profiled oracle is loaded, then "v100", then
    whatever single type the file carries."""
    for wt in ("mi355x", "v100"):
        if wt in parsed_throughputs:
            return wt
    return next(iter(parsed_throughputs))


def build_job_profile(
    job: Job, parsed_throughputs: Dict, job_index: int = 0, worker_type: str = None
) -> Dict:
    model = job.model
    dataset = datasets.dataset_for_model(model)
    batch_size = job.batch_size
    num_epochs = job_num_epochs(job)
    if worker_type is None:
        worker_type = canonical_worker_type(parsed_throughputs)
    bs_every_epoch = bs_patterns.bs_pattern_for_mode(
        job.mode, job.job_type, batch_size, num_epochs, job.scale_factor
    )

    def epoch_duration(bs: int) -> float:
        job_type = f"{model} (batch size {bs})"
        tput = isolated_throughput(
            parsed_throughputs, worker_type, job_type, int(job.scale_factor)
        )
        iters = datasets.dataset_len(dataset) / bs
        return iters / tput

    return {
        "model": model,
        "dataset": dataset,
        "num_epochs": num_epochs,
        "num_samples_per_epoch": datasets.dataset_len(dataset),
        "bs_every_epoch": bs_every_epoch,
        "mem_every_epoch": [datasets.mem_mb(model, bs) for bs in bs_every_epoch],
        "util_every_epoch": [datasets.util_pct(model, bs) for bs in bs_every_epoch],
        "duration_every_epoch": [epoch_duration(bs) for bs in bs_every_epoch],
        "scale_factor": job.scale_factor,
        "duration": job.duration,
    }


def generate_profiles(
    trace_file: str, parsed_throughputs: Dict, pickle_path: str = None
) -> Tuple[List[Job], List[float], List[Dict]]:
    """Build the per-job profile list; optionally persist next to the trace
    (reference writes ``<trace>.pickle`` beside the trace file)."""
    jobs, arrival_times = parse_trace(trace_file)
    profiles = [
        build_job_profile(job, parsed_throughputs, i) for i, job in enumerate(jobs)
    ]
    if pickle_path is None:
        pickle_path = os.path.splitext(trace_file)[0] + ".pickle"
    with open(pickle_path, "wb") as f:
        pickle.dump(profiles, f)
    return jobs, arrival_times, profiles
