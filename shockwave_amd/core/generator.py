"""Synthetic job/trace generation.

Rebuilds the reference's trace generator behavior (utils.generate_job
utils.py:118-276 and scripts/utils/generate_trace.py): jobs are drawn from
the JobTable; scale factors from a Philly-style mix (default
0.6/0.3/0.09/0.01 over 1/2/4/8 GPUs); durations from a 4-bucket power-law
space (72% short, 20% medium, 5% long, 3% longest); modes from a
static/accordion/gns mix; Poisson arrivals; total_steps = duration x
isolated throughput of the (job_type, scale_factor) on the reference worker
type.
"""

from __future__ import annotations

import math
import random
from typing import Dict, List, Optional, Tuple

import numpy as np

from .job import Job
from .job_table import JobTable


def generate_interarrival_time(rng: random.Random, mean_s: float) -> float:
    return -math.log(1.0 - rng.random()) * mean_s


def duration_space(
    min_duration_s: float, max_duration_s: float, nchoices: int, base: float = 1.5
) -> np.ndarray:
    """Power-spaced duration choices from min to max (reference
    construct_duration_space, generate_trace.py:421-429)."""
    assert base > 1.0
    powers = base ** np.linspace(1, nchoices, nchoices - 1)
    powers = np.insert(powers, 0, 0.0)
    powers = powers / powers.max()
    return np.round(powers * (max_duration_s - min_duration_s) + min_duration_s, 2)


def sample_duration(durations: np.ndarray, rng: random.Random) -> float:
    """4-bucket Pollux-style duration draw (generate_trace.py:371-401):
    72% from the shortest 20% of choices, 20% from 20-50%, 5% from 50-90%,
    3% from the top 10%."""
    probs = [0.72, 0.2, 0.05, 0.03]
    bounds = [0.2, 0.5, 0.9, 1.0]
    n = len(durations)
    cut = [round(n * b) for b in bounds[:3]]
    r = rng.random()
    if r < probs[0]:
        pool = durations[: cut[0]]
    elif r < sum(probs[:2]):
        pool = durations[cut[0] : cut[1]]
    elif r < sum(probs[:3]):
        pool = durations[cut[1] : cut[2]]
    else:
        pool = durations[cut[2] :]
    if len(pool) == 0:
        pool = durations
    return float(pool[rng.randrange(len(pool))])


def sample_scale_factor(rng: random.Random, mix: List[float]) -> int:
    assert abs(sum(mix) - 1) <= 1e-3
    r = rng.uniform(0, 1)
    if r <= mix[0]:
        return 1
    if r <= sum(mix[:2]):
        return 2
    if r <= sum(mix[:3]):
        return 4
    return 8


def sample_mode(rng: random.Random, mix: List[float]) -> str:
    r = rng.uniform(0, 1)
    if r <= mix[0]:
        return "static"
    if r <= sum(mix[:2]):
        return "accordion"
    return "gns"


def generate_job(
    throughputs: Dict,
    reference_worker_type: str,
    rng: random.Random,
    duration_s: float,
    scale_factor: int = None,
    mode: str = "static",
    run_dir: Optional[str] = None,
    job_table=None,
    multi_priority: bool = False,
    slo_rng: Optional[random.Random] = None,
) -> Job:
    if job_table is None:
        job_table = JobTable
    # short accordion jobs would shrink into super-short jobs and ruin tail
    # FTF for every policy (reference utils.py:215-217)
    if duration_s < 1000 and mode == "accordion":
        mode = "static"

    while True:
        template = rng.choice(job_table)
        if scale_factor in (None, 1) or template.distributed:
            break
    if scale_factor is None or not template.distributed:
        scale_factor = 1

    command = template.command
    if run_dir is not None:
        if template.needs_data_dir:
            command = command % (run_dir, run_dir)
        else:
            command = command % (run_dir,)

    key = (template.model, scale_factor)
    assert key in throughputs[reference_worker_type], key
    num_steps = duration_s * throughputs[reference_worker_type][key]["null"]
    assert num_steps > 0

    # optional priority tier: 20% of jobs at weight 5 (reference
    # utils.py:242-247)
    priority_weight = 1.0
    if multi_priority and rng.uniform(0, 1) <= 0.2:
        priority_weight = 5.0
    # optional SLO: factor x ideal duration from {1.2, 2, 10}
    # (reference utils.py:249-258)
    SLO = -1.0
    if slo_rng is not None:
        r = slo_rng.uniform(0, 1)
        factor = 1.2 if r < 0.33 else (2.0 if r < 0.67 else 10.0)
        SLO = factor * duration_s

    return Job(
        job_id=None,
        job_type=template.model,
        command=command,
        working_directory=template.working_directory,
        num_steps_arg=template.num_steps_arg,
        total_steps=int(num_steps),
        duration=duration_s,
        scale_factor=scale_factor,
        mode=mode,
        priority_weight=priority_weight,
        SLO=SLO,
        needs_data_dir=template.needs_data_dir,
    )


def generate_trace(
    throughputs: Dict,
    reference_worker_type: str,
    num_jobs: int,
    lam_s: float = 60.0,
    min_duration_s: float = 600.0,
    max_duration_s: float = 4 * 3600.0,
    num_durations: int = 10,
    scale_factor_mix: List[float] = (0.6, 0.3, 0.09, 0.01),
    mode_mix: List[float] = (0.0, 0.5, 0.5),
    seed: int = 0,
    multi_gpu: bool = True,
    run_dir: Optional[str] = None,
    job_table=None,
    multi_priority: bool = False,
    generate_slos: bool = False,
) -> Tuple[List[Job], List[float]]:
    """Generate a TACC-style trace: returns (jobs, arrival_times)."""
    job_rng = random.Random(seed)
    arrival_rng = random.Random(seed + 1)
    duration_rng = random.Random(seed + 2)
    sf_rng = random.Random(seed + 3)
    mode_rng = random.Random(seed + 4)
    slo_rng = random.Random(seed + 5) if generate_slos else None

    durations = duration_space(min_duration_s, max_duration_s, num_durations)

    jobs, arrivals = [], []
    t = 0.0
    for i in range(num_jobs):
        duration = sample_duration(durations, duration_rng)
        sf = sample_scale_factor(sf_rng, list(scale_factor_mix)) if multi_gpu else 1
        mode = sample_mode(mode_rng, list(mode_mix))
        job = generate_job(
            throughputs,
            reference_worker_type,
            job_rng,
            duration_s=duration,
            scale_factor=sf,
            mode=mode,
            run_dir=run_dir,
            job_table=job_table,
            multi_priority=multi_priority,
            slo_rng=slo_rng,
        )
        jobs.append(job)
        arrivals.append(round(t))
        t += generate_interarrival_time(arrival_rng, lam_s)
    return jobs, arrivals
