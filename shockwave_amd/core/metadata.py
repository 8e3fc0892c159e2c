"""Per-job runtime model for the Shockwave planner.

Rebuild of the reference's JobMetaData (/root/reference/scheduler/JobMetaData.py):

* holds the pre-profiled epoch-level profile (batch size / memory /
  utilization / duration per epoch),
* calibrates the pre-profiled epoch durations online against measured
  throughput (``calibrate``), matching JobMetaData.py:225-288,
* forecasts remaining runtime through a Dirichlet posterior over the job's
  batch-size modes (JobMetaData.py:315-370): the planner does not know the
  job's future batch-size switches, so it treats the observed per-epoch
  batch sizes as draws from a categorical distribution with a uniform
  Dirichlet prior over the modes seen in the profile, and prices remaining
  epochs by the posterior-mean mix of per-mode epoch durations.
"""

from __future__ import annotations

import copy
import random
from collections import OrderedDict
from typing import Dict, List, Optional

import numpy as np

INFINITY = 1e9


class JobMetadata:
    def __init__(self, job_id, profile: Dict, overclock: float = 1.0):
        assert isinstance(profile, dict) and profile
        self.job_id = job_id
        self.profile = profile

        self.model_name = profile["model"]
        self.dataset_name = profile["dataset"]
        self.name = f"ID_{job_id}_{self.model_name}_{self.dataset_name}"
        self.nworkers = int(profile.get("scale_factor", 1))

        self.epochs = int(profile["num_epochs"])
        assert self.epochs > 0
        self.epoch_nsamples = profile["num_samples_per_epoch"]

        self.epoch_gpu_req = list(profile["util_every_epoch"])
        assert len(self.epoch_gpu_req) == self.epochs
        # MB -> GB, one decimal (JobMetaData.py:100-103)
        self.epoch_gram_req = [
            round(m / 1024.0, 1) for m in profile["mem_every_epoch"]
        ]
        assert len(self.epoch_gram_req) == self.epochs

        self.epoch_duration = [
            max(1.0, round(d)) / float(overclock)
            for d in profile["duration_every_epoch"]
        ]
        self.epoch_duration = [max(1.0, d) for d in self.epoch_duration]
        assert len(self.epoch_duration) == self.epochs
        self.epoch_duration_preprofiled = list(self.epoch_duration)

        self.bs_schedule = list(profile["bs_every_epoch"])
        assert len(self.bs_schedule) == self.epochs
        self.bs_modes = sorted(set(self.bs_schedule))
        # uniform Dirichlet prior: total concentration = num epochs
        self.bs_dirichlet_prior = {
            bs: self.epochs / len(self.bs_modes) for bs in self.bs_modes
        }

        self.throughput_measurements: Optional[OrderedDict] = None
        self.round_duration: Optional[float] = None

        self.epoch_progress = 0
        self.timestamp_submit: Optional[float] = None
        self.timestamp_completion: Optional[float] = None
        self.waiting_delay = 0.0

    # -- progress bookkeeping ----------------------------------------------

    def set_epoch_progress(self, progress: int) -> None:
        assert 0 <= progress <= self.epochs
        self.epoch_progress = progress

    def add_waiting_delay(self, delay: float) -> None:
        self.waiting_delay += delay

    def reset_waiting_delay(self) -> None:
        self.waiting_delay = 0.0

    def register_submit(self, t: float) -> None:
        if self.timestamp_submit is None:
            self.timestamp_submit = t

    def register_completion(self, t: float) -> None:
        if self.timestamp_completion is None:
            self.timestamp_completion = t

    # -- online calibration -------------------------------------------------

    def set_throughput_measurements(
        self, measurements: OrderedDict, round_duration: float
    ) -> None:
        """measurements: {round_index: (steps/s, batch_size)}."""
        self.throughput_measurements = measurements
        self.round_duration = round_duration

    def calibrate(self, tolerance: float = 0.4) -> None:
        """Rescale pre-profiled epoch durations when measured sample
        throughput disagrees by more than ``tolerance`` (reference
        calibrate_profiled_epoch_duration, JobMetaData.py:225-288)."""
        if not self.throughput_measurements:
            return
        assert self.round_duration is not None
        timeline = sorted(self.throughput_measurements.keys())
        prev_round = 0
        measured_nsamples = 0.0
        for cur in timeline:
            tput, bs = self.throughput_measurements[cur][:2]
            measured_nsamples += bs * tput * self.round_duration * (cur - prev_round)
            prev_round = cur
        measured_time_range = self.round_duration * max(timeline)

        preprof_time, preprof_nsamples = 0.0, 0.0
        iepoch = 0
        for iepoch, duration in enumerate(self.epoch_duration_preprofiled):
            if preprof_time + duration > measured_time_range:
                break
            preprof_time += duration
            preprof_nsamples += self.epoch_nsamples
        in_epoch_deficit = measured_time_range - preprof_time
        if in_epoch_deficit > 0:
            preprof_nsamples += (
                self.epoch_nsamples * in_epoch_deficit / self.epoch_duration[iepoch]
            )

        if measured_nsamples <= 0 or preprof_nsamples <= 0:
            return
        if abs(measured_nsamples - preprof_nsamples) / preprof_nsamples <= tolerance:
            return
        amp = preprof_nsamples / measured_nsamples
        self.epoch_duration = [
            d * amp for d in self.epoch_duration_preprofiled
        ]

    # -- Dirichlet forecast --------------------------------------------------

    def interpolated_epoch_duration(self) -> float:
        """Mean duration of epochs run so far (+ current) — the planner's
        per-epoch price (shockwave.py:322-324)."""
        self.calibrate()
        return float(np.mean(self.epoch_duration[: self.epoch_progress + 1]))

    def bs_epoch_duration_map(self) -> Dict[int, float]:
        self.calibrate()
        out: Dict[int, List[float]] = {}
        for duration, bs in zip(self.epoch_duration, self.bs_schedule):
            out.setdefault(bs, []).append(duration)
        result = {}
        for bs, ds in out.items():
            mean = float(np.mean(ds))
            assert 0 < mean < INFINITY
            result[bs] = mean
        return result

    def remaining_runtime(
        self, progress: int = None, oracle: bool = False, noise_level: float = 0.0
    ) -> float:
        """Dirichlet-posterior forecast of remaining runtime in seconds
        (JobMetaData.py:315-370)."""
        if progress is None:
            progress = self.epoch_progress
        assert 0 <= progress <= self.epochs

        if oracle:
            return float(sum(self.epoch_duration[self.epoch_progress :]))

        observed = self.bs_schedule[: progress + 1]
        posterior = copy.deepcopy(self.bs_dirichlet_prior)
        for bs in observed:
            posterior[bs] += 1

        total = sum(posterior.values())
        rebased = {
            bs: self.epochs * conc / total for bs, conc in posterior.items()
        }
        # subtract the epochs already observed per mode (floored at 0)
        for bs in observed:
            if rebased[bs] >= 1:
                rebased[bs] -= 1

        if not rebased:
            return 1.0

        inflated_remaining = int(sum(rebased.values()) + 1)
        remaining = self.epochs - self.epoch_progress
        inflated_remaining = max(inflated_remaining, remaining)
        if inflated_remaining <= 0 or remaining <= 0:
            return 1.0

        price = self.bs_epoch_duration_map()
        runtime = sum(rebased[bs] * price[bs] for bs in rebased)
        runtime *= remaining / inflated_remaining
        if noise_level:
            runtime *= 1.0 + random.choice([1, -1]) * noise_level
            if noise_level >= 1.0:
                runtime = max(runtime, 1.0)
        return float(runtime)


def build_metadata(
    job_ids: list, profiles: List[Dict], overclock: float = 1.0
) -> OrderedDict:
    assert 0 < len(job_ids) <= len(profiles)
    out = OrderedDict()
    for job_id, profile in zip(job_ids, profiles):
        out[job_id] = JobMetadata(job_id, profile, overclock=overclock)
    return out
