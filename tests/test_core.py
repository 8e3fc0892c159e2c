"""Core data-model tests: Job, JobIdPair, trace I/O, profiles, metadata."""

import math
import os
import random

import pytest

from shockwave_amd.core import Job, JobIdPair, Lease
from shockwave_amd.core import datasets, bs_patterns, generator, trace
from shockwave_amd.core.metadata import JobMetadata, build_metadata


def make_job(job_type="ResNet-18 (batch size 32)", mode="static", sf=1, steps=20000):
    return Job(
        job_id=0,
        job_type=job_type,
        command="python3 main.py --data_dir=%s/cifar10 --batch_size 32",
        working_directory="image_classification/cifar10",
        num_steps_arg="--num_steps",
        total_steps=steps,
        duration=3600,
        scale_factor=sf,
        mode=mode,
    )


class TestJob:
    def test_parse_properties(self):
        j = make_job()
        assert j.batch_size == 32
        assert j.model == "ResNet-18"
        assert j.duration == 3600

    def test_update_bs(self):
        j = make_job()
        j.update_bs(256)
        assert j.batch_size == 256
        assert "--batch_size 256" in j.command
        assert j.job_type == "ResNet-18 (batch size 256)"

    def test_update_bs_imagenet_style(self):
        j = Job(
            job_id=1,
            job_type="ResNet-50 (batch size 64)",
            command="python3 main.py -j 4 -a resnet50 -b 64 %s/imagenet/",
            working_directory="image_classification/imagenet",
            num_steps_arg="--num_minibatches",
            total_steps=100,
            duration=100,
        )
        j.update_bs(128)
        assert "-b 128 %s/imagenet/" in j.command
        assert j.batch_size == 128

    def test_trace_line_roundtrip(self, tmp_path):
        j = make_job()
        p = tmp_path / "t.trace"
        trace.write_trace([j], [42.0], str(p))
        jobs, arrivals = trace.parse_trace(str(p))
        assert len(jobs) == 1
        assert jobs[0].job_type == j.job_type
        assert jobs[0].command == j.command
        assert jobs[0].total_steps == j.total_steps
        assert arrivals == [42.0]


class TestJobIdPair:
    def test_singleton(self):
        p = JobIdPair(3)
        assert not p.is_pair()
        assert p[0] == 3 and p[1] is None
        assert p.singletons() == (p,)

    def test_pair_normalized(self):
        p = JobIdPair(5, 3)
        assert (p[0], p[1]) == (3, 5)
        assert p == JobIdPair(3, 5)
        assert p.overlaps_with(JobIdPair(5))
        assert not p.overlaps_with(JobIdPair(7))
        assert len({JobIdPair(1, 2), JobIdPair(2, 1)}) == 1

    def test_ordering(self):
        assert JobIdPair(1) < JobIdPair(1, 2) < JobIdPair(2)


class TestBsPatterns:
    def test_accordion_resnet18(self):
        pat = bs_patterns.accordion_bs_pattern("ResNet-18 (batch size 32)", 32, 100)
        assert pat[:10] == [32] * 10          # critical regime start
        assert pat[50] == 256                 # out of critical regime -> max bs
        assert len(pat) == 100

    def test_accordion_transformer_static(self):
        pat = bs_patterns.accordion_bs_pattern("Transformer (batch size 32)", 32, 50)
        assert pat == [32] * 50

    def test_gns_ladder(self):
        pat = bs_patterns.gns_bs_pattern("ResNet-18 (batch size 16)", 16, 100, 1)
        assert pat[0] == 16
        assert pat[35] == 32
        assert pat[45] == 64
        assert pat[60] == 128
        assert pat[80] == 256

    def test_gns_unknown_combo_static(self):
        pat = bs_patterns.gns_bs_pattern("ResNet-18 (batch size 256)", 256, 50, 8)
        assert pat == [256] * 50


class TestProfiles:
    def test_build_profile(self, throughputs):
        j = make_job(mode="gns", steps=1563 * 30)  # 30 epochs at bs32
        prof = trace.build_job_profile(j, throughputs)
        assert prof["num_epochs"] == 30
        assert len(prof["bs_every_epoch"]) == 30
        assert len(prof["duration_every_epoch"]) == 30
        assert prof["dataset"] == "CIFAR-10"
        # durations positive and consistent with throughput
        assert all(d > 0 for d in prof["duration_every_epoch"])

    def test_generate_profiles_pickle(self, tmp_path, throughputs):
        j = make_job()
        p = tmp_path / "t.trace"
        trace.write_trace([j, j], [0, 10], str(p))
        jobs, arrivals, profiles = trace.generate_profiles(str(p), throughputs)
        assert len(profiles) == 2
        assert os.path.exists(tmp_path / "t.pickle")


class TestMetadata:
    def _metadata(self, throughputs, mode="gns", steps=1563 * 60):
        j = make_job(mode=mode, steps=steps)
        prof = trace.build_job_profile(j, throughputs)
        return JobMetadata(0, prof)

    def test_remaining_runtime_decreases(self, throughputs):
        md = self._metadata(throughputs)
        r0 = md.remaining_runtime(progress=0)
        md.set_epoch_progress(md.epochs // 2)
        r1 = md.remaining_runtime()
        md.set_epoch_progress(md.epochs - 1)
        r2 = md.remaining_runtime()
        assert r0 > r1 > r2 > 0

    def test_oracle_runtime(self, throughputs):
        md = self._metadata(throughputs)
        assert md.remaining_runtime(oracle=True) == pytest.approx(
            sum(md.epoch_duration)
        )

    def test_calibration_rescales(self, throughputs):
        from collections import OrderedDict

        md = self._metadata(throughputs, mode="static")
        # measured throughput half of profiled -> epoch durations double
        profiled_tput = (
            md.epoch_nsamples / md.bs_schedule[0] / md.epoch_duration[0]
        )
        meas = OrderedDict({10: (profiled_tput / 2, md.bs_schedule[0])})
        md.set_throughput_measurements(meas, 120.0)
        before = md.epoch_duration_preprofiled[0]
        md.calibrate()
        assert md.epoch_duration[0] == pytest.approx(before * 2, rel=0.1)

    def test_build_metadata(self, throughputs):
        j = make_job()
        prof = trace.build_job_profile(j, throughputs)
        md = build_metadata([5], [prof])
        assert list(md.keys()) == [5]


class TestGenerator:
    def test_deterministic(self, throughputs):
        jobs1, arr1 = generator.generate_trace(throughputs, "mi355x", 20, seed=0)
        jobs2, arr2 = generator.generate_trace(throughputs, "mi355x", 20, seed=0)
        assert [str(j) for j in jobs1] == [str(j) for j in jobs2]
        assert arr1 == arr2

    def test_steps_match_duration(self, throughputs):
        jobs, _ = generator.generate_trace(throughputs, "mi355x", 30, seed=1)
        for j in jobs:
            tput = throughputs["mi355x"][(j.job_type, j.scale_factor)]["null"]
            assert j.total_steps == pytest.approx(j.duration * tput, rel=0.01)

    def test_mode_mix(self, throughputs):
        jobs, _ = generator.generate_trace(
            throughputs, "mi355x", 200, seed=2, mode_mix=(0.0, 0.5, 0.5)
        )
        modes = [j.mode for j in jobs]
        # accordion may fall back to static for short jobs
        assert modes.count("gns") > 40
        assert modes.count("accordion") > 20

    def test_trace_file_roundtrip(self, tmp_path, throughputs):
        jobs, arrivals = generator.generate_trace(throughputs, "mi355x", 10, seed=3)
        p = tmp_path / "gen.trace"
        trace.write_trace(jobs, arrivals, str(p))
        jobs2, arrivals2 = trace.parse_trace(str(p))
        assert len(jobs2) == 10
        assert [j.job_type for j in jobs2] == [j.job_type for j in jobs]


class TestGeneratorPrioritySLO:
    def test_multi_priority_and_slos(self, throughputs):
        """reference utils.py:242-258: ~20% priority-5 jobs, SLO =
        {1.2, 2, 10} x ideal duration."""
        from shockwave_amd.core import generator

        jobs, _ = generator.generate_trace(
            throughputs, "mi355x", 80, seed=0,
            multi_priority=True, generate_slos=True,
        )
        frac5 = sum(1 for j in jobs if j.priority_weight == 5.0) / len(jobs)
        assert 0.05 < frac5 < 0.4
        factors = {round(j.SLO / j.duration, 1) for j in jobs}
        assert factors <= {1.2, 2.0, 10.0}
        # defaults unchanged: no priorities/SLOs unless requested
        jobs, _ = generator.generate_trace(throughputs, "mi355x", 10, seed=0)
        assert all(j.priority_weight == 1.0 for j in jobs)
        assert all(j.SLO is None for j in jobs)

    def test_priority_respected_by_policy(self, throughputs):
        """A priority-5 job gets a larger LAS share."""
        import copy

        from shockwave_amd.core import trace as trace_mod
        from shockwave_amd.core.job import JobIdPair
        from shockwave_amd.policies import get_policy

        p = get_policy("max_min_fairness")
        tputs = {JobIdPair(0): {"mi355x": 1.0}, JobIdPair(1): {"mi355x": 1.0}}
        sf = {JobIdPair(0): 1, JobIdPair(1): 1}
        prio = {JobIdPair(0): 5.0, JobIdPair(1): 1.0}
        alloc = p.get_allocation(tputs, sf, prio, {"mi355x": 1})
        assert alloc[JobIdPair(0)]["mi355x"] > alloc[JobIdPair(1)]["mi355x"]
