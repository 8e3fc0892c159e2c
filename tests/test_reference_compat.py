"""Compatibility with the reference's on-disk formats.

These tests run only where the reference checkout is present (this build
container); they prove a user can bring their existing Gavel/Shockwave
traces to this framework unchanged.
"""

import glob
import os

import pytest

from shockwave_amd.core import trace as trace_mod
from shockwave_amd.core.job_table import JobTable
from shockwave_amd.core.job import Job

REFERENCE = "/root/reference/scheduler"
REPO = os.path.join(os.path.dirname(__file__), "..")

needs_reference = pytest.mark.skipif(
    not os.path.isdir(REFERENCE), reason="reference checkout not present"
)


@needs_reference
class TestReferenceTraceCompat:
    def _trace(self):
        paths = glob.glob(
            os.path.join(REFERENCE, "traces", "reproduce", "120_*.trace")
        )
        assert paths
        return paths[0]

    def test_parse_reference_trace(self):
        jobs, arrivals = trace_mod.parse_trace(self._trace())
        assert len(jobs) == 120
        assert all(isinstance(j, Job) for j in jobs)
        assert arrivals == sorted(arrivals)
        modes = {j.mode for j in jobs}
        assert modes <= {"static", "accordion", "gns"}

    def test_commands_map_to_our_workload_tree(self):
        """Every (working_directory, entry script) pair in the reference
        trace exists in our workloads/ trees."""
        jobs, _ = trace_mod.parse_trace(self._trace())
        for j in jobs:
            entry = j.command.split()[1]  # e.g. main.py / train.py
            for tree in ("pytorch", "accordion", "gns"):
                path = os.path.join(
                    REPO, "workloads", tree, j.working_directory, entry
                )
                assert os.path.exists(path), path

    def test_job_types_priced_by_our_oracle(self, throughputs):
        jobs, _ = trace_mod.parse_trace(self._trace())
        wt = "mi355x"
        for j in jobs:
            assert (j.job_type, j.scale_factor) in throughputs[wt], j.job_type

    def test_simulate_reference_trace(self, throughputs):
        """The reference's own 120-job trace runs through our simulator
        (bounded rounds; full completion takes minutes)."""
        from shockwave_amd.engine import RoundScheduler
        from shockwave_amd.policies import get_policy

        jobs, arrivals = trace_mod.parse_trace(self._trace())
        profiles = [
            trace_mod.build_job_profile(j, throughputs) for j in jobs
        ]
        for j, pr in zip(jobs, profiles):
            j.duration = sum(pr["duration_every_epoch"])
        sched = RoundScheduler(
            get_policy("max_min_fairness"),
            simulate=True,
            throughputs=throughputs,
            time_per_iteration=120,
            profiles=profiles,
            worker_type="mi355x",
            max_rounds=40,
        )
        sched.simulate({"mi355x": 32}, arrivals, jobs)
        assert sched._num_completed_rounds == 40
        assert len(sched.get_job_completion_times()) > 0


class TestUpdateBsAllTemplates:
    @pytest.mark.parametrize("template", JobTable, ids=lambda t: t.model)
    def test_update_bs_rewrites_every_template(self, template):
        job = Job(
            job_id=None,
            job_type=template.model,
            command=template.command,
            working_directory=template.working_directory,
            num_steps_arg=template.num_steps_arg,
            total_steps=100,
            duration=100,
        )
        old_bs = job.batch_size
        job.update_bs(old_bs * 2)
        assert job.batch_size == old_bs * 2
        assert str(old_bs * 2) in job.command
