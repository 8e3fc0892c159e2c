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


@needs_reference
class TestReferenceParity:
    """Run OUR simulator on the reference's own trace + V100 oracle and
    compare against the result pickles the reference repo ships
    (reproduce/pickles/tacc_32gpus) — profiles/REFERENCE_PARITY.md."""

    TRACE = os.path.join(
        REFERENCE, "traces", "reproduce",
        "120_0.2_5_100_40_25_0,0.5,0.5_0.6,0.3,0.09,0.01"
        "_multigpu_dynamic.trace",
    )
    PICKLES = os.path.join(REFERENCE, "reproduce", "pickles", "tacc_32gpus")

    @staticmethod
    def _load_reference_pickle(policy):
        import pickle

        class _RefJobIdPair:
            def __init__(self, *a, **k):
                pass

            def __hash__(self):
                return hash((self.__dict__.get("_job0"),
                             self.__dict__.get("_job1")))

            def __eq__(self, o):
                return (isinstance(o, _RefJobIdPair)
                        and self.__dict__ == o.__dict__)

        class RefUnpickler(pickle.Unpickler):
            def find_class(self, module, name):
                if module == "job_id_pair" and name == "JobIdPair":
                    return _RefJobIdPair
                return super().find_class(module, name)

        path = os.path.join(
            TestReferenceParity.PICKLES,
            f"{policy}_120_0.2_5_100_40_25_0,0.5,0.5_0.6,0.3,0.09,0.01"
            "_multigpu_dynamic_simulation.pickle",
        )
        with open(path, "rb") as f:
            return RefUnpickler(f).load()

    def _run_ours(self, policy, shockwave_config=None):
        import copy

        from shockwave_amd.core.throughputs import read_throughputs
        from shockwave_amd.engine import RoundScheduler
        from shockwave_amd.policies import get_policy

        tputs = read_throughputs(
            os.path.join(REFERENCE, "tacc_throughputs.json")
        )
        jobs, arrivals = trace_mod.parse_trace(self.TRACE)
        profiles = [trace_mod.build_job_profile(j, tputs) for j in jobs]
        for j, pr in zip(jobs, profiles):
            j.duration = sum(pr["duration_every_epoch"])
        sched = RoundScheduler(
            get_policy(policy), simulate=True, throughputs=tputs,
            time_per_iteration=120, profiles=profiles,
            shockwave_config=shockwave_config, worker_type="v100",
            seed=0, preemption_overhead_s=20.0,
        )
        makespan = sched.simulate({"v100": 32}, list(arrivals),
                                  copy.deepcopy(jobs))
        return sched, makespan

    def test_max_min_fairness_matches_reference_exactly(self):
        """Deterministic greedy mechanism on identical inputs: makespan
        within 0.5%, avg JCT and worst rho within 2%."""
        ref = self._load_reference_pickle("max_min_fairness")
        sched, makespan = self._run_ours("max_min_fairness")
        assert len(sched.get_job_completion_times()) == 120
        assert abs(makespan - ref["makespan"]) / ref["makespan"] < 0.005
        assert (abs(sched.get_average_jct()[0] - float(ref["avg_jct"]))
                / float(ref["avg_jct"]) < 0.02)
        ftf, _ = sched.get_finish_time_fairness()
        ref_rho = max(ref["finish_time_fairness_list"])
        assert abs(max(ftf) - ref_rho) / ref_rho < 0.02

    def test_allox_matches_reference(self):
        ref = self._load_reference_pickle("allox")
        sched, makespan = self._run_ours("allox")
        assert abs(makespan - ref["makespan"]) / ref["makespan"] < 0.01

    def test_shockwave_matches_reference(self):
        """Independent EG MILP (scipy/HiGHS vs Gurobi): makespan and avg
        JCT within 5% of the reference's committed shockwave result."""
        import json

        cfg = json.load(open(os.path.join(
            REFERENCE, "configurations", "tacc_32gpus.json")))
        cfg["time_per_iteration"] = 120
        cfg["num_gpus"] = 32
        ref = self._load_reference_pickle("shockwave")
        sched, makespan = self._run_ours("shockwave", shockwave_config=cfg)
        assert len(sched.get_job_completion_times()) == 120
        assert abs(makespan - ref["makespan"]) / ref["makespan"] < 0.05
        assert (abs(sched.get_average_jct()[0] - float(ref["avg_jct"]))
                / float(ref["avg_jct"]) < 0.05)
        # round-2 band: worst rho within 10% of the
        # reference and unfair fraction equal
        ftf, _ = sched.get_finish_time_fairness()
        ref_rhos = ref["finish_time_fairness_list"]
        assert max(ftf) <= 1.9
        assert abs(max(ftf) - max(ref_rhos)) / max(ref_rhos) < 0.10
        unfair = sum(1 for r in ftf if r > 1.05)
        ref_unfair = sum(1 for r in ref_rhos if r > 1.05)
        assert unfair == ref_unfair


class TestProtoSchemaParity:
    """Our .proto files (shockwave_amd/rpc/protos/) must carry the
    reference's field names and numbers for every shared message, and the
    msgpack transport must key its maps by those field names.  protoc is
    not in this image; when it exists, `protoc --python_out` over our
    protos yields wire-compatible stubs."""

    OURS = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "shockwave_amd", "rpc", "protos",
    )
    REF_DIR = os.path.join(REFERENCE, "runtime", "protobuf")

    @staticmethod
    def _parse(path):
        import re

        msgs = {}
        cur = None
        for line in open(path):
            line = line.split("//")[0].strip()
            m = re.match(r"message (\w+) \{", line)
            if m:
                cur = m.group(1)
                msgs[cur] = {}
                continue
            if line.startswith("}"):
                cur = None
                continue
            f = re.match(
                r"(?:repeated )?[\w.]+ (\w+) = (\d+);", line
            )
            if cur and f:
                msgs[cur][f.group(1)] = int(f.group(2))
        return msgs

    def test_message_fields_match_reference(self):
        if not os.path.isdir(self.REF_DIR):
            pytest.skip("reference tree not present")
        ours, ref = {}, {}
        for d, out in ((self.OURS, ours), (self.REF_DIR, ref)):
            for name in os.listdir(d):
                if name.endswith(".proto"):
                    out.update(self._parse(os.path.join(d, name)))
        shared = set(ours) & set(ref)
        # every reference message must exist in ours
        assert set(ref) - {"Empty", "JobState"} <= set(ours), (
            set(ref) - set(ours)
        )
        for msg in shared:
            for fname, fnum in ref[msg].items():
                assert ours[msg].get(fname) == fnum, (
                    f"{msg}.{fname}: ours {ours[msg].get(fname)} "
                    f"!= ref {fnum}"
                )

    def test_transport_uses_proto_field_names(self):
        """The msgpack maps sent by the clients use exactly the proto
        field names (schema-level wire compatibility)."""
        ours = {}
        for name in os.listdir(self.OURS):
            if name.endswith(".proto"):
                ours.update(self._parse(os.path.join(self.OURS, name)))
        import inspect

        from shockwave_amd.rpc import services

        src = inspect.getsource(services)
        for msg, fields in ours.items():
            if msg in ("Empty", "JobState", "JobDescription", "Heartbeat"):
                continue
            for fname in fields:
                if fname in ("job_state",):
                    continue
                assert f'"{fname}"' in src, (
                    f"field {msg}.{fname} not used by services.py"
                )
