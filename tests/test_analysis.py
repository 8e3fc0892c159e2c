"""Tests for the analysis/calibration tooling (scripts/) and the
fidelity-mode scheduling behavior they exposed."""

import json
import os
import pickle
import subprocess
import sys

import pytest

REPO = os.path.join(os.path.dirname(__file__), "..")
SCRIPTS = os.path.join(REPO, "scripts")
sys.path.insert(0, SCRIPTS)
sys.path.insert(0, REPO)

PHYS_PICKLE = os.path.join(REPO, "profiles", "fid_phys_warm.pickle")
TRACE = os.path.join(REPO, "traces", "fidelity_5job.trace")
ORACLE = os.path.join(REPO, "traces", "mi355x_throughputs.json")


class TestCalibrateSim:
    @pytest.mark.skipif(not os.path.exists(PHYS_PICKLE),
                        reason="committed physical pickle absent")
    def test_trace_recovers_job_types(self, tmp_path):
        """A pickle that predates the job_types field must still
        calibrate when --trace is given (job id = trace line order)."""
        out_oracle = tmp_path / "oracle.json"
        out_startup = tmp_path / "startup.json"
        r = subprocess.run(
            [sys.executable, os.path.join(SCRIPTS, "calibrate_sim.py"),
             "--physical", PHYS_PICKLE, "--oracle", ORACLE,
             "--trace", TRACE,
             "--out_oracle", str(out_oracle),
             "--out_startup", str(out_startup)],
            capture_output=True, text=True, timeout=120,
        )
        assert r.returncode == 0, r.stderr
        startup = json.load(open(out_startup))
        # keys are job TYPES, not job ids
        assert "ResNet-50 (batch size 16)" in startup
        assert all(not k.isdigit() for k in startup)
        # and the oracle got hot-rate overrides ("N rates updated")
        assert "(17 rates updated)" in r.stdout

    @pytest.mark.skipif(not os.path.exists(PHYS_PICKLE),
                        reason="committed physical pickle absent")
    def test_without_trace_no_silent_id_keys(self, tmp_path):
        """Without --trace on a pre-upgrade pickle the calibration has
        nothing to key on; it must not write job-id-keyed tables that
        silently never match the oracle."""
        out_oracle = tmp_path / "oracle.json"
        out_startup = tmp_path / "startup.json"
        r = subprocess.run(
            [sys.executable, os.path.join(SCRIPTS, "calibrate_sim.py"),
             "--physical", PHYS_PICKLE, "--oracle", ORACLE,
             "--out_oracle", str(out_oracle),
             "--out_startup", str(out_startup)],
            capture_output=True, text=True, timeout=120,
        )
        assert r.returncode == 0, r.stderr
        assert "(0 rates updated)" in r.stdout


class TestFidelityScheduling:
    def test_no_empty_rounds_with_late_arrivals(self):
        """Fidelity mode recomputes the allocation every round, so jobs
        arriving between (reference-throttled) resets must be scheduled
        immediately and no recorded round may be empty while jobs wait
        (the bug behind the r1 9.2% LAS JCT gap)."""
        from simulate import run_simulation

        r = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60, preemption_overhead_s=5,
            warm_overhead_s=0.5, midround_staleness=True,
            fixed_rounds=True,
        )
        # there were always >= 1 runnable jobs until the last completion
        assert all(r["per_round_schedule"]), (
            "empty round recorded while jobs were runnable: %s"
            % r["per_round_schedule"]
        )

    def test_sim_decoder_roundtrip(self, tmp_path):
        from simulate import run_simulation

        r = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60,
        )
        p = tmp_path / "sim.pickle"
        with open(p, "wb") as f:
            pickle.dump(r, f)
        out = subprocess.run(
            [sys.executable, os.path.join(SCRIPTS, "analyze_jobs.py"),
             "--simulation", str(p)],
            capture_output=True, text=True, timeout=120,
        )
        assert out.returncode == 0, out.stderr
        assert "round -> jobs:" in out.stdout
        # one line per job with its jct
        for i in range(5):
            assert f"job {i}:" in out.stdout


class TestDeadlineClip:
    def test_micro_task_clipped_at_deadline_crossing(self, tmp_path):
        """In fixed-rounds fidelity mode a job crossing 1.5x its
        reference duration self-completes mid-round (like the
        iterator's projection abort), instead of running to the round
        boundary first."""
        from simulate import run_simulation

        r = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60, midround_staleness=True,
            fixed_rounds=True,
        )
        r_noclip = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60,
        )
        # both complete all five jobs either way
        assert len(r["jct_list"]) == len(r_noclip["jct_list"]) == 5
        # fidelity mode's makespan is the last completion, not a round
        # boundary multiple
        assert r["makespan_s"] % 60 != pytest.approx(0.0, abs=1e-6)


class TestWorldSplit:
    def test_world_rates_clock_the_sim_not_the_policy(self, tmp_path):
        """Belief/world throughput split: with a world oracle at half
        the believed rates, jobs take about twice as long — the sim's
        clock follows the WORLD table while the policy still plans on
        its beliefs (the physical scheduler's situation)."""
        import json

        from simulate import run_simulation

        belief = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60, midround_staleness=True,
            fixed_rounds=True,
        )
        slow = json.load(open(ORACLE))
        for table in slow.values():
            for entry in table.values():
                if isinstance(entry, dict) and "null" in entry:
                    entry["null"] = (
                        entry["null"] / 2.0
                        if isinstance(entry["null"], (int, float))
                        else entry["null"]
                    )
        world_path = tmp_path / "slow_world.json"
        with open(world_path, "w") as f:
            json.dump(slow, f)
        split = run_simulation(
            TRACE, ORACLE, "max_min_fairness", num_gpus=1,
            time_per_iteration=60, midround_staleness=True,
            fixed_rounds=True, world_throughputs_file=str(world_path),
        )
        # jobs need more rounds at half speed...
        assert split["makespan_s"] > belief["makespan_s"]
        growth = max(
            s_ / b_ for s_, b_ in zip(split["jct_list"], belief["jct_list"])
        )
        assert growth > 1.10
        # ...but the slowdown is CAPPED by the (faithful) deadline clip:
        # at half the believed rate every job crosses 1.5x its believed
        # duration and self-completes, exactly as the iterator would
        assert split["makespan_s"] < 2.0 * belief["makespan_s"]
        assert len(split["jct_list"]) == len(belief["jct_list"]) == 5
