"""End-to-end physical-mode test: head + worker + real training subprocesses
over the live gRPC control plane, on CPU (BASELINE config 2 shape, scaled
down)."""

import json
import os
import socket
import subprocess
import sys
import time

import pytest

from shockwave_amd.core.job import Job
from shockwave_amd.core import trace as trace_mod

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def tiny_job(steps=6, bs=16):
    return Job(
        job_id=None,
        job_type=f"ResNet-18 (batch size {bs})",
        command=f"python3 main.py --batch_size {bs}",
        working_directory="image_classification/cifar10",
        num_steps_arg="--num_steps",
        total_steps=steps,
        duration=600,
        scale_factor=1,
        mode="static",
    )


@pytest.mark.slow
class TestPhysicalEndToEnd:
    @pytest.mark.parametrize("codec", ["msgpack", "proto"])
    def test_single_job_completes_through_rpc(self, tmp_path, throughputs,
                                              codec, monkeypatch):
        # "proto" runs the whole control plane on TRUE protobuf wire
        # bodies from the runtime-built stubs (rpc/pb.py) — the
        # reference's protoc wire format, kept green in every CPU run
        monkeypatch.setenv("SWQ_RPC_CODEC", codec)
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        port = free_port()
        worker_port = free_port()
        jobs = [tiny_job(steps=6)]
        profiles = [trace_mod.build_job_profile(j, throughputs) for j in jobs]

        sched = PhysicalScheduler(
            get_policy("max_min_fairness"),
            port=port,
            expected_num_workers=1,
            throughputs=throughputs,
            time_per_iteration=30,
            profiles=profiles,
            worker_type="mi355x",
        )
        try:
            worker = Worker(
                worker_type="mi355x",
                sched_addr="127.0.0.1",
                sched_port=port,
                worker_port=worker_port,
                num_gpus=1,
                ip_addr="127.0.0.1",
                run_dir=os.path.join(REPO, "workloads", "pytorch"),
                accordion_run_dir=os.path.join(REPO, "workloads", "accordion"),
                gns_run_dir=os.path.join(REPO, "workloads", "gns"),
                checkpoint_dir=str(tmp_path),
            )
            assert worker.round_duration == 30
            sched.add_job(jobs[0])

            deadline = time.time() + 240
            while not sched.is_done() and time.time() < deadline:
                time.sleep(2)
            completions = sched.get_job_completion_times()
            assert len(completions) == 1, "job did not complete"
            (jct,) = completions.values()
            assert 0 < jct < 240
            # the job ran over the real dispatcher: its checkpoint exists
            assert os.path.exists(
                os.path.join(str(tmp_path), "job_id=0", "model.chkpt")
            )
        finally:
            sched.shutdown()


@pytest.mark.slow
class TestPhysicalAdaptation:
    def test_accordion_rescale_through_full_stack(self, tmp_path, throughputs):
        """An accordion job leaves its critical regime inside the real
        training subprocess, reports big_bs over gRPC, the scheduler
        rewrites the command (epoch-preserving step rescale), and the job
        finishes at the new batch size (BASELINE config 4 core loop)."""
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        os.environ["SWQ_DATASET_LEN"] = "96"  # 3 steps/epoch at bs 32
        try:
            port = free_port()
            worker_port = free_port()
            job = Job(
                job_id=None,
                job_type="ResNet-18 (batch size 32)",
                command="python3 main.py --batch_size 32",
                working_directory="image_classification/cifar10",
                num_steps_arg="--num_steps",
                total_steps=60,
                duration=600,
                scale_factor=1,
                mode="accordion",
            )
            profiles = [trace_mod.build_job_profile(job, throughputs)]
            sched = PhysicalScheduler(
                get_policy("max_min_fairness"),
                port=port,
                expected_num_workers=1,
                throughputs=throughputs,
                time_per_iteration=25,
                profiles=profiles,
                worker_type="mi355x",
            )
            try:
                Worker(
                    worker_type="mi355x",
                    sched_addr="127.0.0.1",
                    sched_port=port,
                    worker_port=worker_port,
                    num_gpus=1,
                    ip_addr="127.0.0.1",
                    run_dir=os.path.join(REPO, "workloads", "pytorch"),
                    accordion_run_dir=os.path.join(
                        REPO, "workloads", "accordion"
                    ),
                    gns_run_dir=os.path.join(REPO, "workloads", "gns"),
                    checkpoint_dir=str(tmp_path),
                )
                sched.add_job(job)
                deadline = time.time() + 300
                while not sched.is_done() and time.time() < deadline:
                    time.sleep(2)
                assert len(sched.get_job_completion_times()) == 1, \
                    "accordion job did not complete"
                # the scheduler rescaled the job to the family's max bs
                assert job.batch_size == 256, job.command
                assert "--batch_size 256" in job.command
            finally:
                sched.shutdown()
        finally:
            del os.environ["SWQ_DATASET_LEN"]


@pytest.mark.slow
class TestPhysicalDistributedJob:
    def test_scale_factor_2_job_with_shockwave_policy(self, tmp_path,
                                                      throughputs):
        """A 2-GPU data-parallel job: the scheduler assigns two worker ids,
        injects --master_addr/--master_port/--world_size/--rank at
        dispatch, both ranks rendezvous (gloo on CPU), renew leases via
        the first-requester path, and the job completes under the
        shockwave planner."""
        import json

        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        port = free_port()
        worker_port = free_port()
        job = Job(
            job_id=None,
            job_type="ResNet-18 (batch size 16)",
            command="python3 main.py --batch_size 16",
            working_directory="image_classification/cifar10",
            num_steps_arg="--num_steps",
            total_steps=12,
            duration=600,
            scale_factor=2,
            mode="static",
        )
        profiles = [trace_mod.build_job_profile(job, throughputs)]
        shockwave_config = {
            "future_rounds": 5, "k": 1e-3, "lambda": 12.0, "rhomax": 1.0,
            "time_per_iteration": 30, "num_gpus": 2,
        }
        sched = PhysicalScheduler(
            get_policy("shockwave"),
            port=port,
            expected_num_workers=2,
            throughputs=throughputs,
            time_per_iteration=30,
            profiles=profiles,
            shockwave_config=shockwave_config,
            worker_type="mi355x",
        )
        try:
            Worker(
                worker_type="mi355x",
                sched_addr="127.0.0.1",
                sched_port=port,
                worker_port=worker_port,
                num_gpus=2,
                ip_addr="127.0.0.1",
                run_dir=os.path.join(REPO, "workloads", "pytorch"),
                accordion_run_dir=os.path.join(REPO, "workloads", "accordion"),
                gns_run_dir=os.path.join(REPO, "workloads", "gns"),
                checkpoint_dir=str(tmp_path),
            )
            sched.add_job(job)
            deadline = time.time() + 300
            while not sched.is_done() and time.time() < deadline:
                time.sleep(2)
            completions = sched.get_job_completion_times()
            assert len(completions) == 1, "2-GPU job did not complete"
            # both ranks contributed steps
            steps = sched.get_completed_steps()
            assert list(steps.values())[0] >= 12
        finally:
            sched.shutdown()


@pytest.mark.slow
class TestWatchdog:
    def test_hung_job_killed_and_dropped(self, tmp_path, throughputs):
        """A job whose process never reports (hangs) is killed by the
        watchdog each round, gets a synthesized zero-step done callback,
        and after MAX_FAILED_ATTEMPTS is dropped from the scheduler
        (reference :4201-4281, :4536-4569)."""
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.engine.scheduler import MAX_FAILED_ATTEMPTS
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        port = free_port()
        worker_port = free_port()
        job = Job(
            job_id=None,
            job_type="ResNet-18 (batch size 16)",
            # ignores every appended flag and sleeps forever: no iterator
            # log is ever written
            command="sleep 600 ; true",
            working_directory="image_classification/cifar10",
            num_steps_arg="--num_steps",
            total_steps=100,
            duration=600,
            scale_factor=1,
            mode="static",
        )
        profiles = [trace_mod.build_job_profile(job, throughputs)]
        sched = PhysicalScheduler(
            get_policy("max_min_fairness"),
            port=port,
            expected_num_workers=1,
            completion_buffer_s=4,
            throughputs=throughputs,
            time_per_iteration=10,
            profiles=profiles,
            worker_type="mi355x",
        )
        try:
            Worker(
                worker_type="mi355x",
                sched_addr="127.0.0.1",
                sched_port=port,
                worker_port=worker_port,
                num_gpus=1,
                ip_addr="127.0.0.1",
                run_dir=os.path.join(REPO, "workloads", "pytorch"),
                checkpoint_dir=str(tmp_path),
            )
            sched.add_job(job)
            deadline = time.time() + 240
            while not sched.is_done() and time.time() < deadline:
                time.sleep(2)
            assert sched.is_done(), "hung job was never dropped"
            jid = next(iter(sched.get_job_completion_times()))
            assert sched._num_failures_per_job.get(jid, MAX_FAILED_ATTEMPTS) \
                >= 0  # removed from active accounting
            # the job is recorded as completed (with its elapsed duration)
            assert len(sched.get_job_completion_times()) == 1
        finally:
            sched.shutdown()


@pytest.mark.slow
class TestMultiWorkerCluster:
    def test_four_gpu_worker_mixed_jobs(self, tmp_path, throughputs):
        """4 GPU slots on one node, 3 single-GPU jobs + 1 two-GPU job under
        the shockwave planner: placement respects capacity, the DP job
        spans two worker ids, and everything completes."""
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        port = free_port()
        worker_port = free_port()
        jobs = [tiny_job(steps=6) for _ in range(3)]
        dp = Job(
            job_id=None,
            job_type="ResNet-18 (batch size 16)",
            command="python3 main.py --batch_size 16",
            working_directory="image_classification/cifar10",
            num_steps_arg="--num_steps",
            total_steps=8,
            duration=600,
            scale_factor=2,
            mode="static",
        )
        jobs.append(dp)
        profiles = [trace_mod.build_job_profile(j, throughputs) for j in jobs]
        shockwave_config = {
            "future_rounds": 4, "k": 1e-3, "lambda": 12.0, "rhomax": 1.0,
            "time_per_iteration": 25, "num_gpus": 4,
        }
        sched = PhysicalScheduler(
            get_policy("shockwave"),
            port=port,
            expected_num_workers=4,
            throughputs=throughputs,
            time_per_iteration=25,
            profiles=profiles,
            shockwave_config=shockwave_config,
            worker_type="mi355x",
        )
        try:
            Worker(
                worker_type="mi355x",
                sched_addr="127.0.0.1",
                sched_port=port,
                worker_port=worker_port,
                num_gpus=4,
                ip_addr="127.0.0.1",
                run_dir=os.path.join(REPO, "workloads", "pytorch"),
                checkpoint_dir=str(tmp_path),
            )
            for j in jobs:
                sched.add_job(j)
            deadline = time.time() + 300
            while not sched.is_done() and time.time() < deadline:
                time.sleep(2)
            completions = sched.get_job_completion_times()
            assert len(completions) == 4, f"only {len(completions)}/4 done"
            # capacity was never exceeded in any round
            for rnd in sched.get_per_round_schedule():
                assert sum(len(w) for w in rnd.values()) <= 4
            # the DP job was placed on two distinct workers at least once
            dp_id = jobs[-1].job_id[0]
            dp_rounds = [r[dp_id] for r in sched.get_per_round_schedule()
                         if dp_id in r]
            assert any(len(set(w)) == 2 for w in dp_rounds)
        finally:
            sched.shutdown()


class TestWorkerLiveness:
    def test_silent_worker_deregistered(self, throughputs):
        """Unit-level liveness: a worker that stops heartbeating is
        deregistered at the next round boundary; live workers stay."""
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy

        sched = PhysicalScheduler(
            get_policy("max_min_fairness"),
            port=free_port(),
            expected_num_workers=99,  # round loop stays parked
            throughputs=throughputs,
            time_per_iteration=100,
            profiles=[],
            worker_type="mi355x",
            heartbeat_timeout_s=0.2,
        )
        try:
            a = sched._register_worker_callback("mi355x", 1, "127.0.0.1", 1)
            b = sched._register_worker_callback("mi355x", 1, "127.0.0.1", 2)
            wa, wb = a[0][0], b[0][0]
            assert sched._cluster_spec["mi355x"] == 2
            time.sleep(0.3)
            sched._heartbeat_callback([wa])  # only A beats
            with sched._scheduler_cv:
                sched._check_worker_liveness()
            assert wb not in sched._worker_connections
            assert wa in sched._worker_connections
            assert sched._cluster_spec["mi355x"] == 1
            assert wb not in sched._worker_ids
            # id->type mapping survives for late Done callbacks
            assert sched._worker_id_to_worker_type_mapping[wb] == "mi355x"
        finally:
            sched.shutdown(shutdown_workers=False)


@pytest.mark.slow
class TestWorkerDeathMidTrace:
    def test_jobs_complete_after_worker_dies(self, tmp_path, throughputs):
        """kill a worker mid-trace; its job must be
        rescheduled onto the survivor and all jobs must complete."""
        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        port = free_port()
        jobs = [tiny_job(steps=8), tiny_job(steps=8)]
        profiles = [trace_mod.build_job_profile(j, throughputs) for j in jobs]
        sched = PhysicalScheduler(
            get_policy("max_min_fairness"),
            port=port,
            expected_num_workers=2,
            throughputs=throughputs,
            time_per_iteration=25,
            profiles=profiles,
            worker_type="mi355x",
            heartbeat_timeout_s=8.0,
        )
        workers = []
        try:
            for _ in range(2):
                workers.append(Worker(
                    worker_type="mi355x",
                    sched_addr="127.0.0.1",
                    sched_port=port,
                    worker_port=free_port(),
                    num_gpus=1,
                    ip_addr="127.0.0.1",
                    run_dir=os.path.join(REPO, "workloads", "pytorch"),
                    checkpoint_dir=str(tmp_path),
                    heartbeat_interval_s=2.0,
                ))
            for j in jobs:
                sched.add_job(j)
            # let round 0 dispatch both jobs (one per worker), then kill
            # worker B: stop heartbeats + server, kill its job processes
            time.sleep(6)
            dead = workers[1]
            dead._done.set()
            dead._server.stop(0)
            dead._dispatcher.shutdown()

            deadline = time.time() + 300
            while not sched.is_done() and time.time() < deadline:
                time.sleep(2)
            completions = sched.get_job_completion_times()
            assert len(completions) == 2, (
                f"jobs did not complete after worker death: {completions}"
            )
            # the dead worker must have been deregistered
            assert len(sched._worker_connections) == 1
        finally:
            sched.shutdown()
            for w in workers[:1]:
                try:
                    w._shutdown_callback()
                except Exception:
                    pass


@pytest.mark.slow
class TestEightWorkerDressRehearsal:
    def test_mixed_trace_on_eight_workers(self, tmp_path, throughputs):
        """8 worker slots (2 daemons x 4 GPUs), a
        compressed mixed trace — all three modes, scale factors 1/2/4,
        five families — at short rounds on CPU (gloo).  Shakes out
        dispatch/port/lease races ahead of the driver's 8-GPU node."""
        import random

        from shockwave_amd.engine.physical import PhysicalScheduler
        from shockwave_amd.policies import get_policy
        from shockwave_amd.runtime.worker import Worker

        rng = random.Random(0)
        specs = [
            ("ResNet-18 (batch size 16)",
             "python3 main.py --batch_size 16",
             "image_classification/cifar10", "--num_steps"),
            ("ResNet-18 (batch size 32)",
             "python3 main.py --batch_size 32",
             "image_classification/cifar10", "--num_steps"),
            ("LM (batch size 10)",
             "python3 main.py --batch_size 10",
             "language_modeling", "--steps"),
            ("Recommendation (batch size 512)",
             "python3 train.py --batch_size 512",
             "recommendation", "-n"),
        ]
        jobs = []
        for i in range(12):
            jt, cmd, wd, steps_arg = specs[i % len(specs)]
            sf = rng.choices([1, 2], weights=[0.7, 0.3])[0]
            # sf=4 = 50% of this 8-slot cluster; the reference caps
            # job size at 25% of its 32-GPU cluster — mirrored here
            mode = rng.choice(["static", "static", "gns", "accordion"])
            if "Recommendation" in jt:
                sf = 1  # not distributed in the reference
            jobs.append(Job(
                job_id=None, job_type=jt, command=cmd,
                working_directory=wd, num_steps_arg=steps_arg,
                total_steps=rng.randint(4, 8) * sf,
                duration=600, scale_factor=sf, mode=mode,
            ))
        profiles = [trace_mod.build_job_profile(j, throughputs) for j in jobs]

        port = free_port()
        sched = PhysicalScheduler(
            get_policy("max_min_fairness"),
            port=port,
            expected_num_workers=8,
            throughputs=throughputs,
            time_per_iteration=40,
            profiles=profiles,
            worker_type="mi355x",
            heartbeat_timeout_s=180.0,
            completion_buffer_s=120.0,
        )
        workers = []
        try:
            for w in range(2):
                workers.append(Worker(
                    worker_type="mi355x",
                    sched_addr="127.0.0.1",
                    sched_port=port,
                    worker_port=free_port(),
                    num_gpus=4,
                    ip_addr="127.0.0.1",
                    run_dir=os.path.join(REPO, "workloads", "pytorch"),
                    checkpoint_dir=str(tmp_path / f"node{w}"),
                    heartbeat_interval_s=5.0,
                ))
            for j in jobs:
                sched.add_job(j)
            deadline = time.time() + 720
            while not sched.is_done() and time.time() < deadline:
                time.sleep(3)
            completions = sched.get_job_completion_times()
            assert len(completions) == len(jobs), (
                f"only {len(completions)}/{len(jobs)} jobs completed"
            )
            steps = sched.get_completed_steps()
            for j, (jid, done) in zip(jobs, sorted(steps.items())):
                assert done >= 1, f"job {jid} made no progress"
        finally:
            sched.shutdown()
