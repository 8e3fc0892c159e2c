"""Control-plane tests: RPC loopback, lease iterator protocol, dispatcher."""

import os
import threading
import time

import pytest
import torch

from shockwave_amd.core.job import JobIdPair
from shockwave_amd.rpc.services import (
    IteratorRpcClient,
    SchedulerRpcClient,
    WorkerRpcClient,
    serve_scheduler,
    serve_worker,
)
from shockwave_amd.runtime.lease_iterator import LeaseIterator, NullLeaseClient
from shockwave_amd.runtime.set_queue import SetQueue


def free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class TestSetQueue:
    def test_targeted_get(self):
        q = SetQueue()
        q.put(1)
        q.put(2)
        assert q.get(item=2) == 2
        assert q.get() == 1

    def test_blocking_get(self):
        q = SetQueue()
        result = []

        def getter():
            result.append(q.get(item=7, timeout=5))

        t = threading.Thread(target=getter)
        t.start()
        time.sleep(0.05)
        q.put(7)
        t.join(timeout=5)
        assert result == [7]


class TestSchedulerRpcLoopback:
    def test_register_done_and_lease(self):
        calls = {}

        def register(worker_type, num_gpus, ip_addr, port):
            calls["register"] = (worker_type, num_gpus, ip_addr, port)
            return [0, 1], 120

        def done(job_id, worker_id, num_steps, execution_time, logs):
            calls["done"] = (job_id, worker_id, list(num_steps))

        def init_job(job_id):
            calls["init"] = job_id
            return 500, 60.0, 3.0, 17, 9000

        def update_lease(job_id, worker_id, steps, duration, max_steps,
                         max_duration):
            return 1000, 120.0, 42, 10000

        def update_rr(job_id, worker_id, big_bs, small_bs):
            calls["rr"] = (job_id, big_bs, small_bs)

        port = free_port()
        server = serve_scheduler(
            port,
            {
                "RegisterWorker": register,
                "SendHeartbeat": lambda wids: calls.setdefault(
                    "heartbeats", []).append(list(wids)),
                "Done": done,
                "InitJob": init_job,
                "UpdateLease": update_lease,
                "UpdateResourceRequirement": update_rr,
            },
        )
        try:
            wc = WorkerRpcClient("mi355x", "127.0.0.1", 1234, "127.0.0.1", port)
            worker_ids, round_duration, err = wc.register_worker(2)
            assert err is None
            assert worker_ids == [0, 1]
            assert round_duration == 120
            wc.notify_scheduler(0, [(3, 100, 12.5, "log")])
            assert calls["done"] == (JobIdPair(3), 0, [100])

            ic = IteratorRpcClient(7, 0, "127.0.0.1", port)
            assert ic.init() == (500, 60.0, 3.0, 17, 9000)
            assert ic.update_lease(10, 5.0, 500, 60.0) == (1000, 120.0, 42, 10000)
            ic.update_resource_requirement(True, False)
            assert calls["rr"] == (JobIdPair(7), True, False)
            wc.send_heartbeat([0, 1])
            assert calls["heartbeats"] == [[0, 1]]
        finally:
            server.stop(0)


class TestWorkerRpcLoopback:
    def test_run_and_kill(self):
        got = {}

        def run_job(jobs, worker_id, round_id):
            got["run"] = (jobs, worker_id, round_id)

        port = free_port()
        server = serve_worker(
            port,
            {
                "RunJob": run_job,
                "KillJob": lambda jid: got.setdefault("kill", jid),
                "Reset": lambda: got.setdefault("reset", True),
                "Shutdown": lambda: got.setdefault("shutdown", True),
            },
        )
        try:
            sc = SchedulerRpcClient("127.0.0.1", port)
            jobs = [{"job_id": 5, "command": "echo hi", "job_type": "x",
                     "working_directory": ".", "needs_data_dir": False,
                     "num_steps_arg": "--steps", "num_steps": 10,
                     "mode": "static", "mps_thread_percentage": 100}]
            sc.run_job(jobs, worker_id=1, round_id=2)
            assert got["run"][1] == 1 and got["run"][2] == 2
            sc.kill_job(5)
            assert got["kill"] == 5
            sc.reset()
            sc.shutdown()
            assert got["reset"] and got["shutdown"]
        finally:
            server.stop(0)


class FakeLeaseClient:
    """Scripted lease behavior for iterator state-machine tests."""

    def __init__(self, leases):
        self.leases = list(leases)  # [(max_steps, max_duration)]
        self.init_calls = 0
        self.update_calls = 0
        self.rr_calls = []

    def init(self):
        self.init_calls += 1
        ms, md = self.leases.pop(0)
        return ms, md, 0

    def update_lease(self, steps, duration, max_steps, max_duration):
        self.update_calls += 1
        if self.leases:
            ms, md = self.leases.pop(0)
        else:
            ms, md = max_steps, max_duration
        return ms, md, 0, 1e9

    def update_resource_requirement(self, big, small):
        self.rr_calls.append((big, small))


class TestLeaseIterator:
    def _data(self, n=100):
        return [(torch.zeros(1), torch.zeros(1)) for _ in range(n)]

    def test_expires_at_max_steps(self, tmp_path):
        client = FakeLeaseClient([(5, 1e9)])
        it = LeaseIterator(
            self._data(), str(tmp_path), lambda: None, lambda s: None,
            client=client, write_on_close=False,
        )
        consumed = list(it)
        assert len(consumed) == 5
        assert it.done

    def test_renews_at_75pct(self, tmp_path):
        client = FakeLeaseClient([(8, 1e9), (16, 1e9)])
        it = LeaseIterator(
            self._data(), str(tmp_path), lambda: None, lambda s: None,
            client=client, write_on_close=False,
        )
        consumed = list(it)
        # renewal granted at 75% of 8 steps -> extended to 16 total
        assert client.update_calls >= 1
        assert len(consumed) == 16
        assert it.done

    def test_infinite_lease_runs_all_data(self, tmp_path):
        it = LeaseIterator(
            self._data(30), str(tmp_path), lambda: None, lambda s: None,
            client=NullLeaseClient(), write_on_close=False,
        )
        assert len(list(it)) == 30
        assert not it.done

    def test_update_resource_requirement_sets_done(self, tmp_path):
        client = FakeLeaseClient([(100, 1e9)])
        it = LeaseIterator(
            self._data(), str(tmp_path), lambda: None, lambda s: None,
            client=client, write_on_close=False,
        )
        it.update_resource_requirement(True, False)
        assert it.done
        assert client.rr_calls == [(True, False)]

    def test_deadline_abort(self, tmp_path):
        class DeadlineClient(FakeLeaseClient):
            def update_lease(self, steps, duration, max_steps, max_duration):
                self.update_calls += 1
                return 1000, 1e9, 1e6, 100  # run_time >> deadline

        client = DeadlineClient([(4, 1e9)])
        it = LeaseIterator(
            self._data(), str(tmp_path), lambda: None, lambda s: None,
            client=client, write_on_close=False,
        )
        consumed = list(it)
        assert it.done  # completed via deadline mechanism
        assert len(consumed) < 1000

    def test_log_format_scrapeable(self, tmp_path):
        os.environ["GAVEL_ROUND_ID"] = "3"
        os.environ["GAVEL_WORKER_ID"] = "1"
        try:
            client = FakeLeaseClient([(5, 1e9)])
            it = LeaseIterator(
                self._data(), str(tmp_path), lambda: None, lambda s: None,
                client=client, write_on_close=False,
            )
            list(it)
            it._write_info()
            it._file_handler.flush()
            from shockwave_amd.runtime.dispatcher import LOG_LINE_RE

            log_file = tmp_path / ".gavel" / "round=3" / "worker=1.log"
            steps = None
            for line in open(log_file):
                m = LOG_LINE_RE.match(line)
                if m and m.group("event") == "PROGRESS" and m.group("status") == "STEPS":
                    steps = int(float(m.group("msg")))
            assert steps == 5
        finally:
            del os.environ["GAVEL_ROUND_ID"]
            del os.environ["GAVEL_WORKER_ID"]


class TestDispatcherProcess:
    def test_launch_scrape_and_notify(self, tmp_path):
        """Dispatch a real subprocess that writes an iterator-format log."""
        from shockwave_amd.runtime.dispatcher import Dispatcher

        notified = {}

        class FakeClient:
            def notify_scheduler(self, worker_id, jobs):
                notified["jobs"] = jobs
                notified["worker_id"] = worker_id

        ckpt_dir = tmp_path / "ckpts"
        job_dir = ckpt_dir / "job_id=9" / ".gavel" / "round=0"
        script = (
            "import os; d=r'%s'; os.makedirs(d, exist_ok=True); "
            "open(os.path.join(d,'worker=2.log'),'w').write("
            "'[2026-01-01 00:00:00] [PROGRESS] [STEPS] 17\\n"
            "[2026-01-01 00:00:00] [PROGRESS] [DURATION] 3.5\\n')"
        ) % str(job_dir)
        d = Dispatcher(
            round_duration=10,
            gpu_ids=[0],
            worker_rpc_client=FakeClient(),
            sched_addr="127.0.0.1",
            sched_port=1,
            run_dir=str(tmp_path),
            data_dir=None,
            checkpoint_dir=str(ckpt_dir),
        )
        job = {
            "job_id": 9,
            "command": f"python3 -c \"{script}\" || true; true",
            "working_directory": ".",
            "needs_data_dir": False,
            "num_steps_arg": "--steps",
            "num_steps": 100,
            "mode": "static",
        }
        # bypass CLI arg appending by running helper directly: the fake
        # command ignores the appended args because of the trailing true
        d._dispatch_jobs_helper([job], worker_id=2, round_id=0)
        assert notified["worker_id"] == 2
        job_id, steps, duration, log = notified["jobs"][0]
        assert (job_id, steps, duration) == (9, 17, 3.5)
        assert "[PROGRESS]" in log

    def test_kill_job(self, tmp_path):
        from shockwave_amd.runtime.dispatcher import Dispatcher

        class FakeClient:
            def notify_scheduler(self, worker_id, jobs):
                pass

        d = Dispatcher(
            round_duration=10, gpu_ids=[0], worker_rpc_client=FakeClient(),
            sched_addr="127.0.0.1", sched_port=1, run_dir=str(tmp_path),
            data_dir=None, checkpoint_dir=str(tmp_path),
        )
        job = {
            "job_id": 11,
            "command": "sleep 600; true",
            "working_directory": ".",
            "needs_data_dir": False,
            "num_steps_arg": "--steps",
            "num_steps": 1,
            "mode": "static",
        }
        t = threading.Thread(
            target=d._safe_dispatch, args=([job], 0, 0), daemon=True
        )
        t.start()
        deadline = time.time() + 5
        while 11 not in d._procs and time.time() < deadline:
            time.sleep(0.05)
        assert 11 in d._procs
        start = time.time()
        d.kill_job(11)
        t.join(timeout=15)
        assert not t.is_alive()
        assert time.time() - start < 12


class TestPackedDispatch:
    def test_pair_runs_coresident_one_done(self, tmp_path):
        """Two job_descriptions in one RunJob share ONE GPU slot and are
        reported in a single Done with both ids."""
        from shockwave_amd.runtime.dispatcher import Dispatcher

        notified = {}

        class FakeClient:
            def notify_scheduler(self, worker_id, jobs):
                notified["jobs"] = jobs

        ckpt_dir = tmp_path / "ckpts"

        def mk(job_id, steps):
            d = ckpt_dir / f"job_id={job_id}" / ".gavel" / "round=0"
            script = (
                "import os; d=r'%s'; os.makedirs(d, exist_ok=True); "
                "open(os.path.join(d,'worker=0.log'),'w').write("
                "'[2026-01-01 00:00:00] [PROGRESS] [STEPS] %d\\n"
                "[2026-01-01 00:00:00] [PROGRESS] [DURATION] 1.0\\n')"
            ) % (str(d), steps)
            return {
                "job_id": job_id,
                "command": f"python3 -c \"{script}\" || true; true",
                "working_directory": ".",
                "needs_data_dir": False,
                "num_steps_arg": "--steps",
                "num_steps": 100,
                "mode": "static",
            }

        d = Dispatcher(
            round_duration=10, gpu_ids=[0], worker_rpc_client=FakeClient(),
            sched_addr="127.0.0.1", sched_port=1, run_dir=str(tmp_path),
            data_dir=None, checkpoint_dir=str(ckpt_dir),
        )
        d._dispatch_jobs_helper([mk(1, 11), mk(2, 22)], worker_id=0,
                                round_id=0)
        got = {jid: steps for jid, steps, _, _ in notified["jobs"]}
        assert got == {1: 11, 2: 22}


class TestGpuDiscoveryAndLogging:
    def test_get_num_gpus_cpu_box(self):
        """On a GPU-less box discovery degrades to 0 (or a real count if
        smi tools are present) without raising."""
        from shockwave_amd.runtime.gpu import get_num_gpus

        n = get_num_gpus()
        assert isinstance(n, int) and n >= 0

    def test_gpu_processes_best_effort(self):
        from shockwave_amd.runtime.gpu import get_gpu_processes

        procs = get_gpu_processes()
        assert isinstance(procs, dict)

    def test_scheduler_adapter_prefixes_clock(self):
        """SchedulerAdapter prefixes records with the scheduler clock
        (reference custom_logging.py:5-13)."""
        import logging as _logging

        from shockwave_amd.utils.logging import SchedulerAdapter

        class FakeSched:
            def get_current_timestamp(self):
                return 1234.5

        records = []

        class Capture(_logging.Handler):
            def emit(self, record):
                records.append(record.getMessage())

        lg = _logging.getLogger("swq.adapter.test")
        lg.setLevel(_logging.INFO)
        lg.addHandler(Capture())
        SchedulerAdapter(lg, FakeSched()).info("round %d done", 7)
        assert records and records[0].startswith("[1234.50] ")
        assert "round 7 done" in records[0]
