"""The three control-plane services and their clients.

Mirrors the reference's gRPC surface (runtime/rpc/{scheduler_server,
scheduler_client,worker_server,worker_client,iterator_client}.py) over the
msgpack transport:

* ``WorkerToScheduler``  — RegisterWorker / SendHeartbeat / Done
* ``SchedulerToWorker``  — RunJob / KillJob / Reset / Shutdown
* ``IteratorToScheduler``— InitJob / UpdateLease / UpdateResourceRequirement
"""

from __future__ import annotations

import logging
from typing import Callable, Dict

from ..core.job import JobIdPair
from .transport import RpcClient, make_server

logger = logging.getLogger("shockwave_amd.rpc")


# ---------------------------------------------------------------------------
# head-node server (WorkerToScheduler + IteratorToScheduler)
# ---------------------------------------------------------------------------

def serve_scheduler(port: int, callbacks: Dict[str, Callable]):
    """callbacks: RegisterWorker, SendHeartbeat, Done, InitJob, UpdateLease,
    UpdateResourceRequirement (reference scheduler_server.py:36-215)."""

    def register_worker(req):
        try:
            worker_ids, round_duration = callbacks["RegisterWorker"](
                worker_type=req["worker_type"],
                num_gpus=req["num_gpus"],
                ip_addr=req["ip_addr"],
                port=req["port"],
            )
            return {
                "success": True,
                "worker_ids": list(worker_ids),
                "round_duration": round_duration,
                "error_message": "",
            }
        except Exception as e:
            logger.exception("could not register worker")
            return {"success": False, "worker_ids": [],
                    "round_duration": 0, "error_message": str(e)}

    def send_heartbeat(req):
        import inspect

        fn = callbacks["SendHeartbeat"]
        # liveness: the heartbeat carries the sender's worker ids so the
        # scheduler can track per-worker last-seen times (exceeds the
        # reference, whose heartbeat is a no-op: scheduler_server.py:36-99)
        if inspect.signature(fn).parameters:
            fn(req.get("worker_ids", []))
        else:
            fn()
        return {}

    def done(req):
        ids = req["job_id"]
        job_id = JobIdPair(ids[0], ids[1] if len(ids) > 1 else None)
        try:
            callbacks["Done"](
                job_id,
                req["worker_id"],
                req["num_steps"],
                req["execution_time"],
                req.get("iterator_log"),
            )
        except Exception:
            logger.exception("Done callback failed for %s", job_id)
        return {}

    def init_job(req):
        job_id = JobIdPair(req["job_id"], None)
        resp = callbacks["InitJob"](job_id=job_id)
        # callbacks may return the full 5-field UpdateLeaseResponse
        # (max_steps, max_duration, extra_time, run_time_so_far, deadline
        # — reference iterator_to_scheduler.proto) or the legacy 3-tuple
        if len(resp) == 5:
            max_steps, max_duration, extra_time, run_time, deadline = resp
        else:
            max_steps, max_duration, extra_time = resp
            run_time, deadline = 0, 0
        return {
            "max_steps": max_steps,
            "max_duration": max_duration,
            "extra_time": extra_time,
            "run_time_so_far": run_time,
            "deadline": deadline,
        }

    def update_lease(req):
        job_id = JobIdPair(req["job_id"], None)
        try:
            max_steps, max_duration, run_time_so_far, deadline = callbacks[
                "UpdateLease"
            ](
                job_id=job_id,
                worker_id=req["worker_id"],
                steps=req["steps"],
                duration=req["duration"],
                max_steps=req["max_steps"],
                max_duration=req["max_duration"],
            )
        except Exception:
            logger.exception("could not update lease for %s", job_id)
            max_steps, max_duration = req["max_steps"], req["max_duration"]
            run_time_so_far, deadline = 0, int(1e9)
        return {
            "max_steps": max_steps,
            "max_duration": max_duration,
            "extra_time": 0.0,
            "run_time_so_far": run_time_so_far,
            "deadline": deadline,
        }

    def update_resource_requirement(req):
        job_id = JobIdPair(req["job_id"], None)
        callbacks["UpdateResourceRequirement"](
            job_id, req["worker_id"], req["big_bs"], req["small_bs"]
        )
        return {}

    return make_server(
        port,
        {
            "WorkerToScheduler": {
                "RegisterWorker": register_worker,
                "SendHeartbeat": send_heartbeat,
                "Done": done,
            },
            "IteratorToScheduler": {
                "InitJob": init_job,
                "UpdateLease": update_lease,
                "UpdateResourceRequirement": update_resource_requirement,
            },
        },
    )


# ---------------------------------------------------------------------------
# worker-node server (SchedulerToWorker)
# ---------------------------------------------------------------------------

def serve_worker(port: int, callbacks: Dict[str, Callable]):
    """callbacks: RunJob(jobs, worker_id, round_id), KillJob(job_id),
    Reset(), Shutdown() (reference worker_server.py:1-85)."""

    def run_job(req):
        callbacks["RunJob"](
            req["job_descriptions"], req["worker_id"], req["round_id"]
        )
        return {}

    def kill_job(req):
        callbacks["KillJob"](req["job_id"])
        return {}

    def reset(req):
        callbacks["Reset"]()
        return {}

    def shutdown(req):
        callbacks["Shutdown"]()
        return {}

    def fetch_checkpoint(req):
        fn = callbacks.get("FetchCheckpoint")
        if fn is None:
            return {"found": False, "data": b"", "total": 0}
        return fn(req["job_id"], req["offset"], req["length"])

    return make_server(
        port,
        {
            "SchedulerToWorker": {
                "RunJob": run_job,
                "KillJob": kill_job,
                "Reset": reset,
                "Shutdown": shutdown,
                "FetchCheckpoint": fetch_checkpoint,
            }
        },
    )


# ---------------------------------------------------------------------------
# clients
# ---------------------------------------------------------------------------

class SchedulerRpcClient:
    """Head -> worker (reference scheduler_client.py:1-70)."""

    def __init__(self, server_ip: str, port: int):
        self.addr = server_ip
        self.port = port
        self._client = RpcClient(server_ip, port)

    def run_job(self, job_descriptions, worker_id, round_id):
        self._client.call(
            "SchedulerToWorker",
            "RunJob",
            {
                "job_descriptions": job_descriptions,
                "worker_id": worker_id,
                "round_id": round_id,
            },
        )

    def kill_job(self, job_id):
        self._client.call("SchedulerToWorker", "KillJob", {"job_id": job_id})

    def reset(self):
        self._client.call("SchedulerToWorker", "Reset", {})

    def shutdown(self):
        self._client.call("SchedulerToWorker", "Shutdown", {})


class WorkerRpcClient:
    """Worker -> head (reference worker_client.py:1-95)."""

    def __init__(self, worker_type, ip_addr, port, sched_addr, sched_port):
        self._worker_type = worker_type
        self._ip_addr = ip_addr
        self._port = port
        self._client = RpcClient(sched_addr, sched_port)

    def register_worker(self, num_gpus: int):
        resp = self._client.call(
            "WorkerToScheduler",
            "RegisterWorker",
            {
                "worker_type": self._worker_type,
                "num_gpus": num_gpus,
                "ip_addr": self._ip_addr,
                "port": self._port,
            },
        )
        if not resp.get("success"):
            return None, None, resp.get("error_message", "unknown error")
        return resp["worker_ids"], resp["round_duration"], None

    def send_heartbeat(self, worker_ids=None):
        self._client.call(
            "WorkerToScheduler", "SendHeartbeat",
            {"worker_ids": list(worker_ids or [])},
        )

    def notify_scheduler(self, worker_id, job_descriptions):
        """job_descriptions: [(job_id, num_steps, execution_time, log)]"""
        job_ids, num_steps, execution_times, logs = [], [], [], []
        for jid, steps, t, log in job_descriptions:
            job_ids.append(jid)
            num_steps.append(steps)
            execution_times.append(t)
            logs.append(log)
        self._client.call(
            "WorkerToScheduler",
            "Done",
            {
                "worker_id": worker_id,
                "job_id": job_ids,
                "num_steps": num_steps,
                "execution_time": execution_times,
                "iterator_log": logs,
            },
        )


class IteratorRpcClient:
    """Training process -> head (reference iterator_client.py:1-94)."""

    def __init__(self, job_id, worker_id, sched_addr, sched_port, logger=None):
        self._job_id = job_id
        self._worker_id = worker_id
        self._logger = logger or logging.getLogger("shockwave_amd.rpc")
        self._client = RpcClient(sched_addr, sched_port)

    def init(self):
        resp = self._client.call(
            "IteratorToScheduler", "InitJob", {"job_id": self._job_id},
            timeout=60,
        )
        return (
            resp["max_steps"],
            resp["max_duration"],
            resp["extra_time"],
            resp.get("run_time_so_far", 0),
            resp.get("deadline", 0),
        )

    def update_lease(self, steps, duration, max_steps, max_duration):
        resp = self._client.call(
            "IteratorToScheduler",
            "UpdateLease",
            {
                "job_id": self._job_id,
                "worker_id": self._worker_id,
                "steps": steps,
                "duration": duration,
                "max_steps": max_steps,
                "max_duration": max_duration,
            },
            timeout=60,
        )
        return (
            resp["max_steps"],
            resp["max_duration"],
            resp["run_time_so_far"],
            resp["deadline"],
        )

    def update_resource_requirement(self, big_bs, small_bs):
        self._client.call(
            "IteratorToScheduler",
            "UpdateResourceRequirement",
            {
                "job_id": self._job_id,
                "worker_id": self._worker_id,
                "big_bs": big_bs,
                "small_bs": small_bs,
            },
        )
