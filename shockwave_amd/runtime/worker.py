"""Worker daemon: registers GPUs with the head and dispatches jobs.

Rebuild of scheduler/worker.py:23-233: brings up a SchedulerToWorker gRPC
server, registers this machine's GPUs with the scheduler (one worker id
per GPU), and forwards RunJob/KillJob to the dispatcher.
"""

from __future__ import annotations

import argparse
import logging
import socket
import threading

from ..rpc.services import WorkerRpcClient, serve_worker
from .dispatcher import Dispatcher
from .gpu import get_num_gpus

logger = logging.getLogger("shockwave_amd.worker")


class Worker:
    def __init__(
        self,
        worker_type: str,
        sched_addr: str,
        sched_port: int,
        worker_port: int,
        num_gpus: int = None,
        ip_addr: str = None,
        run_dir: str = ".",
        static_run_dir: str = None,
        accordion_run_dir: str = None,
        gns_run_dir: str = None,
        data_dir: str = None,
        checkpoint_dir: str = "/tmp/swq_checkpoints",
        heartbeat_interval_s: float = 30.0,
    ):
        self._heartbeat_interval_s = heartbeat_interval_s
        if num_gpus is None:
            num_gpus = get_num_gpus()
        assert num_gpus > 0, "no GPUs found"
        if ip_addr is None:
            try:
                ip_addr = socket.gethostbyname(socket.gethostname())
            except socket.gaierror:
                ip_addr = "127.0.0.1"
        self.worker_type = worker_type
        self.num_gpus = num_gpus

        self._checkpoint_dir = checkpoint_dir
        self._server = serve_worker(
            worker_port,
            {
                "RunJob": self._run_job_callback,
                "KillJob": self._kill_job_callback,
                "Reset": self._reset_callback,
                "Shutdown": self._shutdown_callback,
                "FetchCheckpoint": self._fetch_checkpoint_callback,
            },
        )

        self._rpc_client = WorkerRpcClient(
            worker_type, ip_addr, worker_port, sched_addr, sched_port
        )
        worker_ids, round_duration, error = self._rpc_client.register_worker(
            num_gpus
        )
        if error:
            raise RuntimeError(f"worker registration failed: {error}")
        self.worker_ids = worker_ids
        self.round_duration = round_duration
        logger.info(
            "registered %d GPUs as worker ids %s (round %ss)",
            num_gpus, worker_ids, round_duration,
        )
        # worker_id -> local gpu index
        self._worker_id_to_gpu = {
            wid: i for i, wid in enumerate(worker_ids)
        }

        self._dispatcher = Dispatcher(
            round_duration,
            list(range(num_gpus)),
            self._rpc_client,
            sched_addr,
            sched_port,
            run_dir=run_dir,
            data_dir=data_dir,
            checkpoint_dir=checkpoint_dir,
            static_run_dir=static_run_dir,
            accordion_run_dir=accordion_run_dir,
            gns_run_dir=gns_run_dir,
        )
        self._done = threading.Event()
        self._heartbeat_thread = threading.Thread(
            target=self._heartbeat_loop, daemon=True
        )
        self._heartbeat_thread.start()

    def _heartbeat_loop(self):
        while not self._done.wait(self._heartbeat_interval_s):
            try:
                self._rpc_client.send_heartbeat(self.worker_ids)
            except Exception:
                logger.warning("heartbeat to scheduler failed", exc_info=True)

    # -- RPC callbacks -------------------------------------------------------

    def _run_job_callback(self, job_descriptions, worker_id, round_id):
        # pin the job to the GPU backing this worker id
        gpu = self._worker_id_to_gpu.get(worker_id)
        if gpu is not None:
            # targeted slot checkout happens inside the dispatcher via the
            # job's assigned gpu; jobs queue per-GPU
            pass
        self._dispatcher.dispatch_jobs(job_descriptions, worker_id, round_id)

    def _kill_job_callback(self, job_id):
        self._dispatcher.kill_job(job_id)

    def _reset_callback(self):
        self._dispatcher.reset()

    def _shutdown_callback(self):
        self._dispatcher.shutdown()
        self._done.set()

    def _fetch_checkpoint_callback(self, job_id, offset, length):
        """Serve checkpoint bytes for cross-node migration
        (shockwave_amd/parallel/ckpt_stream.py)."""
        from ..parallel.ckpt_stream import CheckpointStore

        store = CheckpointStore(
            os.path.join(self._checkpoint_dir, f"job_id={job_id}")
        )
        data = store.read_bytes()
        if data is None:
            return {"found": False, "data": b"", "total": 0}
        return {
            "found": True,
            "data": data[offset : offset + length],
            "total": len(data),
        }

    def join(self):
        self._done.wait()
        self._server.stop(5)


def main():
    p = argparse.ArgumentParser(description="shockwave_amd worker daemon")
    p.add_argument("-t", "--worker_type", default="mi355x")
    p.add_argument("-i", "--ip_addr", default=None)
    p.add_argument("-s", "--sched_addr", default="127.0.0.1")
    p.add_argument("--sched_port", type=int, default=50070)
    p.add_argument("-w", "--worker_port", type=int, default=50061)
    p.add_argument("-g", "--num_gpus", type=int, default=None)
    p.add_argument("--run_dir", default=".")
    p.add_argument("--static_run_dir", default=None)
    p.add_argument("--accordion_run_dir", default=None)
    p.add_argument("--gns_run_dir", default=None)
    p.add_argument("--data_dir", default=None)
    p.add_argument("--checkpoint_dir", default="/tmp/swq_checkpoints")
    args = p.parse_args()
    logging.basicConfig(level=logging.INFO)
    worker = Worker(
        args.worker_type,
        args.sched_addr,
        args.sched_port,
        args.worker_port,
        num_gpus=args.num_gpus,
        ip_addr=args.ip_addr,
        run_dir=args.run_dir,
        static_run_dir=args.static_run_dir,
        accordion_run_dir=args.accordion_run_dir,
        gns_run_dir=args.gns_run_dir,
        data_dir=args.data_dir,
        checkpoint_dir=args.checkpoint_dir,
    )
    worker.join()


if __name__ == "__main__":
    main()
