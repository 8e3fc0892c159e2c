"""Per-worker job dispatcher.

Rebuild of the reference dispatcher (runtime/rpc/dispatcher.py:1-634):

* GPU-slot queue: one slot per local GPU, checked out per job
  (:64-69, :521),
* builds the shell command — job command + ``--local_rank`` +
  ``<num_steps_arg> N`` + ``--checkpoint_dir`` + ``--enable_gavel_iterator``
  (:179-206) — and runs it from the mode-specific run dir (:352-358),
* exports the iterator env contract (GAVEL_JOB_ID/WORKER_ID/ROUND_ID/
  SCHED_ADDR/SCHED_PORT) and pins the GPU via HIP/ROCR_VISIBLE_DEVICES
  (the reference pins CUDA_VISIBLE_DEVICES; :385-399),
* scrapes the iterator round log for (steps, duration) (:208-237),
* kills job processes by tracked process group (:239-295; the reference
  greps command lines for pids — tracking the Popen handle is safer),
* reports Done to the scheduler with the full iterator log (:611).

CUDA MPS has no ROCm equivalent and is not needed: packed jobs simply
share the GPU, and HIP's hardware scheduler time-slices them (SURVEY §2.4
row 10).
"""

from __future__ import annotations

import logging
import os
import re
import signal
import subprocess
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Optional

from .set_queue import SetQueue

logger = logging.getLogger("shockwave_amd.dispatcher")

LOG_LINE_RE = re.compile(
    r"^\[(?P<ts>[^\]]+)\] \[(?P<event>[^\]]+)\] \[(?P<status>[^\]]+)\] ?(?P<msg>.*)$"
)


class Dispatcher:
    def __init__(
        self,
        round_duration: float,
        gpu_ids: List[int],
        worker_rpc_client,
        sched_addr: str,
        sched_port: int,
        run_dir: str,
        data_dir: Optional[str],
        checkpoint_dir: str,
        static_run_dir: Optional[str] = None,
        accordion_run_dir: Optional[str] = None,
        gns_run_dir: Optional[str] = None,
        warm: bool = True,
    ):
        self._round_duration = round_duration
        self._gpu_ids = gpu_ids
        self._gpu_queue = SetQueue()
        for gpu_id in gpu_ids:
            self._gpu_queue.put(gpu_id)
        self._worker_rpc_client = worker_rpc_client
        self._sched_addr = sched_addr
        self._sched_port = sched_port
        self._run_dir = run_dir
        self._data_dir = data_dir
        self._checkpoint_dir = checkpoint_dir
        self._run_dirs = {
            "static": static_run_dir or run_dir,
            "accordion": accordion_run_dir or run_dir,
            "gns": gns_run_dir or run_dir,
        }
        self._lock = threading.Lock()
        self._procs: Dict[int, subprocess.Popen] = {}  # job_id -> proc
        self._killed = set()
        self._pool = ThreadPoolExecutor(max_workers=max(8, 2 * len(gpu_ids)))
        # persistent warm runners (one+ per GPU slot): jobs run in-process
        # in a long-lived python that holds torch/HIP/MIOpen state, so a
        # dispatch costs ~0.1 s instead of full process startup
        self._warm = warm
        self._runner_pool = None
        self._job_runners: Dict[int, object] = {}  # job_id -> WarmRunner
        if warm:
            from .warm_runner import RunnerPool

            self._runner_pool = RunnerPool(
                os.path.join(checkpoint_dir, ".runners"),
                extra_env={
                    # set at spawn so MIOpen picks it up at library init
                    "MIOPEN_USER_DB_PATH": os.path.join(
                        checkpoint_dir, ".miopen"
                    ),
                },
            )

    # -- command construction ----------------------------------------------

    def _construct_command(self, job, gpu_id: int, worker_id: int) -> str:
        command = job["command"]
        if job.get("needs_data_dir") and self._data_dir and "%s" in command:
            n = command.count("%s")
            command = command % tuple([self._data_dir] * n)
        command = f"{command} --local_rank {gpu_id}"
        command = f"{command} {job['num_steps_arg']} {job['num_steps']}"
        ckpt = os.path.join(self._checkpoint_dir, f"job_id={job['job_id']}")
        command = f"{command} --checkpoint_dir {ckpt}"
        command = f"{command} --enable_gavel_iterator"
        mode = job.get("mode")
        if mode:
            command = f"{command} --mode {mode}"
        return command

    def _job_working_dir(self, job) -> str:
        base = self._run_dirs.get(job.get("mode", "static"), self._run_dir)
        return os.path.join(base, job["working_directory"])

    # -- log scraping --------------------------------------------------------

    def _get_steps_and_execution_time(self, job_id, worker_id, round_id):
        log_file = os.path.join(
            self._checkpoint_dir,
            f"job_id={job_id}",
            ".gavel",
            f"round={round_id}",
            f"worker={worker_id}.log",
        )
        steps, duration, lines = 0, 0.0, []
        try:
            with open(log_file) as f:
                for line in f:
                    lines.append(line.rstrip("\n"))
                    m = LOG_LINE_RE.match(line)
                    if not m:
                        continue
                    if m.group("event") == "PROGRESS":
                        if m.group("status") == "STEPS":
                            steps = int(float(m.group("msg")))
                        elif m.group("status") == "DURATION":
                            duration = float(m.group("msg"))
        except FileNotFoundError:
            logger.warning("no iterator log at %s", log_file)
        return steps, duration, "\n".join(lines)

    # -- launch / kill -------------------------------------------------------

    def _job_env(self, job, worker_id, round_id, gpu_id, pin_gpu=True):
        env = {
            "GAVEL_JOB_ID": str(job["job_id"]),
            "GAVEL_WORKER_ID": str(worker_id),
            "GAVEL_ROUND_ID": str(round_id),
            "GAVEL_SCHED_ADDR": self._sched_addr,
            "GAVEL_SCHED_PORT": str(self._sched_port),
            "SWQ_MODE": job.get("mode", "static"),
            # MIOpen find-db persists under the (typically shared)
            # checkpoint dir: a migrated job's destination node skips the
            # convolution-algorithm search its source already paid for
            "MIOPEN_USER_DB_PATH": os.path.join(
                self._checkpoint_dir, ".miopen"
            ),
        }
        if pin_gpu:
            env.update(
                {
                    "HIP_VISIBLE_DEVICES": str(gpu_id),
                    "ROCR_VISIBLE_DEVICES": str(gpu_id),
                    "CUDA_VISIBLE_DEVICES": str(gpu_id),
                }
            )
        return env

    def launch_job_warm(self, job, command, worker_id, round_id, gpu_id):
        """Run a lease inside the GPU slot's persistent runner."""
        job_id = job["job_id"]
        runner = self._runner_pool.acquire(gpu_id)
        cwd = self._job_working_dir(job)
        # runner already pins the GPU through its own env
        env = self._job_env(job, worker_id, round_id, gpu_id, pin_gpu=False)
        log_path = os.path.join(
            self._checkpoint_dir, f"job_id={job_id}", "job_output.log"
        )
        logger.info(
            "[worker %s round %s] warm-dispatching job %s on gpu %s: %s",
            worker_id, round_id, job_id, gpu_id, command,
        )
        with self._lock:
            self._job_runners[job_id] = runner
        try:
            rc = runner.run(command, cwd, env, log_path)
        finally:
            with self._lock:
                self._job_runners.pop(job_id, None)
                was_killed = job_id in self._killed
                self._killed.discard(job_id)
            self._runner_pool.release(runner)
        if rc != 0 and not was_killed:
            tail = ""
            try:
                with open(log_path, "rb") as f:
                    tail = b"\n".join(
                        f.read().splitlines()[-20:]
                    ).decode(errors="replace")
            except OSError:
                pass
            logger.error("job %s exited rc=%s; output tail:\n%s",
                         job_id, rc, tail)
        return rc, b""

    def launch_job(self, job, command, worker_id, round_id, gpu_id):
        # the warm runner executes python entrypoints in-process; anything
        # else (shell pipelines, non-python tools) takes the cold path
        if self._warm and os.path.basename(
            command.split()[0]
        ).startswith("python"):
            return self.launch_job_warm(job, command, worker_id, round_id,
                                        gpu_id)
        job_id = job["job_id"]
        env = dict(os.environ)
        env.update(self._job_env(job, worker_id, round_id, gpu_id))
        cwd = self._job_working_dir(job)
        logger.info(
            "[worker %s round %s] launching job %s on gpu %s: %s (cwd %s)",
            worker_id, round_id, job_id, gpu_id, command, cwd,
        )
        proc = subprocess.Popen(
            command,
            shell=True,
            cwd=cwd,
            env=env,
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
            start_new_session=True,  # own process group for clean kills
        )
        with self._lock:
            self._procs[job_id] = proc
        stdout, _ = proc.communicate()
        with self._lock:
            self._procs.pop(job_id, None)
            was_killed = job_id in self._killed
            self._killed.discard(job_id)
        if proc.returncode != 0 and not was_killed:
            logger.error(
                "job %s exited rc=%s; output tail:\n%s",
                job_id, proc.returncode,
                b"\n".join(stdout.splitlines()[-20:]).decode(errors="replace"),
            )
        return proc.returncode, stdout

    def _dispatch_jobs_helper(self, job_descriptions, worker_id, round_id):
        """One RunJob request = one worker (GPU) for one round.  A packed
        pair arrives as two job_descriptions and runs CO-RESIDENT on the
        same GPU — HIP's hardware scheduler time-slices them (the
        reference used CUDA MPS percentages for this).
        """
        gpu_id = self._gpu_queue.get()
        results = []
        try:
            threads = []
            out = {}

            def run_one(job):
                command = self._construct_command(job, gpu_id, worker_id)
                self.launch_job(job, command, worker_id, round_id, gpu_id)
                out[job["job_id"]] = self._get_steps_and_execution_time(
                    job["job_id"], worker_id, round_id
                )

            for job in job_descriptions:
                t = threading.Thread(target=run_one, args=(job,))
                t.start()
                threads.append(t)
            for t in threads:
                t.join()
            for job in job_descriptions:
                steps, duration, log = out.get(job["job_id"], (0, 0.0, ""))
                results.append((job["job_id"], steps, duration, log))
        finally:
            self._gpu_queue.put(gpu_id)
        self._worker_rpc_client.notify_scheduler(worker_id, results)

    def dispatch_jobs(self, job_descriptions, worker_id, round_id):
        self._pool.submit(
            self._safe_dispatch, job_descriptions, worker_id, round_id
        )

    def _safe_dispatch(self, job_descriptions, worker_id, round_id):
        try:
            self._dispatch_jobs_helper(job_descriptions, worker_id, round_id)
        except Exception:
            logger.exception(
                "dispatch of jobs %s failed",
                [j.get("job_id") for j in job_descriptions],
            )
            try:
                self._worker_rpc_client.notify_scheduler(
                    worker_id,
                    [(j["job_id"], 0, 0.0, "") for j in job_descriptions],
                )
            except Exception:
                logger.exception("failed to notify scheduler of failure")

    def kill_job(self, job_id):
        with self._lock:
            proc = self._procs.get(job_id)
            runner = self._job_runners.get(job_id)
            if proc is not None or runner is not None:
                self._killed.add(job_id)
        if runner is not None:
            # SIGTERM the runner: the workload loop's graceful handler
            # checkpoints and returns (runner survives); if the job is
            # hung the runner gets SIGKILLed and replaced
            logger.info("killing warm job %s (runner pid %s)",
                        job_id, runner.proc.pid)
            runner.terminate()
            deadline = time.time() + 10
            while time.time() < deadline:
                with self._lock:
                    if job_id not in self._job_runners:
                        return
                time.sleep(0.25)
            logger.warning("job %s did not stop in 10 s; killing runner",
                           job_id)
            runner.kill()
            return
        if proc is None:
            logger.info("kill_job(%s): no running process", job_id)
            return
        logger.info("killing job %s (pid %s)", job_id, proc.pid)
        try:
            os.killpg(os.getpgid(proc.pid), signal.SIGTERM)
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                os.killpg(os.getpgid(proc.pid), signal.SIGKILL)
        except ProcessLookupError:
            pass

    def reset(self):
        with self._lock:
            job_ids = list(self._procs.keys()) + list(self._job_runners.keys())
        for job_id in job_ids:
            self.kill_job(job_id)

    def shutdown(self):
        self.reset()
        if self._runner_pool is not None:
            self._runner_pool.shutdown()
        self._pool.shutdown(wait=False)
