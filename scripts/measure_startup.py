#!/usr/bin/env python3
"""Measure per-dispatch startup cost: cold subprocess vs warm runner.

VERDICT r1 weak #1 asked for <10 s (from the measured ~35 s).  This
drives the REAL dispatcher paths:

* cold: ``subprocess`` launch of the trace-compatible shim (r1 behavior),
* warm-first: first dispatch into a fresh WarmRunner (runner spawn +
  torch import + HIP context + model build + MIOpen + capture),
* warm-hit: re-dispatch of the same configuration (session-cache hit —
  checkpoint load into existing tensors, graph replayed).

"startup" here = wall time of the dispatch minus pure step time
(steps x per-step time measured separately), i.e. what a lease pays
before useful work.  Writes gpurun_out/startup.json.
"""

import json
import os
import shutil
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run_cold(ckpt_dir, steps):
    cwd = os.path.join(REPO, "workloads/pytorch/image_classification/cifar10")
    cmd = (
        f"python3 main.py --batch_size 16 --num_steps {steps} "
        f"--checkpoint_dir {ckpt_dir} --enable_gavel_iterator --mode static"
    )
    env = dict(os.environ)
    env.update({"GAVEL_JOB_ID": "100", "GAVEL_WORKER_ID": "0",
                "GAVEL_ROUND_ID": "0"})
    t0 = time.time()
    subprocess.run(cmd, shell=True, cwd=cwd, env=env, check=True,
                   stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
    return time.time() - t0


def main():
    from shockwave_amd.runtime.warm_runner import WarmRunner

    out = {}
    steps = 150
    base = "/tmp/swq_startup"
    shutil.rmtree(base, ignore_errors=True)
    os.makedirs(base, exist_ok=True)

    # pure step time (for subtracting) from a long warm run later; first
    # the r1 cold path, twice (second run has MIOpen find-db warm on disk)
    for label, job in (("cold_first_s", "101"), ("cold_second_s", "102")):
        d = os.path.join(base, f"job_id={job}")
        os.makedirs(d, exist_ok=True)
        out[label] = round(run_cold(d, steps), 2)
        print(f"{label}: {out[label]} s", flush=True)

    t0 = time.time()
    runner = WarmRunner(0, os.path.join(base, "runner.log"))
    out["runner_spawn_s"] = round(time.time() - t0, 2)

    cwd = os.path.join(REPO, "workloads/pytorch/image_classification/cifar10")

    def warm(job, steps):
        d = os.path.join(base, f"job_id={job}")
        os.makedirs(d, exist_ok=True)
        cmd = (
            f"python3 main.py --batch_size 16 --num_steps {steps} "
            f"--checkpoint_dir {d} --enable_gavel_iterator --mode static"
        )
        env = {"GAVEL_JOB_ID": str(job), "GAVEL_WORKER_ID": "0",
               "GAVEL_ROUND_ID": "0"}
        t0 = time.time()
        rc = runner.run(cmd, cwd, env, os.path.join(d, "out.log"))
        if rc != 0:
            try:
                print("JOB LOG TAIL:\n" +
                      open(os.path.join(d, "out.log")).read()[-4000:],
                      flush=True)
            except OSError:
                pass
            raise AssertionError(f"warm job {job} rc={rc}")
        return time.time() - t0

    out["warm_first_s"] = round(warm(111, steps), 2)
    print(f"warm_first_s: {out['warm_first_s']} s", flush=True)
    out["warm_hit_new_job_s"] = round(warm(112, steps), 2)
    print(f"warm_hit_new_job_s: {out['warm_hit_new_job_s']} s", flush=True)
    out["warm_hit_resume_s"] = round(warm(112, steps), 2)
    print(f"warm_hit_resume_s: {out['warm_hit_resume_s']} s", flush=True)

    # per-step time from a longer run on the hot session
    t_long = warm(113, 600)
    per_step = t_long / 600
    out["hot_ms_per_step"] = round(per_step * 1e3, 3)

    # startup = wall - useful step time.  The resume case ran 0 steps
    # (target already reached), so its wall time IS the dispatch cost.
    for k in ("cold_first_s", "cold_second_s", "warm_first_s",
              "warm_hit_new_job_s"):
        out["startup_" + k] = round(out[k] - steps * per_step, 2)
    out["startup_warm_hit_resume_s"] = out["warm_hit_resume_s"]
    runner.shutdown()

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/startup.json", "w") as f:
        json.dump(out, f, indent=2)
    print(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
