#!/usr/bin/env python3
"""Measure sf=2 scaling BOUNDS by time-slicing 2 DDP ranks on one MI355X.

gpurun leases expose a single GPU and RCCL refuses two ranks on one
device (profiles/MULTIGPU_PROBE.md), so true xGMI scaling cannot be
measured here; what CAN be measured — and what VERDICT r1 item 4 asks
for as the fallback — is a conservative per-(model,bs) LOWER BOUND on
the sf=2 data-parallel efficiency the oracle assumes
(scripts/make_throughputs.py XGMI_EFF):

* t1      = per-step time of the eager sf=1 training loop (one process)
* t2r     = per-rank step time with TWO DDP ranks time-slicing the GPU
            (gloo backend with host-staged comm — strictly slower than
            RCCL over xGMI for every message)
* oh      = max(0, t2r - 2*t1)   # everything beyond pure 2x time-slicing:
            bucket all-reduce + sync + DDP bookkeeping, all through the
            slow transport, so an UPPER bound on real exposed comm
* e_lb(2) = t1 / (t1 + oh)       # lower bound on real 2-GPU efficiency

Both runs are eager (SWQ_GRAPHS=0): gloo collectives are not hipGraph-
capturable, and the bound must compare like with like.  The reference
measured its sf>1 oracle entries on a 32-V100 cluster
(tacc_throughputs.json keys ``('<type>', 2|4|8)``); this is the closest
single-GPU-lease equivalent, recorded as bounds, not point estimates.

Writes profiles/measured_sf2_timesliced.json; profiles/SF2_BOUNDS.md is
generated from it by --report.
"""

import argparse
import json
import os
import socket
import subprocess
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from measure_throughput import FAMILY_ARGS, STEPS_ARG  # noqa: E402

# (family, bs, steps, warmup) — the trace's most common configs
CONFIGS = [
    ("ResNet-18", 16, 300, 60),
    ("ResNet-18", 64, 150, 40),
    ("ResNet-50", 32, 80, 20),
    ("Transformer", 64, 120, 30),
    ("LM", 80, 200, 40),
]


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def run_child(family, bs, steps, warmup, world, rank, port, tmp):
    """Launch one rank as a subprocess; returns the Popen."""
    env = dict(os.environ)
    env["SWQ_GRAPHS"] = "0"          # eager on both sides of the bound
    env["SWQ_SESSION_CACHE"] = "0"   # fresh process = fresh session anyway
    if world > 1:
        env["SWQ_DIST_BACKEND"] = "gloo"   # RCCL refuses 2 ranks/1 GPU
        env["SWQ_RENDEZVOUS_TIMEOUT"] = "300"
    cmd = [
        sys.executable, os.path.abspath(__file__), "--child",
        "--family", family, "--bs", str(bs), "--steps", str(steps),
        "--warmup", str(warmup), "--world", str(world), "--rank", str(rank),
        "--port", str(port), "--tmp", tmp,
    ]
    return subprocess.Popen(
        cmd, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True, cwd=os.path.join(os.path.dirname(__file__), ".."),
    )


def child_main(args):
    import faulthandler

    faulthandler.dump_traceback_later(60, repeat=True)
    from shockwave_amd.runtime.dispatcher import LOG_LINE_RE
    from shockwave_amd.runtime.lease_iterator import NullLeaseClient
    from shockwave_amd.workloads import families as fam_mod

    fn_name, make_args = FAMILY_ARGS[args.family]
    fn = getattr(fam_mod, fn_name)
    world_args = []
    if args.world > 1:
        world_args = [
            "--world_size", str(args.world), "--rank", str(args.rank),
            "--master_addr", "127.0.0.1", "--master_port", str(args.port),
        ]
    # warmup (MIOpen find, allocator, rendezvous) — untimed by the
    # iterator log of the SECOND run; world>1 warmup also runs world>1 so
    # both ranks' communicators and buckets are hot
    fn(make_args(args.bs) + [STEPS_ARG[args.family], str(args.warmup)]
       + world_args, client=NullLeaseClient())
    if args.world > 1:
        # fresh port for the timed run: the warmup run's TCPStore on
        # `port` may still be tearing down, and a rank that reconnects to
        # the dying store hangs rendezvous
        world_args[world_args.index(str(args.port))] = str(args.port + 1)
    ckpt_dir = os.path.join(args.tmp, f"rank{args.rank}")
    os.makedirs(ckpt_dir, exist_ok=True)
    argv = make_args(args.bs) + [
        STEPS_ARG[args.family], str(args.steps),
        "--checkpoint_dir", ckpt_dir, "--enable_gavel_iterator",
    ] + world_args
    fn(argv, client=NullLeaseClient())
    log_path = os.path.join(
        ckpt_dir, ".gavel", "round=0", f"worker={args.rank}.log"
    )
    if not os.path.exists(log_path):
        log_path = os.path.join(ckpt_dir, ".gavel", "round=0", "worker=0.log")
    duration = steps_logged = None
    with open(log_path) as f:
        for line in f:
            m = LOG_LINE_RE.match(line)
            if m and m.group("event") == "PROGRESS":
                if m.group("status") == "STEPS":
                    steps_logged = int(float(m.group("msg")))
                elif m.group("status") == "DURATION":
                    duration = float(m.group("msg"))
    faulthandler.cancel_dump_traceback_later()
    print(json.dumps({"rank": args.rank, "steps": steps_logged,
                      "duration": duration}), flush=True)


def measure_config(family, bs, steps, warmup, tmp_root):
    import shutil
    import tempfile

    out = {"family": family, "bs": bs, "steps": steps}
    for world in (1, 2):
        tmp = tempfile.mkdtemp(prefix=f"swq_sf2_{world}_", dir=tmp_root)
        port = _free_port()
        procs = [
            run_child(family, bs, steps, warmup, world, r, port, tmp)
            for r in range(world)
        ]
        durations = []
        logs = []
        for p in procs:
            stdout, _ = p.communicate(timeout=1800)
            logs.append(stdout)
            if p.returncode != 0:
                print(f"  world={world} child failed:\n{stdout[-2000:]}")
                shutil.rmtree(tmp, ignore_errors=True)
                return None
            rec = json.loads(stdout.strip().splitlines()[-1])
            assert rec["duration"], f"no duration (world={world})"
            durations.append(rec["duration"])
        shutil.rmtree(tmp, ignore_errors=True)
        t_step = max(durations) / steps
        out[f"world{world}_s_per_step"] = t_step
        print(f"  world={world}: {t_step*1000:.2f} ms/step "
              f"({steps/max(durations):.2f} steps/s/rank)", flush=True)
    t1, t2r = out["world1_s_per_step"], out["world2_s_per_step"]
    oh = max(0.0, t2r - 2.0 * t1)
    out["overhead_s_ub"] = oh
    out["e2_lower_bound"] = t1 / (t1 + oh)
    return out


XGMI_LINK_GBPS = 153e9  # per-link xGMI bandwidth (MI355X: 7 links/GPU)


def report(results_path, md_path,
           allreduce_path="profiles/allreduce_gloo.json"):
    data = json.load(open(results_path))
    comm = {}
    if os.path.exists(allreduce_path):
        comm = json.load(open(allreduce_path))
    lines = [
        "# sf=2 data-parallel efficiency bounds (time-sliced, 1x MI355X)",
        "",
        "Two DDP ranks share ONE GPU (gloo host-staged comm; RCCL refuses",
        "duplicate devices — profiles/MULTIGPU_PROBE.md).  `oh` bounds the",
        "real exposed comm of a 2-GPU xGMI run from above, so `e_lb` bounds",
        "the true sf=2 efficiency from below.  `t_gloo` is the SAME bucket",
        "layout all-reduced with no compute (scripts/bench_allreduce.py);",
        "`e_est` replaces the measured gloo transport inside `oh` with the",
        "xGMI ring model (bytes x 2(n-1)/n / 153 GB/s per link), keeping",
        "the measured sync/bookkeeping residual.  Oracle assumption:",
        "XGMI_EFF[2] = 0.96 (scripts/make_throughputs.py).",
        "",
        "| config | t1 (ms/step) | t2 rank (ms) | oh ub (ms) | e(2) lb "
        "| t_gloo (ms) | residual (ms) | t_xgmi model (ms) | e(2) est |",
        "|---|---|---|---|---|---|---|---|",
    ]
    for r in data["configs"]:
        t1 = r["world1_s_per_step"]
        oh = r["overhead_s_ub"]
        row = (
            f"| {r['family']} bs{r['bs']} | {t1*1e3:.2f} "
            f"| {r['world2_s_per_step']*1e3:.2f} "
            f"| {oh*1e3:.2f} | {r['e2_lower_bound']:.3f} "
        )
        c = comm.get(r["family"])
        if c:
            t_gloo = c["allreduce_s"]
            resid = max(0.0, oh - t_gloo)
            t_xgmi = c["total_mb"] * 2**20 / XGMI_LINK_GBPS  # 2(n-1)/n = 1
            e_est = t1 / (t1 + resid + t_xgmi)
            row += (f"| {t_gloo*1e3:.2f} | {resid*1e3:.2f} "
                    f"| {t_xgmi*1e3:.3f} | {e_est:.3f} |")
        else:
            row += "| - | - | - | - |"
        lines.append(row)
    lines += [
        "",
        "Eager path on both sides (`SWQ_GRAPHS=0`): gloo collectives are",
        "not capturable, and the subtraction must compare like with like.",
        "`e_lb` below 0.96 does NOT refute the oracle — host-staged gloo",
        "transport dominates `oh` (see the t_gloo column) and is far",
        "slower than xGMI; `e_est` is the measured-sync/modeled-transport",
        "estimate.  `e_est` also still pays the 2x time-slicing residual,",
        "so the truth lies between `e_est` and 1.0.  The driver's",
        "round-end SCALE run (one rank per GPU over real RCCL/xGMI) is",
        "the point measurement these bounds bracket.",
    ]
    with open(md_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {md_path}")


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--child", action="store_true")
    p.add_argument("--family")
    p.add_argument("--bs", type=int)
    p.add_argument("--steps", type=int)
    p.add_argument("--warmup", type=int)
    p.add_argument("--world", type=int, default=1)
    p.add_argument("--rank", type=int, default=0)
    p.add_argument("--port", type=int, default=29600)
    p.add_argument("--tmp", default="/tmp")
    p.add_argument("--out", default="profiles/measured_sf2_timesliced.json")
    p.add_argument("--report", action="store_true",
                   help="regenerate profiles/SF2_BOUNDS.md from --out")
    args = p.parse_args()
    if args.child:
        child_main(args)
        return
    if args.report:
        report(args.out, "profiles/SF2_BOUNDS.md")
        return

    results = {"configs": [], "host": "mi355x", "transport": "gloo-timesliced"}
    t0 = time.time()
    for family, bs, steps, warmup in CONFIGS:
        print(f"{family} bs{bs}:", flush=True)
        r = measure_config(family, bs, steps, warmup, "/tmp")
        if r is not None:
            results["configs"].append(r)
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(results, f, indent=1)
    print(f"done in {time.time()-t0:.0f}s -> {args.out}")
    report(args.out, "profiles/SF2_BOUNDS.md")


if __name__ == "__main__":
    main()
