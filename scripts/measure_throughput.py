#!/usr/bin/env python3
"""Measure per-job-type isolated throughputs on a live MI355X.

Rebuild of the reference's offline profiler
(scripts/profiling/measure_throughput.py:28-80): runs each job type for a
bounded window and records steps/s, here by driving the unified workload
runner in-process with a wall-clock-bounded lease.

Writes ``profiles/measured_throughputs.json`` with ``"<model>|<bs>"`` keys;
``scripts/make_throughputs.py`` folds these into the oracle file.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

FAMILY_ARGS = {
    "ResNet-18": ("cifar10_main", lambda bs: ["--batch_size", str(bs)]),
    "ResNet-50": ("imagenet_main", lambda bs: ["-b", str(bs)]),
    "Transformer": (
        "translation_main",
        lambda bs: ["-batch_size", str(bs), "-proj_share_weight"],
    ),
    "LM": ("lm_main", lambda bs: ["--batch_size", str(bs)]),
    "Recommendation": ("recommendation_main", lambda bs: ["--batch_size", str(bs)]),
}

FAMILY_BS = {
    "ResNet-18": [16, 32, 64, 128, 256],
    "ResNet-50": [16, 32, 64, 128],
    "Transformer": [16, 32, 64, 128],
    "LM": [5, 10, 20, 40, 80],
    "Recommendation": [512, 1024, 2048, 4096, 8192],
}

STEPS_ARG = {
    "ResNet-18": "--num_steps",
    "ResNet-50": "--num_minibatches",
    "Transformer": "-step",
    "LM": "--steps",
    "Recommendation": "-n",
}


def measure_one(family, bs, steps, warmup):
    """steps/s measured from the lease iterator's own PROGRESS log, which
    times the training loop only (model construction, synthetic-data
    generation and MIOpen algorithm search are excluded: the warmup call
    absorbs them, and the timed call's iterator clock starts at the loop).

    Round 2: the warmup run is long enough (>= SWQ_GRAPH_MIN_STEPS) to
    build the SESSION including the captured hipGraph, so the timed run
    measures the PRODUCTION path — session hit + graph replay — i.e. the
    rate leases actually sustain under the warm-runner dispatcher, not
    the eager rate (oracle used eager numbers)."""
    import shutil
    import tempfile

    from shockwave_amd.runtime.dispatcher import LOG_LINE_RE
    from shockwave_amd.workloads import families as fam_mod

    fn_name, make_args = FAMILY_ARGS[family]
    fn = getattr(fam_mod, fn_name)
    # warmup run (MIOpen find, allocator, L3) — untimed
    fn(make_args(bs) + [STEPS_ARG[family], str(warmup)])

    ckpt_dir = tempfile.mkdtemp(prefix="swq_measure_")
    try:
        argv = make_args(bs) + [
            STEPS_ARG[family], str(steps),
            "--checkpoint_dir", ckpt_dir, "--enable_gavel_iterator",
        ]
        from shockwave_amd.runtime.lease_iterator import NullLeaseClient

        done = fn(argv, client=NullLeaseClient())
        log_path = os.path.join(ckpt_dir, ".gavel", "round=0", "worker=0.log")
        steps_logged, duration = done, None
        with open(log_path) as f:
            for line in f:
                m = LOG_LINE_RE.match(line)
                if m and m.group("event") == "PROGRESS":
                    if m.group("status") == "STEPS":
                        steps_logged = int(float(m.group("msg")))
                    elif m.group("status") == "DURATION":
                        duration = float(m.group("msg"))
        assert duration and duration > 0, "no duration in iterator log"
        return steps_logged / duration
    finally:
        shutil.rmtree(ckpt_dir, ignore_errors=True)


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--families", nargs="*", default=list(FAMILY_ARGS))
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--warmup", type=int, default=120)
    p.add_argument("--out", default="profiles/measured_throughputs.json")
    p.add_argument("--timeout_per_config", type=float, default=180)
    args = p.parse_args()

    measured = {}
    if os.path.exists(args.out):
        measured = json.load(open(args.out))
    for family in args.families:
        for bs in FAMILY_BS[family]:
            key = f"{family}|{bs}"
            try:
                tput = measure_one(family, bs, args.steps, args.warmup)
                measured[key] = tput
                print(f"{key}: {tput:.2f} steps/s", flush=True)
            except Exception as e:
                print(f"{key}: FAILED ({e})", flush=True)
            os.makedirs(os.path.dirname(args.out), exist_ok=True)
            with open(args.out, "w") as f:
                json.dump(measured, f, indent=1)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
