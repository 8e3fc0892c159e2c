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
    from shockwave_amd.workloads import families as fam_mod

    fn_name, make_args = FAMILY_ARGS[family]
    fn = getattr(fam_mod, fn_name)
    argv = make_args(bs) + [STEPS_ARG[family], str(warmup)]
    # warmup run (algo selection, allocator) — untimed
    fn(argv)
    t0 = time.time()
    argv = make_args(bs) + [STEPS_ARG[family], str(steps)]
    done = fn(argv)
    elapsed = time.time() - t0
    return done / elapsed


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--families", nargs="*", default=list(FAMILY_ARGS))
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--warmup", type=int, default=15)
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
