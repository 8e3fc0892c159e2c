#!/usr/bin/env python3
"""Generate a synthetic TACC-style trace (reference:
scripts/utils/generate_trace.py)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from shockwave_amd.core import generator, trace
from shockwave_amd.core.throughputs import read_throughputs
from shockwave_amd.core.trace import canonical_worker_type


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--num_jobs", type=int, required=True)
    p.add_argument("--lam", type=float, default=60.0, help="mean interarrival (s)")
    p.add_argument("--min_duration", type=float, default=0.2, help="hours")
    p.add_argument("--max_duration", type=float, default=4.0, help="hours")
    p.add_argument("--num_durations", type=int, default=10)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--throughputs_file", default="traces/mi355x_throughputs.json")
    p.add_argument("--mode_mix", default="0,0.5,0.5",
                   help="static,accordion,gns fractions")
    p.add_argument("--scale_factor_mix", default="0.6,0.3,0.09,0.01")
    p.add_argument("--single_gpu", action="store_true")
    p.add_argument("--generate_multi_priority_jobs", action="store_true",
                   help="~20%% of jobs get priority weight 5")
    p.add_argument("--generate_SLOs", action="store_true",
                   help="assign SLO = {1.2,2,10} x ideal duration")
    p.add_argument("-o", "--output_file", required=True)
    args = p.parse_args()

    throughputs = read_throughputs(args.throughputs_file)
    wt = canonical_worker_type(throughputs)
    jobs, arrivals = generator.generate_trace(
        throughputs,
        wt,
        args.num_jobs,
        lam_s=args.lam,
        min_duration_s=args.min_duration * 3600,
        max_duration_s=args.max_duration * 3600,
        num_durations=args.num_durations,
        scale_factor_mix=[float(x) for x in args.scale_factor_mix.split(",")],
        mode_mix=[float(x) for x in args.mode_mix.split(",")],
        seed=args.seed,
        multi_gpu=not args.single_gpu,
        multi_priority=args.generate_multi_priority_jobs,
        generate_slos=args.generate_SLOs,
    )
    trace.write_trace(jobs, arrivals, args.output_file)
    modes = [j.mode for j in jobs]
    sfs = [j.scale_factor for j in jobs]
    print(f"wrote {len(jobs)} jobs -> {args.output_file}")
    print("modes:", {m: modes.count(m) for m in set(modes)})
    print("scale factors:", {s: sfs.count(s) for s in set(sfs)})
    print("durations (h):", sorted(round(j.duration/3600, 2) for j in jobs)[:10], "...")


if __name__ == "__main__":
    main()
