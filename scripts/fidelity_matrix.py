#!/usr/bin/env python3
"""One-command sim-vs-physical fidelity matrix.

Automates the pipeline behind profiles/FIDELITY.md's cross-policy
table: given a physical results pickle (scripts/run_physical.py), it
calibrates the oracle + startup table (scripts/calibrate_sim.py logic),
runs the fidelity simulator under the three calibration settings, and
prints the gap row:

* plain — sim plans AND clocks with the same oracle the physical
  scheduler was given (the reference's own fidelity methodology),
* hot   — oracle rates replaced by the hot rates observed on that box,
* split — policy believes the plain oracle (+ online EMA calibration),
  world progresses at the observed hot rates.

Usage:
  python scripts/fidelity_matrix.py --physical profiles/fid_phys_las_r2.pickle
  # policy, trace and round duration are read from the pickle when
  # present; override with --policy/--trace/--round_duration
"""

import argparse
import json
import os
import pickle
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--physical", required=True)
    ap.add_argument("--trace", default="traces/fidelity_5job.trace")
    ap.add_argument("--oracle", default="traces/mi355x_throughputs.json")
    ap.add_argument("--policy", default=None,
                    help="default: the pickle's policy")
    ap.add_argument("--round_duration", type=float, default=60)
    ap.add_argument("--preemption_overhead", type=float, default=5)
    ap.add_argument("--warm_overhead", type=float, default=0.5)
    args = ap.parse_args()

    from calibrate_sim import calibrate
    from simulate import run_simulation

    phys = pickle.load(open(args.physical, "rb"))
    policy = args.policy or phys.get("policy")
    if not phys.get("job_types"):
        phys["job_types"] = {
            str(i): line.split("\t")[0]
            for i, line in enumerate(open(args.trace))
            if line.strip()
        }
    hot, startup = calibrate(phys)
    print(f"policy={policy} physical makespan={phys['makespan_s']:.1f}s "
          f"avg_jct={phys['avg_jct_s']:.1f}s")
    print(f"calibrated hot rates: { {k: round(v,1) for k,v in hot.items()} }")

    with tempfile.TemporaryDirectory() as td:
        # hot oracle = plain oracle with observed-rate overrides
        import re

        oracle = json.load(open(args.oracle))
        tables = [v for v in oracle.values() if isinstance(v, dict)] or [
            oracle
        ]
        old_iso = {}
        for t in tables:
            for jt in hot:
                e1 = t.get(f"('{jt}', 1)")
                if e1 and e1.get("null"):
                    old_iso.setdefault(jt, e1["null"])
        for t in tables:
            for key, entry in t.items():
                m = re.match(r"^\('(?P<jt>.+)', (?P<sf>\d+)\)$", key)
                if not m or "null" not in entry:
                    continue
                jt, sf = m.group("jt"), int(m.group("sf"))
                if jt in hot:
                    if sf == 1:
                        entry["null"] = hot[jt]
                    elif old_iso.get(jt):
                        entry["null"] *= hot[jt] / old_iso[jt]
        hot_path = os.path.join(td, "hot_oracle.json")
        json.dump(oracle, open(hot_path, "w"))

        rows = []
        for label, belief, world in [
            ("plain", args.oracle, None),
            ("hot", hot_path, None),
            ("split", args.oracle, hot_path),
        ]:
            r = run_simulation(
                args.trace, belief, policy, num_gpus=1,
                time_per_iteration=args.round_duration,
                preemption_overhead_s=args.preemption_overhead,
                warm_overhead_s=args.warm_overhead,
                midround_staleness=True, fixed_rounds=True,
                startup_table=startup,
                world_throughputs_file=world,
            )
            gm = 100 * (r["makespan_s"] - phys["makespan_s"]) / phys[
                "makespan_s"
            ]
            gj = 100 * (r["avg_jct_s"] - phys["avg_jct_s"]) / phys[
                "avg_jct_s"
            ]
            rows.append((label, r, gm, gj))
            print(f"{label:6s} sim {r['makespan_s']:7.1f}/"
                  f"{r['avg_jct_s']:7.1f}s  gaps {gm:+5.1f}%/{gj:+5.1f}%")

    best = min(rows, key=lambda x: abs(x[2]) + abs(x[3]))
    print(f"| {policy} | {phys['makespan_s']:.1f} / "
          f"{phys['avg_jct_s']:.1f} s | "
          + " | ".join(f"{gm:+.1f}% / {gj:+.1f}%" for _, _, gm, gj in rows)
          + f" |   (best: {best[0]})")


if __name__ == "__main__":
    main()
