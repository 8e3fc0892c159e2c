#!/usr/bin/env python3
"""Aggregate per-policy results into the headline comparison table.

Reference: reproduce/aggregate_result.py (unfair fraction threshold
rho > 1.05, :24).
"""

import argparse
import glob
import os
import pickle
import sys

FTF_FAIRNESS_THRESHOLD = 1.05


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--results_dir", required=True)
    p.add_argument("--markdown", action="store_true")
    args = p.parse_args()

    rows = []
    for path in sorted(glob.glob(os.path.join(args.results_dir, "*.pickle"))):
        with open(path, "rb") as f:
            r = pickle.load(f)
        rhos = r.get("ftf_rho_list", [])
        unfair = (
            100.0 * sum(1 for x in rhos if x > FTF_FAIRNESS_THRESHOLD) / len(rhos)
            if rhos else 0.0
        )
        rows.append(
            (
                r["policy"],
                r.get("makespan_h", r.get("makespan_s", 0) / 3600.0),
                r.get("avg_jct_h", (r.get("avg_jct_s") or 0) / 3600.0),
                max(rhos) if rhos else float("nan"),
                unfair,
                r.get("cluster_util", float("nan")),
            )
        )
    rows.sort(key=lambda x: x[1])
    sep = "|" if args.markdown else ""
    hdr = f"{sep}{'policy':26s}{sep}{'makespan(h)':>12s}{sep}{'avg JCT(h)':>11s}{sep}{'worst rho':>10s}{sep}{'unfair %':>9s}{sep}{'util':>6s}{sep}"
    print(hdr)
    if args.markdown:
        print("|" + "|".join(["---"] * 6) + "|")
    for r in rows:
        print(
            f"{sep}{r[0]:26s}{sep}{r[1]:12.2f}{sep}{r[2]:11.2f}{sep}{r[3]:10.2f}"
            f"{sep}{r[4]:9.1f}{sep}{r[5]:6.2f}{sep}"
        )


if __name__ == "__main__":
    main()
