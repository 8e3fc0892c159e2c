#!/usr/bin/env python3
"""Simulation-vs-physical fidelity analysis.

Reference: reproduce/analyze_fidelity.py:20-57 — compares makespan, avg
JCT and unfair fraction between a policy's simulation and physical result
pickles; the Table-3 claim is a sim/physical gap of a few percent.
"""

import argparse
import pickle


def summarize(r):
    rhos = r.get("ftf_rho_list", [])
    unfair = (
        100.0 * sum(1 for x in rhos if x > 1.05) / len(rhos) if rhos else 0.0
    )
    makespan_h = r.get("makespan_h", r.get("makespan_s", 0) / 3600.0)
    jct_h = r.get("avg_jct_h", (r.get("avg_jct_s") or 0) / 3600.0)
    return makespan_h, jct_h, unfair


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--simulation", required=True)
    p.add_argument("--physical", required=True)
    args = p.parse_args()
    with open(args.simulation, "rb") as f:
        sim = pickle.load(f)
    with open(args.physical, "rb") as f:
        phys = pickle.load(f)

    s = summarize(sim)
    ph = summarize(phys)
    names = ["makespan (h)", "avg JCT (h)", "unfair %"]
    print(f"{'metric':14s} {'sim':>10s} {'physical':>10s} {'gap %':>8s}")
    for name, a, b in zip(names, s, ph):
        gap = 100.0 * abs(a - b) / b if b else float("nan")
        print(f"{name:14s} {a:10.3f} {b:10.3f} {gap:8.1f}")


if __name__ == "__main__":
    main()
