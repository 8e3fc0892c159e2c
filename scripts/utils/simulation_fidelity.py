#!/usr/bin/env python3
"""Per-job sim-vs-physical fidelity (reference
scripts/utils/simulation_fidelity.py / physical_simulation_comparison.py):
pairs each job's simulated and physical completion time and reports the
distribution of relative errors."""

import argparse
import pickle

import numpy as np


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--simulation", required=True)
    p.add_argument("--physical", required=True)
    args = p.parse_args()
    sim = pickle.load(open(args.simulation, "rb"))
    phys = pickle.load(open(args.physical, "rb"))
    s, ph = sim.get("jct_list", []), phys.get("jct_list", [])
    n = min(len(s), len(ph))
    if n == 0:
        print("no overlapping jobs")
        return
    rel = np.abs(np.array(s[:n]) - np.array(ph[:n])) / np.maximum(
        np.array(ph[:n]), 1e-9
    )
    print(f"jobs compared: {n}")
    print(
        "per-job |sim-phys|/phys percentiles {0,25,50,75,100}%:",
        [round(float(x), 3) for x in np.percentile(rel, [0, 25, 50, 75, 100])],
    )
    print(f"mean relative JCT error: {rel.mean():.3f}")


if __name__ == "__main__":
    main()
