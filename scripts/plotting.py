#!/usr/bin/env python3
"""Result plotting: JCT/FTF CDFs, makespan bars, per-round schedule Gantt.

Rebuild of the reference's scheduler/plotting.py:1-643 over this repo's
results pickles (scripts/simulate.py / run_physical.py output).
"""

import argparse
import glob
import os
import pickle

import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt
import numpy as np


def load_results(results_dir):
    out = {}
    for path in sorted(glob.glob(os.path.join(results_dir, "*_simulation.pickle"))) + \
            sorted(glob.glob(os.path.join(results_dir, "*_physical.pickle"))):
        with open(path, "rb") as f:
            r = pickle.load(f)
        out[r["policy"]] = r
    return out


def plot_cdf(results, key, xlabel, out_path, scale=1 / 3600.0):
    plt.figure(figsize=(5, 3.5))
    for policy, r in results.items():
        vals = sorted(np.array(r[key]) * scale)
        if not len(vals):
            continue
        ys = np.arange(1, len(vals) + 1) / len(vals)
        plt.plot(vals, ys, label=policy)
    plt.xlabel(xlabel)
    plt.ylabel("CDF")
    plt.legend(fontsize=7)
    plt.tight_layout()
    plt.savefig(out_path, dpi=150)
    plt.close()


def plot_makespan_bars(results, out_path):
    plt.figure(figsize=(6, 3.5))
    policies = list(results)
    makespans = [results[p]["makespan_h"] for p in policies]
    plt.bar(policies, makespans)
    plt.ylabel("Makespan (h)")
    plt.xticks(rotation=30, ha="right", fontsize=7)
    plt.tight_layout()
    plt.savefig(out_path, dpi=150)
    plt.close()


def plot_per_round_schedule(result, out_path, max_rounds=400):
    """Gantt: one row per job, colored spans for scheduled rounds
    (reference plot_per_round_schedule, plotting.py:267)."""
    sched = result.get("per_round_schedule", [])[:max_rounds]
    if not sched:
        return
    jobs = sorted({j for rnd in sched for j in rnd})
    job_idx = {j: i for i, j in enumerate(jobs)}
    plt.figure(figsize=(10, max(3, len(jobs) * 0.12)))
    for r, rnd in enumerate(sched):
        for j, workers in rnd.items():
            plt.barh(job_idx[j], 1, left=r, height=0.8,
                     color=plt.cm.tab20(job_idx[j] % 20))
    plt.xlabel("Round")
    plt.ylabel("Job")
    plt.tight_layout()
    plt.savefig(out_path, dpi=150)
    plt.close()


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--results_dir", required=True)
    p.add_argument("--out_dir", default=None)
    args = p.parse_args()
    out_dir = args.out_dir or args.results_dir
    os.makedirs(out_dir, exist_ok=True)

    results = load_results(args.results_dir)
    if not results:
        print("no results found")
        return
    plot_cdf(results, "jct_list", "JCT (h)",
             os.path.join(out_dir, "jct_cdf.png"))
    plot_cdf(results, "ftf_rho_list", "Finish-time fairness rho",
             os.path.join(out_dir, "ftf_cdf.png"), scale=1.0)
    plot_makespan_bars(results, os.path.join(out_dir, "makespan.png"))
    for policy, r in results.items():
        plot_per_round_schedule(
            r, os.path.join(out_dir, f"schedule_{policy}.png")
        )
    print(f"plots -> {out_dir}")


if __name__ == "__main__":
    main()
