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


def plot_load_sweep(jsonl_paths, out_path):
    """Steady-state avg JCT vs input load from run_sweep jsonl rows."""
    import collections
    import json

    rows = []
    for p in jsonl_paths:
        with open(p) as f:
            rows += [json.loads(line) for line in f]
    agg = collections.defaultdict(list)
    for r in rows:
        if r.get("status") == "ok":
            agg[(r["policy"], r["jobs_per_hr"])].append(
                r["avg_jct_s"] / 3600.0
            )
    pols = sorted({p for p, _ in agg})
    loads = sorted({l for _, l in agg})
    fig, ax = plt.subplots(figsize=(7, 4.5))
    for pol in pols:
        ys = [np.mean(agg[(pol, l)]) for l in loads if (pol, l) in agg]
        xs = [l for l in loads if (pol, l) in agg]
        kw = ({"linewidth": 2.5, "marker": "o"} if pol == "shockwave"
              else {"linewidth": 1.2, "marker": ".", "alpha": 0.8})
        ax.plot(xs, ys, label=pol, **kw)
    ax.set_xlabel("input load (jobs/hr)")
    ax.set_ylabel("steady-state avg JCT (h)")
    ax.grid(alpha=0.3)
    ax.legend(fontsize=8)
    fig.tight_layout()
    fig.savefig(out_path, dpi=130)
    plt.close(fig)


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--results_dir", required=True)
    p.add_argument("--out_dir", default=None)
    p.add_argument("--load_sweep_jsonl", nargs="*", default=None,
                   help="run_sweep result files -> load_sweep.png")
    args = p.parse_args()
    out_dir = args.out_dir or args.results_dir
    os.makedirs(out_dir, exist_ok=True)

    if args.load_sweep_jsonl:
        plot_load_sweep(args.load_sweep_jsonl,
                        os.path.join(out_dir, "load_sweep.png"))

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
