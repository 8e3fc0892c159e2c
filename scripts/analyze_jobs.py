#!/usr/bin/env python3
"""Per-job timeline decoder.

Joins a physical run's iterator event logs, per-round schedules and
throughput timelines (saved by scripts/run_physical.py in the results
pickle) into per-job, per-round reports:

* per dispatch: init time, lease events, steps, loop duration,
  steps/s, checkpoint save/load spans,
* per job: total steps vs trace, per-round throughput vs the oracle,
  time attributed to startup / training / idle,
* cluster: per-round occupancy gaps (when no job was training).

With --simulation, also prints a per-job sim-vs-physical comparison
(the reference's parse_simulation_trace.py, at ~1/20 the size).

Usage:
  python scripts/analyze_jobs.py --physical fid_phys.pickle \
      [--simulation results/fid/*.pickle] [--oracle traces/mi355x_throughputs.json]
"""

import argparse
import json
import pickle
import re
import sys
from collections import defaultdict
from datetime import datetime

LOG_RE = re.compile(
    r"^\[(?P<ts>[^\]]+)\] \[(?P<event>[^\]]+)\] \[(?P<status>[^\]]+)\] ?(?P<msg>.*)$"
)


def parse_timeline(lines):
    """One worker-slot's iterator log lines -> list of event dicts."""
    events = []
    for line in lines:
        m = LOG_RE.match(line)
        if not m:
            continue
        try:
            ts = datetime.strptime(m.group("ts"), "%Y-%m-%d %H:%M:%S")
        except ValueError:
            continue
        events.append(
            {
                "t": ts,
                "event": m.group("event"),
                "status": m.group("status"),
                "msg": m.group("msg"),
            }
        )
    return events


def split_dispatches(events):
    """Split a slot's event stream into dispatches.  Each lease starts by
    logging PROGRESS STEPS 0; a drop in the cumulative step counter marks
    the next dispatch."""
    dispatches = []
    cur = None
    last_steps = -1
    for ev in events:
        is_boundary = ev["event"] == "INIT"
        if ev["event"] == "PROGRESS" and ev["status"] == "STEPS":
            try:
                v = int(float(ev["msg"]))
            except ValueError:
                v = last_steps
            if v < last_steps:
                is_boundary = True
            last_steps = v
        if cur is None or is_boundary:
            if cur:
                dispatches.append(cur)
            cur = {"events": []}
        cur["events"].append(ev)
    if cur:
        dispatches.append(cur)
    return dispatches


def summarize_dispatch(d):
    evs = d["events"]
    out = {
        "t_start": evs[0]["t"],
        "t_end": evs[-1]["t"],
        "steps": 0,
        "duration": 0.0,
        "lease_events": [],
    }
    for ev in evs:
        if ev["event"] == "PROGRESS":
            try:
                if ev["status"] == "STEPS":
                    out["steps"] = max(out["steps"], int(float(ev["msg"])))
                elif ev["status"] == "DURATION":
                    out["duration"] = max(out["duration"], float(ev["msg"]))
            except ValueError:
                pass
        elif ev["event"] == "LEASE":
            out["lease_events"].append((ev["t"], ev["status"]))
    wall = (out["t_end"] - out["t_start"]).total_seconds()
    out["wall"] = wall
    out["overhead"] = max(0.0, wall - out["duration"])
    out["steps_per_s"] = (
        out["steps"] / out["duration"] if out["duration"] > 0 else 0.0
    )
    return out


def analyze(results, oracle=None):
    report = {}
    timelines = results.get("job_timelines", {})
    totals = results.get("job_total_steps", {})
    for jid, slots in sorted(timelines.items(), key=lambda kv: kv[0]):
        job_report = {"dispatches": [], "job_id": jid}
        for slot_lines in slots:
            for d in split_dispatches(parse_timeline(slot_lines)):
                s = summarize_dispatch(d)
                if s["steps"] or s["duration"]:
                    job_report["dispatches"].append(s)
        job_report["dispatches"].sort(key=lambda s: s["t_start"])
        job_report["n_dispatches"] = len(job_report["dispatches"])
        job_report["total_steps"] = sum(
            s["steps"] for s in job_report["dispatches"]
        )
        job_report["train_time"] = sum(
            s["duration"] for s in job_report["dispatches"]
        )
        job_report["overhead_time"] = sum(
            s["overhead"] for s in job_report["dispatches"]
        )
        rates = [
            s["steps_per_s"]
            for s in job_report["dispatches"]
            if s["steps"] > 10
        ]
        job_report["median_steps_per_s"] = (
            sorted(rates)[len(rates) // 2] if rates else 0.0
        )
        job_report["trace_total_steps"] = totals.get(jid)
        report[jid] = job_report
    return report


def print_report(report, results, oracle, sim=None):
    print("=" * 78)
    print(
        f"policy={results.get('policy')} makespan={results.get('makespan_s', 0):.1f}s "
        f"avg_jct={results.get('avg_jct_s', 0):.1f}s util={results.get('cluster_util')}"
    )
    print("=" * 78)
    tput_tl = results.get("throughput_timeline", {})
    for jid, r in sorted(report.items(), key=lambda kv: int(kv[0])):
        line = (
            f"job {jid}: {r['n_dispatches']} dispatches, "
            f"{r['total_steps']} steps, train {r['train_time']:.1f}s, "
            f"overhead {r['overhead_time']:.1f}s, "
            f"median {r['median_steps_per_s']:.2f} steps/s"
        )
        tl = tput_tl.get(int(jid))
        if tl:
            jt = [f"r{k}:{v[0]:.1f}" for k, v in sorted(tl.items())]
            line += f" | sched-observed steps/s {' '.join(jt[:6])}"
        print(line)
        for i, d in enumerate(r["dispatches"]):
            lease = ",".join(s for _, s in d["lease_events"][:4])
            print(
                f"    [{i}] {d['t_start'].strftime('%H:%M:%S')} wall "
                f"{d['wall']:.1f}s loop {d['duration']:.1f}s steps "
                f"{d['steps']} ({d['steps_per_s']:.2f}/s) overhead "
                f"{d['overhead']:.1f}s lease[{lease}]"
            )
    if sim is not None:
        print("-" * 78)
        print("sim vs physical per job (JCT seconds):")
        sim_jcts = sim.get("jct_list", [])
        phys_jcts = results.get("jct_list", [])
        for i, (s, p) in enumerate(zip(sim_jcts, phys_jcts)):
            gap = (s - p) / p * 100 if p else 0.0
            print(f"  job {i}: sim {s:.1f}  phys {p:.1f}  gap {gap:+.1f}%")


def print_sim_report(sim):
    """Standalone decoder for a simulation results pickle (the
    reference's parse_simulation_trace.py role for SIM output): per-job
    JCT/FTF and scheduled-round counts from the per-round schedule."""
    print("=" * 78)
    print(
        f"SIM policy={sim.get('policy')} "
        f"makespan={sim.get('makespan_s', 0):.1f}s "
        f"avg_jct={sim.get('avg_jct_s', 0):.1f}s "
        f"worst_rho={sim.get('worst_ftf_rho')}"
    )
    print("=" * 78)
    prs = sim.get("per_round_schedule") or []
    sched_rounds = defaultdict(list)
    for i, rd in enumerate(prs):
        for jid in rd:
            sched_rounds[str(jid)].append(i)
    jcts = sim.get("jct_list", [])
    rhos = sim.get("ftf_rho_list", [])
    for i, jct in enumerate(jcts):
        rounds = sched_rounds.get(str(i), [])
        rho = rhos[i] if i < len(rhos) else None
        rtxt = (
            f"rounds {rounds[0]}-{rounds[-1]} ({len(rounds)} scheduled)"
            if rounds else "never scheduled"
        )
        print(f"job {i}: jct {jct:8.1f}s  rho {rho}  {rtxt}")
    if prs:
        width = max((len(str(j)) for rd in prs for j in rd), default=1)
        print("-" * 78)
        print("round -> jobs:")
        for i, rd in enumerate(prs):
            jobs = " ".join(str(j).rjust(width) for j in sorted(rd))
            print(f"  r{i:03d}: {jobs if jobs else '(idle)'}")


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--physical", default=None)
    ap.add_argument("--simulation", default=None)
    ap.add_argument("--oracle", default=None)
    ap.add_argument("--json_out", default=None)
    args = ap.parse_args()

    if args.physical is None:
        if args.simulation is None:
            ap.error("need --physical and/or --simulation")
        print_sim_report(pickle.load(open(args.simulation, "rb")))
        return

    results = pickle.load(open(args.physical, "rb"))
    oracle = json.load(open(args.oracle)) if args.oracle else None
    sim = pickle.load(open(args.simulation, "rb")) if args.simulation else None
    report = analyze(results, oracle)
    print_report(report, results, oracle, sim)
    if args.json_out:
        ser = {
            jid: {
                k: v
                for k, v in r.items()
                if k != "dispatches"
            }
            for jid, r in report.items()
        }
        with open(args.json_out, "w") as f:
            json.dump(ser, f, indent=1, default=str)
        print(f"wrote {args.json_out}")


if __name__ == "__main__":
    main()
