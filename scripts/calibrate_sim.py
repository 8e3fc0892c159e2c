#!/usr/bin/env python3
"""Calibrate the simulator from a physical run's timelines.

Derives, per job type observed in a physical results pickle
(scripts/run_physical.py):

* **hot rate** — the best per-round steps/s the scheduler observed
  (post-capture, session warm).  Written as a measured-throughputs
  overlay and folded into a calibrated oracle JSON.
* **first-dispatch startup** — the first round's wall time minus
  steps/hot_rate: what MIOpen find + model build + hipGraph capture
  cost inside that job type's first lease on the box.

This mirrors on the sim side what the live scheduler does online
(JobMetaData.calibrate_profiled_epoch_duration, core/metadata.py):
replace a-priori rates with observed ones.

Usage:
  python scripts/calibrate_sim.py --physical fid_phys.pickle \
      --oracle traces/mi355x_throughputs.json \
      --out_oracle /tmp/calibrated.json --out_startup /tmp/startup.json
"""

import argparse
import json
import pickle
import re
import sys

sys.path.insert(0, __import__("os").path.join(
    __import__("os").path.dirname(__file__), ".."))

BS_RE = re.compile(r"^(?P<family>.+) \(batch size (?P<bs>\d+)\)$")


def calibrate(results):
    """-> (hot_rates {job_type: steps/s}, startup {job_type: s})."""
    from scripts.analyze_jobs import (  # reuse the decoder
        parse_timeline, split_dispatches, summarize_dispatch,
    )

    # job_id -> job_type from the per-round schedule is not stored;
    # recover job types from the throughput timeline order + trace?  The
    # timelines are keyed by job id; job types come from ftf ordering.
    # run_physical stores throughput_timeline {job_int: {round: (tput,
    # bs)}} — rate only; dispatch summaries give (steps, duration).
    hot, startup = {}, {}
    job_types = results.get("job_types", {})
    for jid, slots in results.get("job_timelines", {}).items():
        dispatches = []
        for slot in slots:
            dispatches += [
                summarize_dispatch(d)
                for d in split_dispatches(parse_timeline(slot))
            ]
        dispatches = [d for d in dispatches if d["steps"] > 0]
        if not dispatches:
            continue
        dispatches.sort(key=lambda d: d["t_start"])
        rates = [d["steps_per_s"] for d in dispatches if d["steps"] > 50]
        if not rates:
            continue
        hot_rate = max(rates)
        first = dispatches[0]
        su = max(0.0, first["wall"] - first["steps"] / hot_rate)
        jt = job_types.get(jid, jid)
        hot[jt] = max(hot.get(jt, 0.0), hot_rate)
        startup[jt] = max(startup.get(jt, 0.0), su)
    return hot, startup


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--physical", required=True)
    ap.add_argument("--oracle", default="traces/mi355x_throughputs.json")
    ap.add_argument("--out_oracle", required=True)
    ap.add_argument("--out_startup", required=True)
    ap.add_argument("--trace", default=None,
                    help="trace file: recover job_id -> job_type by line "
                         "order when the pickle predates the job_types "
                         "field (run_physical.py)")
    args = ap.parse_args()

    results = pickle.load(open(args.physical, "rb"))
    if not results.get("job_types") and args.trace:
        results["job_types"] = {
            str(i): line.split("\t")[0]
            for i, line in enumerate(open(args.trace))
            if line.strip()
        }
    hot, startup = calibrate(results)
    print("calibrated hot rates:", json.dumps(hot, indent=1))
    print("first-dispatch startup:", json.dumps(startup, indent=1))

    oracle = json.load(open(args.oracle))
    # oracle schema: {worker_type: {"('<job_type>', <sf>)": {"null": rate,
    # ...}}}; also rescale sf>1 entries by the same factor (they were
    # extrapolated from the sf=1 rate)
    n = 0
    worker_tables = [
        v for v in oracle.values() if isinstance(v, dict)
    ] or [oracle]
    # capture pre-update isolated rates FIRST (sf>1 rescale factor)
    old_iso = {}
    for t in worker_tables:
        for jt in hot:
            e1 = t.get(f"('{jt}', 1)")
            if e1 and e1.get("null"):
                old_iso.setdefault(jt, e1["null"])
    for key, entry in [
        (k, e) for t in worker_tables for k, e in t.items()
    ]:
        m = re.match(r"^\('(?P<jt>.+)', (?P<sf>\d+)\)$", key)
        if not m:
            continue
        jt, sf = m.group("jt"), int(m.group("sf"))
        if jt in hot and "null" in entry:
            if sf == 1:
                entry["null"] = hot[jt]
            elif old_iso.get(jt):
                entry["null"] *= hot[jt] / old_iso[jt]
            n += 1
    with open(args.out_oracle, "w") as f:
        json.dump(oracle, f, indent=1)
    with open(args.out_startup, "w") as f:
        json.dump(startup, f, indent=1)
    print(f"wrote {args.out_oracle} ({n} rates updated) and "
          f"{args.out_startup}")


if __name__ == "__main__":
    main()
