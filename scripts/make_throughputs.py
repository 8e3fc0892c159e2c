#!/usr/bin/env python3
"""Generate the MI355X throughput-oracle JSON.

Produces ``traces/mi355x_throughputs.json`` in the reference "v2" schema
(shockwave_amd/core/throughputs.py).  Isolated steps/s start from estimated
MI355X-vs-V100 speedups per model family and are overridden by any measured
numbers in ``profiles/measured_throughputs.json`` (written by
scripts/measure_throughput.py on a real MI355X box).

Scale-factor entries model RCCL-over-xGMI data parallelism: aggregate
steps/s = sf * isolated * eff(sf), with efficiencies reflecting 7-link xGMI
all-reduce overlap (far better than the reference's PCIe/IB V100 numbers,
where DDP cost ~30-50% per doubling).

Pairwise (packing) entries use a utilization contention model: two packed
jobs each run at iso/(1 + (u_a/100)*(u_b/100)) — contention scales with
the product of their GPU utilizations.
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from shockwave_amd.core import datasets
from shockwave_amd.core.throughputs import format_job_type_key

# Estimated MI355X single-GPU steps/s per (model, batch size).
# Seeded from V100-class profiles x per-family speedup estimates;
# overridden by measurements when available.
SPEEDUP = {
    "ResNet-18": 3.0,     # tiny 32x32 convs, launch-bound -> modest speedup
    "ResNet-50": 6.0,     # 224x224 convs, MFMA-bound
    "Transformer": 6.0,
    "LM": 2.0,            # small LSTM, kernel-launch bound
    "Recommendation": 4.0,
}

V100_BASE = {
    ("ResNet-18", 16): 57.68, ("ResNet-18", 32): 42.97, ("ResNet-18", 64): 21.43,
    ("ResNet-18", 128): 11.78, ("ResNet-18", 256): 6.32,
    ("ResNet-50", 16): 10.60, ("ResNet-50", 32): 5.90, ("ResNet-50", 64): 3.11,
    ("ResNet-50", 128): 1.60,
    ("Transformer", 16): 8.71, ("Transformer", 32): 4.53, ("Transformer", 64): 2.08,
    ("Transformer", 128): 1.15,
    ("LM", 5): 133.84, ("LM", 10): 94.91, ("LM", 20): 68.05, ("LM", 40): 46.82,
    ("LM", 80): 21.71,
    ("Recommendation", 512): 169.35, ("Recommendation", 1024): 107.32,
    ("Recommendation", 2048): 59.26, ("Recommendation", 4096): 25.07,
    ("Recommendation", 8192): 11.63,
}

# DP scaling efficiency over xGMI (aggregate = sf * iso * eff)
XGMI_EFF = {1: 1.0, 2: 0.96, 4: 0.93, 8: 0.89}

WORKER_TYPE = "mi355x"


def isolated(model, bs, measured):
    key = f"{model}|{bs}"
    if key in measured:
        return measured[key]
    return V100_BASE[(model, bs)] * SPEEDUP[model]


def main(out_path="traces/mi355x_throughputs.json"):
    measured = {}
    mpath = "profiles/measured_throughputs.json"
    if os.path.exists(mpath):
        measured = json.load(open(mpath))
        print(f"using {len(measured)} measured entries from {mpath}")

    # enumerate (job_type, bs) pairs incl. every batch size an adaptive job
    # can scale to (update_bs can move within the family's bs grid)
    family_bs = {
        "ResNet-18": [16, 32, 64, 128, 256],
        "ResNet-50": [16, 32, 64, 128],
        "Transformer": [16, 32, 64, 128],
        "LM": [5, 10, 20, 40, 80],
        "Recommendation": [512, 1024, 2048, 4096, 8192],
    }
    entries = {}
    for model, bss in family_bs.items():
        distributed = model not in ("Recommendation",)
        for bs in bss:
            iso = isolated(model, bs, measured)
            for sf in ([1, 2, 4, 8] if distributed else [1]):
                jt = f"{model} (batch size {bs})"
                entries[(jt, sf)] = iso * sf * XGMI_EFF[sf]

    # pairwise colocation model (single-GPU jobs only, like the reference)
    raw = {WORKER_TYPE: {}}
    keys = sorted(entries.keys())
    for (jt, sf) in keys:
        e = {"null": entries[(jt, sf)]}
        if sf == 1:
            my_model = jt.split(" (")[0]
            my_bs = int(jt[jt.rfind(" ") + 1 : -1])
            for (jt2, sf2) in keys:
                if sf2 != 1:
                    continue
                other_model = jt2.split(" (")[0]
                other_bs = int(jt2[jt2.rfind(" ") + 1 : -1])
                u_mine = datasets.util_pct(my_model, my_bs)
                u_theirs = datasets.util_pct(other_model, other_bs)
                # contention grows with the PRODUCT of utilizations, so a
                # job's colocation fingerprint depends on its own intensity
                contention = (u_mine / 100.0) * (u_theirs / 100.0)
                mine = entries[(jt, 1)] / (1.0 + contention)
                theirs = entries[(jt2, 1)] / (1.0 + contention)
                e[format_job_type_key((jt2, 1))] = [mine, theirs]
        raw[WORKER_TYPE][format_job_type_key((jt, sf))] = e

    out_dir = os.path.dirname(out_path)
    if out_dir:
        os.makedirs(out_dir, exist_ok=True)
    with open(out_path, "w") as f:
        json.dump(raw, f, indent=1)
    print(f"wrote {len(raw[WORKER_TYPE])} job-type entries -> {out_path}")


if __name__ == "__main__":
    main(*sys.argv[1:])
