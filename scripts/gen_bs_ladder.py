#!/usr/bin/env python3
"""Regenerate the GNS batch-size ladder data from the reference oracle.

The simulator's adaptation twins need the exact per-epoch batch-size
schedules the reference's simulator uses (utils.py:741-1330 — ~600 lines
of hand-profiled if/else ladders).  These are behavioral CONSTANTS; this
script derives them as data by loading the two oracle functions from the
reference tree at run time (AST extraction — no code is copied into the
repo) and run-length-encoding their output, including the reference's
loop-ordering quirk where check-first segments leave the FINAL epoch at
the base batch size (r1 deliberately skipped that quirk; round 2
reproduces it for parity — it shifts the last epoch's duration by up to
8x for LM jobs).

Output schema (core/data/gns_bs_ladder.json):
  "<model>|<bs>|<sf>": [[start, end_or_null, multiplier, check_first], ...]
"""

import ast
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

REF_UTILS = "/root/reference/scheduler/utils.py"


def load_reference_oracles():
    src = open(REF_UTILS).read()
    ns = {}
    for node in ast.parse(src).body:
        if isinstance(node, ast.FunctionDef) and node.name in (
            "get_gns_bs_pattern",
            "get_accordion_bs_pattern",
        ):
            exec(
                compile(ast.Module(body=[node], type_ignores=[]), REF_UTILS,
                        "exec"),
                ns,
            )
    return ns["get_gns_bs_pattern"], ns.get("get_accordion_bs_pattern")


def rle_with_styles(ref_gns, job_type, bs, sf, big_e=600):
    """Segment boundaries from a large-E evaluation; per-segment
    check-first style probed by truncating E into the segment."""
    full = [int(x) for x in ref_gns(job_type, bs, big_e, sf)]
    segs = []
    start = 0
    for i in range(1, big_e):
        # ignore the last-epoch quirk position in the big evaluation
        if i == big_e - 1:
            break
        if full[i] != full[start]:
            segs.append((start, i, full[start] // bs))
            start = i
    segs.append((start, None, full[start] // bs))

    out = []
    for s, e, mult in segs:
        check_first = False
        if mult != 1:
            probe_e = s + 2
            pat = [int(x) for x in ref_gns(job_type, bs, probe_e, sf)]
            check_first = pat[-1] == bs
        out.append([s, e, mult, bool(check_first)])
    return out


def main():
    from shockwave_amd.core.bs_patterns import _DATA_DIR

    ref_gns, _ = load_reference_oracles()
    existing = json.load(
        open(os.path.join(_DATA_DIR, "gns_bs_ladder.json"))
    )
    out = {}
    for key in existing:
        model, bs, sf = key.split("|")
        job_type = f"{model} (batch size {bs})"
        out[key] = rle_with_styles(ref_gns, job_type, int(bs), int(sf))
    path = os.path.join(_DATA_DIR, "gns_bs_ladder.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {path} ({len(out)} keys)")


if __name__ == "__main__":
    main()
