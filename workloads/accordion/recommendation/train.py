"""Trace-compatible entry shim -> shockwave_amd.workloads.families.recommendation_main
(mode: accordion).  Keeps the reference's run-dir/CLI layout so its traces
dispatch unchanged."""
import os
import sys

_d = os.path.dirname(os.path.abspath(__file__))
while not os.path.isdir(os.path.join(_d, "shockwave_amd")):
    parent = os.path.dirname(_d)
    if parent == _d:
        raise RuntimeError("repo root not found")
    _d = parent
sys.path.insert(0, _d)

from shockwave_amd.workloads.families import recommendation_main

if __name__ == "__main__":
    sys.exit(0 if recommendation_main(mode="accordion") is not None else 1)
