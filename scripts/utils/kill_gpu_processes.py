#!/usr/bin/env python3
"""Kill leftover GPU training processes on this node (cleanup utility;
reference scripts/utils GPU process killer).  Uses amd-smi process listing
plus the dispatcher's env marker (GAVEL_JOB_ID) to avoid unrelated
processes; never kills by bare name pattern."""

import argparse
import os
import signal
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from shockwave_amd.runtime.gpu import get_gpu_processes


def job_pids():
    """PIDs whose environment carries the scheduler's job marker."""
    pids = []
    for pid in os.listdir("/proc"):
        if not pid.isdigit():
            continue
        try:
            with open(f"/proc/{pid}/environ", "rb") as f:
                env = f.read()
            if b"GAVEL_JOB_ID=" in env:
                pids.append(int(pid))
        except (PermissionError, FileNotFoundError):
            continue
    return pids


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--dry_run", action="store_true")
    args = p.parse_args()
    targets = set(job_pids())
    gpu_procs = set(get_gpu_processes().keys())
    if gpu_procs:
        targets &= gpu_procs | targets  # keep job-marked even if not listed
    for pid in sorted(targets):
        print(f"{'would kill' if args.dry_run else 'killing'} pid {pid}")
        if not args.dry_run:
            try:
                os.kill(pid, signal.SIGTERM)
            except ProcessLookupError:
                pass


if __name__ == "__main__":
    main()
