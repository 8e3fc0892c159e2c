"""GPU discovery on ROCm hosts.

Reference uses nvidia-smi/pynvml (utils.py:289-327); the MI355X-native path
asks the HIP runtime via torch first, then amd-smi/rocm-smi.
"""

from __future__ import annotations

import shutil
import subprocess


def get_num_gpus() -> int:
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    for tool, args in (
        ("amd-smi", ["list", "--csv"]),
        ("rocm-smi", ["--showid", "--csv"]),
    ):
        if shutil.which(tool):
            try:
                out = subprocess.run(
                    [tool] + args, capture_output=True, text=True, timeout=30
                ).stdout
                rows = [
                    l for l in out.splitlines()
                    if l.strip() and not l.lower().startswith(("gpu", "device", "#"))
                ]
                if rows:
                    return len(rows)
            except Exception:
                continue
    return 0


def get_gpu_processes():
    """pid -> memory MB per GPU process (best effort)."""
    if shutil.which("amd-smi"):
        try:
            out = subprocess.run(
                ["amd-smi", "process", "--csv"],
                capture_output=True, text=True, timeout=30,
            ).stdout
            procs = {}
            for line in out.splitlines()[1:]:
                parts = line.split(",")
                if len(parts) >= 2 and parts[1].strip().isdigit():
                    procs[int(parts[1])] = parts
            return procs
        except Exception:
            pass
    return {}
