"""Lease value object.

A lease bounds how long a dispatched job may run in the current round before
it must cooperatively preempt itself: at most ``max_steps`` iterations and at
most ``max_duration`` seconds, whichever is hit first.  Mirrors
/root/reference/scheduler/lease.py:1-26.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class Lease:
    max_steps: int
    max_duration: float

    def __str__(self):
        return f"Lease(max_steps={self.max_steps}, max_duration={self.max_duration:.1f})"
