"""Logging helpers: scheduler-clock log adapter.

Reference: scheduler/custom_logging.py:5-13 and Utility.py — prefixes
every record with the scheduler's own clock (simulated time in simulation,
elapsed wall time physically) so logs line up with round boundaries.
"""

from __future__ import annotations

import logging


class SchedulerAdapter(logging.LoggerAdapter):
    def __init__(self, logger, scheduler):
        super().__init__(logger, {})
        self._scheduler = scheduler

    def process(self, msg, kwargs):
        ts = self._scheduler.get_current_timestamp()
        return f"[{ts:.2f}] {msg}", kwargs


def build_logger(name: str, level=logging.INFO, log_file=None):
    logger = logging.getLogger(name)
    logger.setLevel(level)
    fmt = logging.Formatter("{name}:{levelname} {message}", style="{")
    handler = logging.StreamHandler()
    handler.setFormatter(fmt)
    logger.addHandler(handler)
    if log_file:
        fh = logging.FileHandler(log_file)
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    return logger


def enable_hang_diagnosis(path: str = ".stack_trace.log",
                          interval_s: float = 30.0):
    """Periodically dump every thread's stack to ``path`` — if the head
    process wedges (lost cv notify, stuck RPC), the last dump shows where
    (reference scheduler.py:450-455 faulthandler hook).

    Returns a cancel() callable; the file handle stays open for
    faulthandler's lifetime."""
    import faulthandler

    f = open(path, "w")
    faulthandler.dump_traceback_later(
        interval_s, repeat=True, file=f
    )

    def cancel():
        faulthandler.cancel_dump_traceback_later()
        f.close()

    return cancel
