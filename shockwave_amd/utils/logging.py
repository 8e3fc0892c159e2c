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
