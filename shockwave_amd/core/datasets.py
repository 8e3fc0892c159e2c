"""Model/dataset profile tables, loaded from data files.

The reference hard-codes these in Python (utils.py:40-54 dataset sizes,
utils.py:706-737 per-(model, batch-size) memory/utilization tables); here
they live in ``data/job_profiles.json`` so new profiles can be dropped in
without code changes.
"""

from __future__ import annotations

import json
import os
from functools import lru_cache

_DATA_DIR = os.path.join(os.path.dirname(__file__), "data")


@lru_cache(maxsize=None)
def _profiles():
    with open(os.path.join(_DATA_DIR, "job_profiles.json")) as f:
        return json.load(f)


def dataset_for_model(model: str) -> str:
    return _profiles()["model_dataset"][model]


def dataset_len(dataset: str) -> int:
    return _profiles()["dataset_len"][dataset]


def mem_mb(model: str, batch_size: int) -> float:
    return _profiles()["mem_mb"][model][str(batch_size)]


def util_pct(model: str, batch_size: int) -> float:
    return _profiles()["util_pct"][model][str(batch_size)]


def max_batch_size(model: str, default: int = None) -> int:
    return _profiles()["max_bs"].get(model, default)


def steps_per_epoch(model: str, batch_size: int) -> int:
    import math

    return math.ceil(dataset_len(dataset_for_model(model)) / batch_size)
