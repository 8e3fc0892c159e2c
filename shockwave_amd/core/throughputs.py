"""Throughput-oracle file I/O.

File schema is the reference's "v2" JSON (utils.py:562-601):

.. code-block:: json

    { "<worker_type>": {
        "('<job type>', <scale_factor>)": {
            "null": <isolated steps/s>,
            "('<other job type>', <sf>)": [<my steps/s>, <their steps/s>]
        } } }

Keys parse to ``(job_type, scale_factor)`` tuples; ``"null"`` holds the
isolated throughput and tuple keys hold pairwise colocated throughputs.
"""

from __future__ import annotations

import json
import re
from typing import Dict, Optional, Tuple

_KEY_RE = re.compile(r"\('(.*)', (\d+)\)")

JobTypeKey = Tuple[str, int]


def parse_job_type_key(s: str) -> Optional[JobTypeKey]:
    m = _KEY_RE.match(s)
    if m is None:
        return None
    return (m.group(1), int(m.group(2)))


def format_job_type_key(key: JobTypeKey) -> str:
    return "('%s', %d)" % key


def read_throughputs(path: str) -> Dict[str, Dict]:
    with open(path) as f:
        raw = json.load(f)
    parsed: Dict[str, Dict] = {}
    for worker_type, per_type in raw.items():
        parsed[worker_type] = {}
        for job_type_str, entry in per_type.items():
            key = parse_job_type_key(job_type_str)
            assert key is not None, job_type_str
            parsed[worker_type][key] = {}
            for other, value in entry.items():
                other_key = "null" if other == "null" else parse_job_type_key(other)
                assert other_key is not None, other
                parsed[worker_type][key][other_key] = value
    return parsed


def write_throughputs(parsed: Dict[str, Dict], path: str) -> None:
    raw = {}
    for worker_type, per_type in parsed.items():
        raw[worker_type] = {}
        for key, entry in per_type.items():
            raw[worker_type][format_job_type_key(key)] = {
                ("null" if ok == "null" else format_job_type_key(ok)): v
                for ok, v in entry.items()
            }
    with open(path, "w") as f:
        json.dump(raw, f, indent=2)


def isolated_throughput(
    parsed: Dict[str, Dict], worker_type: str, job_type: str, scale_factor: int
) -> float:
    return parsed[worker_type][(job_type, scale_factor)]["null"]
