"""Oracle batch-size adaptation schedules for the simulator.

The simulator needs to know, without running a job, how Accordion / GNS would
change its batch size over epochs.  Parity targets:

* Accordion — the critical-regime windows per model family from the
  reference's ``get_accordion_bs_pattern`` (utils.py:741-800), expressed as
  rules: outside a critical-regime epoch window AND past 30% of training, the
  job jumps to its family's max batch size.
* GNS — the reference encodes a ~500-line if/else ladder of hand-profiled
  doubling schedules (utils.py:801-1330).  That is pure data; it lives here
  as run-length-encoded multiplier segments in ``data/gns_bs_ladder.json``
  keyed by ``"<model>|<bs>|<scale_factor>"``.  (One deliberate deviation: the
  reference's loop-ordering quirk that leaves the final epoch un-multiplied
  in some branches is not reproduced.)
"""

from __future__ import annotations

import json
import os
from functools import lru_cache
from typing import List

from . import datasets

_DATA_DIR = os.path.join(os.path.dirname(__file__), "data")


@lru_cache(maxsize=None)
def _gns_ladder():
    with open(os.path.join(_DATA_DIR, "gns_bs_ladder.json")) as f:
        return json.load(f)


def accordion_bs_pattern(
    job_type: str, initial_batch_size: int, num_epochs: int
) -> List[int]:
    model = job_type[: job_type.find(" ")]
    bs = [initial_batch_size] * num_epochs

    if model == "ResNet-18":
        regime = 20 if initial_batch_size >= 256 else 10
        critical = set(range(regime)) | set(range(150, 160)) | set(range(250, 260))
    elif model == "ResNet-50":
        critical = {e for e in range(600) if e % 30 < 10}
    elif model == "LM":
        critical = set(range(10))
    elif model == "Recommendation":
        if initial_batch_size in (512, 1024):
            regime = 30
        elif initial_batch_size == 2048:
            regime = 40
        else:
            regime = 10
        critical = set(range(regime)) | set(range(60, 70)) | set(range(80, 90))
    else:
        # Transformer / CycleGAN / A3C: Accordion not applicable
        # (scheduler.py:1670-1672 in the reference)
        return bs

    max_bs = datasets.max_batch_size(model, initial_batch_size)
    for epoch in range(num_epochs):
        if epoch not in critical and epoch > num_epochs * 0.3:
            bs[epoch] = max_bs
    return bs


def gns_bs_pattern(
    job_type: str, batch_size: int, num_epochs: int, scale_factor: int
) -> List[int]:
    model = job_type[: job_type.find(" ")]
    key = f"{model}|{batch_size}|{scale_factor}"
    segments = _gns_ladder().get(key)
    bs = [batch_size] * num_epochs
    if segments is None:
        return bs
    for seg in segments:
        start, end, mult = seg[0], seg[1], seg[2]
        # check_first: the reference's check-then-multiply loops leave
        # the FINAL training epoch at the base batch size (loop-ordering
        # quirk in utils.py:801-1330, reproduced as data for parity)
        check_first = bool(seg[3]) if len(seg) > 3 else False
        end = num_epochs if end is None else min(end, num_epochs)
        for e in range(start, end):
            if check_first and e + 1 >= num_epochs:
                break
            bs[e] = batch_size * mult
    return bs


def bs_pattern_for_mode(
    mode: str, job_type: str, batch_size: int, num_epochs: int, scale_factor: int
) -> List[int]:
    if mode == "accordion":
        return accordion_bs_pattern(job_type, batch_size, num_epochs)
    if mode == "gns":
        return gns_bs_pattern(job_type, batch_size, num_epochs, scale_factor)
    return [batch_size] * num_epochs
