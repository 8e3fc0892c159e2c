"""Job record and job-id pairing.

Behavioral parity with the reference's job model
(/root/reference/scheduler/job.py:1-166 and job_id_pair.py:1-93): a job is an
immutable-ish record of a training command plus scheduling metadata; dynamic
batch-size adaptation rewrites the batch-size token in both the command and
the job_type string.  Job-type strings follow the reference convention
``"<Model> (batch size <N>)"`` so throughput-oracle keys and traces
inter-operate.
"""

from __future__ import annotations

import re
from typing import Optional


_BS_ARG_RE = re.compile(
    r"(?P<flag>--batch[_-]size[= ]|-batch_size |-b |--bs[= ])(?P<bs>\d+)"
)


class Job:
    """One training job as it appears in a trace.

    Fields mirror the 12-field trace line format (SURVEY.md §2.1 "Traces"):
    job_type, command, working_directory, num_steps_arg, needs_data_dir,
    total_steps, scale_factor, mode, priority_weight, SLO, duration,
    arrival_time (arrival kept outside the Job).
    """

    def __init__(
        self,
        job_id,
        job_type: str,
        command: str,
        working_directory: str,
        num_steps_arg: str,
        total_steps: int,
        duration,
        mps_thread_percentage: int = 100,
        scale_factor: int = 1,
        mode: str = "static",
        priority_weight: float = 1.0,
        SLO: Optional[float] = None,
        needs_data_dir: bool = False,
    ):
        self.job_id = job_id
        self.job_type = job_type
        self.command = command
        self.working_directory = working_directory
        self.needs_data_dir = needs_data_dir
        self.num_steps_arg = num_steps_arg
        self.total_steps = int(total_steps)
        self._duration = duration
        self.scale_factor = int(scale_factor)
        self.mode = mode
        self.priority_weight = priority_weight
        self.mps_thread_percentage = mps_thread_percentage
        # Negative SLO in trace files means "no SLO".
        self.SLO = None if (SLO is not None and SLO < 0) else SLO

    # -- trace serialization ------------------------------------------------

    def __str__(self):
        slo = -1.0 if self.SLO is None else self.SLO
        return "%s\t%s\t%s\t%s\t%d\t%d\t%d\t%s\t%d\t%f\t%d" % (
            self.job_type,
            self.command,
            self.working_directory,
            self.num_steps_arg,
            int(self.needs_data_dir),
            self.total_steps,
            self.scale_factor,
            self.mode,
            self.priority_weight,
            slo,
            int(float(self._duration)) if self._duration is not None else 0,
        )

    def __repr__(self):
        return f"Job(id={self.job_id}, type={self.job_type!r}, sf={self.scale_factor}, mode={self.mode})"

    # -- derived properties -------------------------------------------------

    @property
    def duration(self) -> int:
        return int(float(self._duration)) if self._duration is not None else 0

    @duration.setter
    def duration(self, value):
        self._duration = value

    @property
    def batch_size(self) -> int:
        """Parse N out of '<Model> (batch size <N>)'."""
        jt = self.job_type
        return int(jt[jt.rfind(" ") + 1 : -1])

    @property
    def model(self) -> str:
        jt = self.job_type
        return jt[: jt.find(" ")]

    def update_bs(self, new_bs: int) -> None:
        """Rewrite the batch-size token in both the command and job_type.

        The reference does this positionally (job.py:142-166); we match the
        batch-size *flag* instead, which handles every command template in
        the job table regardless of argument order.
        """
        m = _BS_ARG_RE.search(self.command)
        if m is None:
            raise ValueError(
                f"no batch-size flag found in command {self.command!r}"
            )
        self.command = (
            self.command[: m.start("bs")]
            + str(new_bs)
            + self.command[m.end("bs") :]
        )
        self.job_type = self.job_type[: self.job_type.rfind(" ")] + f" {new_bs})"


class JobIdPair:
    """A single job id or a colocated (packed) pair of ids.

    Hashable, ordered; mirrors the reference's JobIdPair semantics
    (job_id_pair.py:1-93): ``JobIdPair(3, None)`` is the singleton job 3,
    ``JobIdPair(3, 5)`` is jobs 3 and 5 space-sharing one accelerator.
    """

    __slots__ = ("_ids",)

    def __init__(self, job0: Optional[int], job1: Optional[int] = None):
        if job0 is None and job1 is None:
            raise ValueError("at least one job id required")
        if job0 is None:
            job0, job1 = job1, None
        if job1 is not None and job1 < job0:
            job0, job1 = job1, job0
        self._ids = (job0, job1)

    def __getitem__(self, i):
        return self._ids[i]

    def __eq__(self, other):
        return isinstance(other, JobIdPair) and self._ids == other._ids

    def __lt__(self, other):
        a = (self._ids[0], -1 if self._ids[1] is None else self._ids[1])
        b = (other._ids[0], -1 if other._ids[1] is None else other._ids[1])
        return a < b

    def __hash__(self):
        return hash(self._ids)

    def __repr__(self):
        if self._ids[1] is None:
            return str(self._ids[0])
        return f"({self._ids[0]}, {self._ids[1]})"

    def is_pair(self) -> bool:
        return self._ids[1] is not None

    def overlaps_with(self, other: "JobIdPair") -> bool:
        mine = set(i for i in self._ids if i is not None)
        theirs = set(i for i in other._ids if i is not None)
        return bool(mine & theirs)

    def singletons(self):
        if self._ids[1] is None:
            return (self,)
        return (JobIdPair(self._ids[0]), JobIdPair(self._ids[1]))
