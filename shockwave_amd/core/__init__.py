from .job import Job, JobIdPair
from .lease import Lease
from .job_table import JobTable, JobTemplate, build_job_table

__all__ = ["Job", "JobIdPair", "Lease", "JobTable", "JobTemplate", "build_job_table"]
