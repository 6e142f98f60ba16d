"""SRTF: shortest-remaining-time-first, preemptive.

Priority = estimated remaining service time (remaining iterations x
measured or hinted per-iteration time).  A newly submitted short job
preempts a long-running one.  Ties break by submit order so the
schedule is deterministic.
"""
from __future__ import annotations

from typing import Dict, List

from ...trace.job import Job
from .base import Policy, Request, active_jobs


class SRTFPolicy(Policy):
    name = "srtf"
    preemptive = True

    def __init__(self, default_iter_time: float = 1.0):
        self.default_iter_time = default_iter_time

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        order = sorted(
            active_jobs(jobs),
            key=lambda j: (j.remaining_time(self.default_iter_time),
                           j.submit_t, j.job_id))
        return [(j.job_id, j.spec.n_gpus) for j in order]
