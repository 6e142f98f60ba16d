"""FIFO: run jobs in arrival order, never preempt, block at the head.

Semantics: running jobs run to completion.  Queued jobs are considered
strictly in (submit_time, job_id) order; if the head job cannot be
placed, nothing behind it starts (no backfill — classic FIFO).
"""
from __future__ import annotations

from typing import Dict, List

from ...trace.job import Job
from .base import Policy, Request, active_jobs


class FIFOPolicy(Policy):
    name = "fifo"
    preemptive = False
    strict_order = True

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        order = sorted(active_jobs(jobs),
                       key=lambda j: (j.submit_t, j.job_id))
        return [(j.job_id, j.spec.n_gpus) for j in order]
