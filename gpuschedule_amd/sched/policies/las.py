"""Tiresias-style LAS: discretized least-attained-service MLFQ.

Jobs start in the highest-priority queue (level 0) and are demoted as
their *attained GPU-service* (GPU-seconds = wall seconds x gang width)
crosses level thresholds.  Across levels, lower level wins; within a
level, FIFO by submit time (Tiresias' discretized 2DAS with K queues).

Starvation guard (Tiresias' PROMOTE): a job that has waited longer than
``starve_limit`` x its total executed time since it last ran is
promoted back to level 0.
"""
from __future__ import annotations

from typing import Dict, List, Sequence

from ...trace.job import Job, JobState
from .base import Policy, Request, active_jobs


class LASPolicy(Policy):
    name = "las"
    preemptive = True

    def __init__(self,
                 thresholds: Sequence[float] = (60.0, 600.0),
                 starve_limit: float = 4.0):
        # thresholds[k] = attained GPU-seconds above which a job sits in
        # queue level k+1.  len(thresholds)+1 levels total.
        self.thresholds = tuple(thresholds)
        self.starve_limit = starve_limit
        self._last_ran: Dict[str, float] = {}
        self._promoted: Dict[str, float] = {}  # job -> promote time

    def _level(self, job: Job) -> int:
        lvl = 0
        for k, th in enumerate(self.thresholds):
            if job.attained_service >= th:
                lvl = k + 1
        return lvl

    def on_tick(self, now: float, jobs: Dict[str, Job]) -> None:
        for j in active_jobs(jobs):
            if j.state == JobState.RUNNING:
                self._last_ran[j.job_id] = now
                self._promoted.pop(j.job_id, None)
            else:
                last = self._last_ran.get(j.job_id, j.submit_t)
                waited = now - last
                executed = j.attained_service / max(1, j.n_gpus)
                if (executed > 0 and waited > self.starve_limit * executed
                        and j.job_id not in self._promoted):
                    self._promoted[j.job_id] = now
            j.queue_level = (0 if j.job_id in self._promoted
                             else self._level(j))

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        order = sorted(active_jobs(jobs),
                       key=lambda j: (j.queue_level, j.submit_t, j.job_id))
        return [(j.job_id, j.spec.n_gpus) for j in order]
