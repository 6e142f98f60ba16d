"""Gandiva-style time-slicing with suspend/resume and migration.

All active jobs round-robin over the GPUs with a time-slice quantum.
Suspension is cheap by design on MI355X: a suspended job's state stays
RESIDENT in the GPU's 288 GB HBM (no copy at all); eviction to a host
checkpoint happens only under memory pressure.  The policy also enables
the core's migration path so fragmented singles are moved to open
contiguous windows for gangs (Gandiva's "migration for locality").

Rotation rule: every ``quantum`` seconds the set of running jobs is
re-ranked by least-recently-scheduled first (round-robin); between
quantum boundaries the current ranking is kept stable so jobs actually
hold the GPU for a full slice.
"""
from __future__ import annotations

from typing import Dict, List

from ...trace.job import Job, JobState
from .base import Policy, Request, active_jobs


class GandivaPolicy(Policy):
    name = "gandiva"
    preemptive = True
    migration = True

    def __init__(self, quantum: float = 60.0, packing: bool = False,
                 pack_capacity: int = 2):
        self.quantum = quantum
        self.packing = packing
        self.pack_capacity = pack_capacity
        self._slice_start: Dict[str, float] = {}
        self._next_rotate = 0.0

    def on_tick(self, now: float, jobs: Dict[str, Job]) -> None:
        for j in active_jobs(jobs):
            if j.state == JobState.RUNNING:
                self._slice_start.setdefault(j.job_id, now)
            else:
                self._slice_start.pop(j.job_id, None)

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        act = active_jobs(jobs)
        rotate = now >= self._next_rotate
        if rotate:
            self._next_rotate = now + self.quantum

        def key(j: Job):
            running = j.state == JobState.RUNNING
            if rotate:
                # Round-robin: least-recently-started-slice first; jobs
                # currently holding a slice go to the back.
                slice_age = now - self._slice_start.get(j.job_id, -1e18)
                return (1 if running else 0, -slice_age, j.submit_t, j.job_id)
            # Mid-quantum: keep running jobs in place, queue others.
            return (0 if running else 1, j.submit_t, j.job_id)

        return [(j.job_id, j.spec.n_gpus) for j in sorted(act, key=key)]
