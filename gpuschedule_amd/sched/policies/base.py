"""Scheduling-policy interface.

A policy ranks active jobs; the scheduler core (``sched/scheduler.py``)
turns the ranking into placements, preemptions, migrations and resizes
against the cluster state.  This split keeps policies tiny and lets the
simulator and the real Torch backend share every policy unchanged.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

from ...trace.job import Job

# A ranked request: (job_id, desired gang width).  Highest priority
# first.  Width differs from spec only for elastic jobs.
Request = Tuple[str, int]


class Policy:
    name = "base"
    #: False => running jobs are never preempted by the core (FIFO).
    preemptive = True
    #: True  => stop allocating at the first job that does not fit
    #: (head-of-line blocking, classic FIFO semantics).
    strict_order = False
    #: True  => the core may move running jobs to defragment the
    #: cluster when a gang cannot find a contiguous window (Gandiva).
    migration = False
    #: Elastic jobs may shrink below desired width to fit.
    elastic = False
    #: Gandiva packing: allow co-running a second 1-GPU job on a busy
    #: GPU (separate HIP streams) when no idle GPU exists.
    packing = False
    pack_capacity = 2

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        """Return active jobs in descending priority with desired
        widths.  Jobs omitted from the list are not scheduled."""
        raise NotImplementedError

    def on_tick(self, now: float, jobs: Dict[str, Job]) -> None:
        """Per-tick bookkeeping (queue levels, quanta)."""

    def describe(self) -> dict:
        return {"name": self.name, "preemptive": self.preemptive,
                "migration": self.migration, "elastic": self.elastic}


def active_jobs(jobs: Dict[str, Job]) -> List[Job]:
    return [j for j in jobs.values() if j.is_active]
