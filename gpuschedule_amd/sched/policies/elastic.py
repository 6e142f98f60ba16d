"""Elastic-resize policy: grow elastic DP jobs into idle GPUs.

Non-elastic jobs are scheduled SRTF-like at their fixed width.  Elastic
jobs then expand to soak up whatever GPUs remain idle (up to
``max_gpus``), and shrink (down to ``min_gpus``) when fixed jobs need
the space — config 5's "Llama-3-8B DP job growing 1->8 GPUs".

Resize hysteresis: a job is only resized if its width would change by
at least one GPU and its last resize was more than ``resize_interval``
seconds ago (resizing re-shards the optimizer and re-splits the global
batch, which costs a step).
"""
from __future__ import annotations

from typing import Dict, List

from ...trace.job import Job, JobState
from .base import Policy, Request, active_jobs


class ElasticPolicy(Policy):
    name = "elastic"
    preemptive = True
    elastic = True

    def __init__(self, default_iter_time: float = 1.0,
                 resize_interval: float = 10.0):
        self.default_iter_time = default_iter_time
        self.resize_interval = resize_interval
        self._last_resize: Dict[str, float] = {}
        self._total_gpus = 0  # set by the scheduler core each tick

    def rank(self, now: float, jobs: Dict[str, Job]) -> List[Request]:
        act = active_jobs(jobs)
        fixed = [j for j in act if not j.spec.elastic]
        elas = [j for j in act if j.spec.elastic]
        fixed.sort(key=lambda j: (j.remaining_time(self.default_iter_time),
                                  j.submit_t, j.job_id))
        elas.sort(key=lambda j: (j.submit_t, j.job_id))

        reqs: List[Request] = [(j.job_id, j.spec.n_gpus) for j in fixed]
        # GPUs left over after fixed jobs (upper bound; the allocator
        # shrinks elastic widths further if placement fails).
        used = sum(j.spec.n_gpus for j in fixed)
        avail = max(0, self._total_gpus - used)
        for j in elas:
            share = max(j.spec.min_gpus, avail // max(1, len(elas)))
            want = min(j.spec.max_gpus, share)
            cur = len(j.placement) if j.state == JobState.RUNNING else 0
            last = self._last_resize.get(j.job_id, -1e18)
            if cur and want != cur and now - last < self.resize_interval:
                want = cur  # hysteresis: hold width
            if cur and want != cur:
                self._last_resize[j.job_id] = now
            reqs.append((j.job_id, max(j.spec.min_gpus, want)))
        return reqs
