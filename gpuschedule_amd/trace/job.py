"""Job specification and runtime lifecycle state.

A job is the scheduler's unit of work: a DL training run that needs
``n_gpus`` GPUs as a gang for ``total_iters`` training iterations.
The scheduler never looks inside the model; backends (simulator or the
real Torch backend) execute iterations and report progress.

States::

    QUEUED ──► RUNNING ──► DONE
                 │  ▲
                 ▼  │            (suspend keeps state in HBM or evicts
              SUSPENDED           it to a host checkpoint — Gandiva)
                 │
                 ▼
               FAILED
"""
from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import List, Optional

# Per-GPU HBM on MI355X. Placement/bin-packing is sized against this.
MI355X_HBM_BYTES = 288 * 1024**3


class JobState(enum.Enum):
    QUEUED = "QUEUED"
    RUNNING = "RUNNING"
    SUSPENDED = "SUSPENDED"
    DONE = "DONE"
    FAILED = "FAILED"


@dataclass
class JobSpec:
    """Immutable description of a submitted job (one trace row)."""

    job_id: str
    submit_time: float          # seconds from trace start
    n_gpus: int                 # gang width (DP degree)
    model: str                  # model family key, see models.registry
    total_iters: int            # iterations to completion
    batch_size: int = 0         # per-GPU batch (0 = model default)
    mem_bytes: int = 0          # per-GPU HBM estimate (0 = model default)
    elastic: bool = False       # may be resized by the elastic policy
    min_gpus: int = 1           # elastic lower bound
    max_gpus: int = 8           # elastic upper bound
    # Optional oracle iteration time (seconds) for the simulator; the
    # real backend measures it instead.
    iter_time_hint: float = 0.0
    # Best-effort class: scheduled only onto GPUs no normal job wants
    # this tick (the scheduler core ranks background jobs last under
    # every policy).  Used for fill/background workloads.
    background: bool = False
    # Fault injection (simulator): the job fails once its progress
    # reaches this iteration — exercises mid-run failure handling
    # (resource release, cluster continues) in scheduler studies.
    fail_at_iter: int = 0       # 0 = never

    def __post_init__(self) -> None:
        if self.n_gpus < 1:
            raise ValueError(f"job {self.job_id}: n_gpus must be >= 1")
        if self.total_iters < 1:
            raise ValueError(f"job {self.job_id}: total_iters must be >= 1")
        if self.elastic and not (1 <= self.min_gpus <= self.max_gpus):
            raise ValueError(f"job {self.job_id}: bad elastic bounds")


@dataclass
class Job:
    """Mutable scheduler-side view of a job."""

    spec: JobSpec
    state: JobState = JobState.QUEUED
    placement: List[int] = field(default_factory=list)  # global GPU ids
    done_iters: int = 0
    attained_service: float = 0.0   # GPU-seconds of service received
    tokens_done: float = 0.0        # tokens/samples processed (all GPUs)
    queue_level: int = 0            # LAS/Tiresias MLFQ level
    submit_t: float = 0.0
    first_start_t: Optional[float] = None
    finish_t: Optional[float] = None
    last_sched_t: Optional[float] = None  # last (re)placement time
    preemptions: int = 0
    migrations: int = 0
    resizes: int = 0
    # Measured per-iteration wall time (real backend) or hint (sim).
    iter_time: float = 0.0
    # Set when the job's GPU state was evicted to a host checkpoint
    # (as opposed to suspended-in-HBM).
    evicted: bool = False

    @property
    def job_id(self) -> str:
        return self.spec.job_id

    @property
    def n_gpus(self) -> int:
        """Current gang width (elastic jobs: width of current placement
        when running, else the spec width clamped to elastic bounds)."""
        if self.placement:
            return len(self.placement)
        return self.spec.n_gpus

    @property
    def remaining_iters(self) -> int:
        return max(0, self.spec.total_iters - self.done_iters)

    def remaining_time(self, default_iter_time: float = 1.0) -> float:
        """Remaining-service estimate in seconds (for SRTF)."""
        it = self.iter_time or self.spec.iter_time_hint or default_iter_time
        return self.remaining_iters * it

    @property
    def is_active(self) -> bool:
        return self.state in (JobState.QUEUED, JobState.RUNNING,
                              JobState.SUSPENDED)

    @property
    def jct(self) -> Optional[float]:
        if self.finish_t is None:
            return None
        return self.finish_t - self.submit_t
