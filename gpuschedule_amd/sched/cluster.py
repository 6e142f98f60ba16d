"""Cluster state model and HBM-aware gang placement.

MI355X-first placement rules:

* Each GPU has 288 GB of HBM3E (``MI355X_HBM_BYTES``).  Suspended jobs
  may stay RESIDENT in HBM (Gandiva-style suspend/resume without any
  copy) as long as the sum of resident jobs' footprints fits; the
  scheduler only evicts (host checkpoint) under memory pressure.  The
  cluster model therefore tracks *resident* memory per GPU separately
  from which job is *executing*.
* Multi-GPU gangs are placed on CONTIGUOUS global GPU indices within one
  node.  On a single MI355X node the xGMI fabric is point-to-point
  (7 links/GPU) and RCCL ring collectives are per-link bound; contiguous
  ranges also let the real backend pre-create every possible RCCL
  subgroup (O(N^2) of them) instead of racing dynamic group creation.
* Exactly one job executes on a GPU at a time (time-slicing = swapping
  which resident job is marked running).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Set

from ..trace.job import MI355X_HBM_BYTES, Job


@dataclass
class GPUState:
    gpu_id: int                  # global index
    node_id: int
    hbm_bytes: int = MI355X_HBM_BYTES
    # jobs currently EXECUTING here (normally one; Gandiva packing may
    # co-run two on separate HIP streams)
    running_set: Set[str] = field(default_factory=set)
    resident: Set[str] = field(default_factory=set)  # jobs with state in HBM
    resident_bytes: int = 0

    @property
    def running(self) -> Optional[str]:
        """First running job (compat accessor; None when idle)."""
        return next(iter(self.running_set), None)

    @property
    def free_bytes(self) -> int:
        return self.hbm_bytes - self.resident_bytes


class Cluster:
    """State of all GPUs; placement and memory accounting."""

    def __init__(self, n_gpus: int, gpus_per_node: int = 8,
                 hbm_bytes: int = MI355X_HBM_BYTES):
        self.n_gpus = n_gpus
        self.gpus_per_node = gpus_per_node
        self.gpus = [GPUState(i, i // gpus_per_node, hbm_bytes)
                     for i in range(n_gpus)]
        self._mem: Dict[str, int] = {}  # job_id -> per-GPU bytes

    # ---- queries -------------------------------------------------

    def gpu(self, i: int) -> GPUState:
        return self.gpus[i]

    def free_gpus(self) -> List[int]:
        return [g.gpu_id for g in self.gpus if not g.running_set]

    def n_free(self) -> int:
        return sum(1 for g in self.gpus if not g.running_set)

    def running_jobs(self) -> Set[str]:
        out: Set[str] = set()
        for g in self.gpus:
            out |= g.running_set
        return out

    def running_on(self, gpu_id: int) -> Set[str]:
        return set(self.gpus[gpu_id].running_set)

    def utilization(self) -> float:
        return 1.0 - self.n_free() / max(1, self.n_gpus)

    # ---- placement search ----------------------------------------

    def find_placement(self, n: int, mem_bytes: int,
                       resident_ok: Optional[Set[str]] = None,
                       prefer: Optional[Sequence[int]] = None) -> Optional[List[int]]:
        """Find ``n`` contiguous idle GPUs (within one node) with
        ``mem_bytes`` of free HBM each.

        ``resident_ok``: job_ids whose resident memory may be ignored
        (e.g. the job being placed is already resident there).
        ``prefer``: a previous placement to reuse if still legal
        (resume in place without migration).

        Best-fit: among legal windows pick the one whose GPUs have the
        least total free memory (pack tight, keep big holes for big
        gangs).  Returns global GPU ids or None.
        """
        resident_ok = resident_ok or set()

        def window_ok(ids: Sequence[int]) -> bool:
            for i in ids:
                g = self.gpus[i]
                if g.running_set:
                    return False
                extra = sum(self._mem.get(j, 0) for j in g.resident
                            if j in resident_ok)
                if g.free_bytes + extra < mem_bytes:
                    return False
            return True

        if prefer and len(prefer) == n and window_ok(prefer):
            return list(prefer)

        best: Optional[List[int]] = None
        best_score = None
        for node_start in range(0, self.n_gpus, self.gpus_per_node):
            node_end = min(node_start + self.gpus_per_node, self.n_gpus)
            for s in range(node_start, node_end - n + 1):
                ids = list(range(s, s + n))
                if not window_ok(ids):
                    continue
                score = sum(self.gpus[i].free_bytes for i in ids)
                if best_score is None or score < best_score:
                    best, best_score = ids, score
        return best

    def can_fit_with_migration(self, n: int, mem_bytes: int) -> bool:
        """Would ``n`` GPUs with enough memory exist if running singles
        could be moved?  (Gandiva migration trigger.)"""
        ok = 0
        for node_start in range(0, self.n_gpus, self.gpus_per_node):
            node_end = min(node_start + self.gpus_per_node, self.n_gpus)
            ok = max(ok, sum(1 for i in range(node_start, node_end)
                             if not self.gpus[i].running_set
                             and self.gpus[i].free_bytes >= mem_bytes))
        return ok >= n

    # ---- mutation ------------------------------------------------

    def place(self, job: Job, gpus: List[int], mem_bytes: int,
              capacity: int = 1) -> None:
        """Mark job executing on ``gpus``; make it resident there.
        ``capacity``: max co-running jobs per GPU (Gandiva packing)."""
        self._mem[job.job_id] = mem_bytes
        for i in gpus:
            g = self.gpus[i]
            assert len(g.running_set) < capacity, \
                f"GPU {i} full ({g.running_set}), placing {job.job_id}"
            g.running_set.add(job.job_id)
            if job.job_id not in g.resident:
                g.resident.add(job.job_id)
                g.resident_bytes += mem_bytes
                assert g.resident_bytes <= g.hbm_bytes, \
                    f"GPU {i} HBM over-committed placing {job.job_id}"

    def stop(self, job: Job, evict: bool = False) -> None:
        """Stop executing; keep resident unless ``evict``."""
        for i in job.placement:
            g = self.gpus[i]
            g.running_set.discard(job.job_id)
            if evict and job.job_id in g.resident:
                g.resident.remove(job.job_id)
                g.resident_bytes -= self._mem.get(job.job_id, 0)

    def evict_resident(self, job_id: str) -> None:
        """Drop a (non-running) job's resident footprint everywhere."""
        for g in self.gpus:
            if job_id not in g.running_set and job_id in g.resident:
                g.resident.remove(job_id)
                g.resident_bytes -= self._mem.get(job_id, 0)

    def release(self, job: Job) -> None:
        """Job finished/failed: free everything."""
        self.stop(job, evict=True)
        self.evict_resident(job.job_id)
        self._mem.pop(job.job_id, None)

    def job_mem(self, job_id: str) -> int:
        return self._mem.get(job_id, 0)

    def update_mem(self, job_id: str, measured: int) -> None:
        """Refine a job's footprint from the backend's measured HBM use
        (catalog numbers are conservative priors; measured values give
        tighter bin-packing).  Floor at 256 MB."""
        old = self._mem.get(job_id)
        new = max(measured, 256 << 20)
        if old is None or new == old:
            return
        delta = new - old
        for g in self.gpus:
            if job_id in g.resident:
                g.resident_bytes += delta
        self._mem[job_id] = new
