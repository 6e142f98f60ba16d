"""Scheduler core: admission -> ranking -> placement -> action plan.

Backend-agnostic.  Each tick:

1. admit arrivals whose submit_time has passed,
2. let the policy rank active jobs (``policies/base.py``),
3. allocate GPUs in rank order against a shadow of the cluster
   (contiguous windows, HBM bin-packing, elastic shrink-to-fit,
   eviction under memory pressure, compact/migration retry),
4. diff against current state into a :class:`Plan` of preempts,
   evictions, placements (with resume/migrate/resize provenance),
5. apply the plan to the cluster model and the event log.

The backend (simulator or Torch) then EXECUTES the plan and reports
progress/completions back via :meth:`on_progress` / :meth:`on_fail`.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Set, Tuple

from .. import catalog
from ..trace.job import Job, JobSpec, JobState
from .cluster import Cluster
from .events import EventLog
from .policies.base import Policy


@dataclass
class Placement:
    job_id: str
    gpus: List[int]
    resume: bool = False                  # was SUSPENDED (state exists)
    from_evicted: bool = False            # state must come from host ckpt
    migrate_from: Optional[List[int]] = None
    resize_from: Optional[int] = None
    # old placement of a resize; when it overlaps the new gpus, the
    # backend moves state by RCCL broadcast from a surviving rank
    # instead of a host-checkpoint round-trip
    resize_from_gpus: Optional[List[int]] = None


@dataclass
class Plan:
    """One tick's worth of actions, in execution order."""
    preempts: List[Tuple[str, str]] = field(default_factory=list)  # (job, reason)
    evictions: List[str] = field(default_factory=list)
    places: List[Placement] = field(default_factory=list)

    @property
    def empty(self) -> bool:
        return not (self.preempts or self.evictions or self.places)


class Scheduler:
    def __init__(self, cluster: Cluster, policy: Policy,
                 log: Optional[EventLog] = None):
        self.cluster = cluster
        self.policy = policy
        self.log = log if log is not None else EventLog()
        self.jobs: Dict[str, Job] = {}
        self._pending: List[JobSpec] = []   # sorted by submit_time
        self._ticks = 0

    # ---- submission ---------------------------------------------

    def submit(self, specs: Sequence[JobSpec]) -> None:
        self._pending.extend(specs)
        self._pending.sort(key=lambda s: (s.submit_time, s.job_id))

    def submit_now(self, spec: JobSpec, now: float) -> None:
        """Admit immediately (used by interactive API)."""
        self._admit(spec, now)

    def _admit(self, spec: JobSpec, now: float) -> None:
        job = Job(spec=spec, submit_t=max(now, spec.submit_time))
        self.jobs[spec.job_id] = job
        # ideal_s: the job's contention-free solo runtime (for the
        # finish-time-slowdown metric in EventLog.summary); unknown
        # models are rejected just below — log their submit without it
        try:
            ideal = spec.total_iters * catalog.sim_iter_time(
                spec, spec.n_gpus)
        except ValueError:
            ideal = 0.0
        self.log.emit(job.submit_t, "submit", spec.job_id,
                      n_gpus=spec.n_gpus, model=spec.model,
                      iters=spec.total_iters,
                      ideal_s=round(ideal, 6))
        # Reject unsatisfiable jobs at admission (they would otherwise
        # starve in the queue forever).
        width = spec.min_gpus if spec.elastic else spec.n_gpus
        try:
            mem = catalog.job_mem_bytes(spec)
        except ValueError as e:  # unknown model family
            self.on_fail(spec.job_id, str(e), job.submit_t)
            return
        too_wide = width > min(self.cluster.n_gpus,
                               self.cluster.gpus_per_node)
        too_big = mem > max(g.hbm_bytes for g in self.cluster.gpus)
        if too_wide or too_big:
            reason = ("gang wider than cluster/node" if too_wide
                      else "per-GPU memory exceeds HBM")
            self.on_fail(spec.job_id, f"unsatisfiable: {reason}",
                         job.submit_t)

    def next_arrival(self) -> Optional[float]:
        return self._pending[0].submit_time if self._pending else None

    @property
    def all_done(self) -> bool:
        return not self._pending and not any(
            j.is_active for j in self.jobs.values())

    def active(self) -> List[Job]:
        return [j for j in self.jobs.values() if j.is_active]

    # ---- the tick -----------------------------------------------

    def tick(self, now: float) -> Plan:
        while self._pending and self._pending[0].submit_time <= now:
            self._admit(self._pending.pop(0), now)

        if hasattr(self.policy, "_total_gpus"):
            self.policy._total_gpus = self.cluster.n_gpus
        self.policy.on_tick(now, self.jobs)
        requests = self.policy.rank(now, self.jobs)
        # Best-effort class ranks strictly below normal jobs (stable).
        requests = ([r for r in requests
                     if not self.jobs[r[0]].spec.background] +
                    [r for r in requests
                     if self.jobs[r[0]].spec.background])

        alloc, evictions = self._allocate(requests)
        plan = self._diff(alloc, evictions, now)
        self._apply(plan, now)
        self._ticks += 1
        if self._ticks % 16 == 1:  # periodic cluster utilization sample
            self.log.emit(now, "tick",
                          util=round(self.cluster.utilization(), 3),
                          running=len(self.cluster.running_jobs()),
                          queued=sum(1 for j in self.jobs.values()
                                     if j.state == JobState.QUEUED))
        return plan

    # ---- allocation ---------------------------------------------

    def _allocate(self, requests) -> Tuple[Dict[str, List[int]], List[str]]:
        alloc, evictions, failed_gang = self._allocate_pass(
            requests, stable=True)
        if failed_gang and self.policy.migration:
            # Gandiva defrag: re-place everything compactly, allowing
            # running jobs to move (backend pays a migration).
            alloc2, evictions2, failed2 = self._allocate_pass(
                requests, stable=False)
            if len(alloc2) > len(alloc):
                return alloc2, evictions2
        return alloc, evictions

    def _allocate_pass(self, requests, stable: bool):
        """One greedy pass in rank order over a shadow cluster."""
        C = self.cluster
        cap = self.policy.pack_capacity if self.policy.packing else 1
        slots: Dict[int, int] = {}  # gpu -> allocated run slots
        busy = slots  # truthiness helper below uses slot counts
        # Shadow per-GPU memory delta: newly placed non-resident jobs
        # add, evictions subtract.
        extra: Dict[int, int] = {}
        evicted: Set[str] = set()
        alloc: Dict[str, List[int]] = {}
        failed_gang = False

        def occupy(ids):
            for i in ids:
                slots[i] = slots.get(i, 0) + 1

        pinned: Set[str] = set()
        if not self.policy.preemptive:
            # Running jobs cannot be preempted: pin them first.
            for j in self.jobs.values():
                if j.state == JobState.RUNNING:
                    alloc[j.job_id] = list(j.placement)
                    occupy(j.placement)
                    pinned.add(j.job_id)

        def free_mem(g: int, job: Job) -> int:
            gs = C.gpu(g)
            used = gs.resident_bytes + extra.get(g, 0)
            for jid in gs.resident:
                if jid in evicted or jid == job.job_id:
                    used -= C.job_mem(jid)
            return gs.hbm_bytes - used

        def evictable(g: int) -> List[str]:
            # Resident jobs that are not allocated this tick (running
            # ones will be preempted by the diff) may be evicted to a
            # host checkpoint to make room.
            out = []
            for jid in C.gpu(g).resident:
                j = self.jobs.get(jid)
                if (j is not None and j.is_active
                        and jid not in alloc and jid not in evicted):
                    out.append(jid)
            # LRU: least-recently scheduled first
            out.sort(key=lambda jid: self.jobs[jid].last_sched_t or 0.0)
            return out

        def try_window(ids: List[int], job: Job, mem: int,
                       allow_evict: bool, max_slots: int = 1) -> bool:
            if any(slots.get(i, 0) >= max_slots for i in ids):
                return False
            need_evict: List[str] = []
            for i in ids:
                fm = free_mem(i, job)
                if fm >= mem:
                    continue
                if not allow_evict:
                    return False
                for jid in evictable(i):
                    need_evict.append(jid)
                    fm += C.job_mem(jid)
                    if fm >= mem:
                        break
                if fm < mem:
                    return False
            for jid in need_evict:
                evicted.add(jid)
            return True

        def windows(n: int) -> List[List[int]]:
            out = []
            for ns in range(0, C.n_gpus, C.gpus_per_node):
                ne = min(ns + C.gpus_per_node, C.n_gpus)
                for s in range(ns, ne - n + 1):
                    out.append(list(range(s, s + n)))
            return out

        def place(job: Job, width: int, allow_evict: bool) -> bool:
            mem = C.job_mem(job.job_id) or catalog.job_mem_bytes(job.spec)
            # Preference 1: keep current placement (running, stable).
            if (stable and job.state == JobState.RUNNING
                    and len(job.placement) == width
                    and try_window(list(job.placement), job, mem,
                                   allow_evict)):
                alloc[job.job_id] = list(job.placement)
                occupy(job.placement)
                return True
            # Preference 2: resume where resident (suspended in HBM).
            if stable and job.placement and len(job.placement) == width:
                if try_window(list(job.placement), job, mem, allow_evict):
                    alloc[job.job_id] = list(job.placement)
                    occupy(job.placement)
                    return True
            # Preference 3: best-fit contiguous window (least free mem).
            best, best_score = None, None
            for ids in windows(width):
                if try_window(ids, job, mem, allow_evict=False):
                    score = sum(free_mem(i, job) for i in ids)
                    if best_score is None or score < best_score:
                        best, best_score = ids, score
            if best is None and allow_evict:
                for ids in windows(width):
                    if try_window(ids, job, mem, allow_evict=True):
                        best = ids
                        break
            # Preference 4 (Gandiva packing): co-run a 1-GPU job on the
            # least-loaded busy GPU with memory headroom.
            if best is None and cap > 1 and width == 1:
                cands = sorted(
                    (i for i in range(C.n_gpus)
                     if slots.get(i, 0) < cap),
                    key=lambda i: (slots.get(i, 0), -free_mem(i, job)))
                for i in cands:
                    if try_window([i], job, mem, allow_evict=False,
                                  max_slots=cap):
                        best = [i]
                        break
            if best is None:
                return False
            alloc[job.job_id] = best
            occupy(best)
            # Record the shadow memory delta: jobs placed later this
            # tick must see this job's footprint on GPUs where it is
            # not already resident (free_mem only reads the cluster's
            # committed residency + `extra`).
            for i in best:
                if job.job_id not in C.gpu(i).resident:
                    extra[i] = extra.get(i, 0) + mem
            return True

        for job_id, width in requests:
            job = self.jobs.get(job_id)
            if job is None or not job.is_active or job_id in pinned:
                continue
            if sum(slots.values()) >= C.n_gpus * cap:
                break
            widths = [width]
            if self.policy.elastic and job.spec.elastic:
                widths = list(range(width, job.spec.min_gpus - 1, -1))
            ok = False
            n_full = sum(1 for v in slots.values() if v >= 1)
            for w in widths:
                if w > C.n_gpus or (w > 1 and n_full + w > C.n_gpus):
                    continue
                if place(job, w, allow_evict=True):
                    ok = True
                    break
            if not ok:
                if width > 1:
                    failed_gang = True
                if self.policy.strict_order:
                    break
        return alloc, sorted(evicted), failed_gang

    # ---- diff & apply -------------------------------------------

    def _diff(self, alloc: Dict[str, List[int]], evictions: List[str],
              now: float) -> Plan:
        plan = Plan()
        for job in self.jobs.values():
            if job.state != JobState.RUNNING:
                continue
            new = alloc.get(job.job_id)
            if new is None:
                plan.preempts.append((job.job_id, "policy"))
            elif new != job.placement:
                reason = ("resize" if len(new) != len(job.placement)
                          else "migrate")
                plan.preempts.append((job.job_id, reason))
        preempting = {p[0] for p in plan.preempts}
        plan.evictions = [j for j in evictions
                          if self.jobs[j].state == JobState.SUSPENDED
                          or j in preempting]
        for job_id, gpus in alloc.items():
            job = self.jobs[job_id]
            if job.state == JobState.RUNNING and gpus == job.placement:
                continue
            had_state = job.done_iters > 0 or job.state == JobState.SUSPENDED
            p = Placement(job_id=job_id, gpus=gpus,
                          resume=had_state,
                          from_evicted=job.evicted or job_id in plan.evictions)
            if had_state and job.placement and gpus != job.placement:
                if len(gpus) != len(job.placement):
                    p.resize_from = len(job.placement)
                    p.resize_from_gpus = list(job.placement)
                else:
                    p.migrate_from = list(job.placement)
            plan.places.append(p)
        return plan

    def _apply(self, plan: Plan, now: float) -> None:
        C = self.cluster
        for job_id, reason in plan.preempts:
            job = self.jobs[job_id]
            C.stop(job)
            job.state = JobState.SUSPENDED
            job.preemptions += 1
            self.log.emit(now, "preempt", job_id, reason=reason)
        for job_id in plan.evictions:
            job = self.jobs[job_id]
            C.evict_resident(job_id)
            job.evicted = True
            self.log.emit(now, "evict", job_id)
        for p in plan.places:
            job = self.jobs[p.job_id]
            mem = C.job_mem(p.job_id) or catalog.job_mem_bytes(job.spec)
            if p.migrate_from is not None:
                C.evict_resident(p.job_id)
                job.migrations += 1
                self.log.emit(now, "migrate", p.job_id,
                              src=p.migrate_from, dst=p.gpus)
            if p.resize_from is not None:
                C.evict_resident(p.job_id)
                job.resizes += 1
                self.log.emit(now, "resize", p.job_id,
                              old_gpus=p.resize_from, new_gpus=len(p.gpus))
            C.place(job, p.gpus, mem,
                    capacity=(self.policy.pack_capacity
                              if self.policy.packing else 1))
            job.placement = list(p.gpus)
            job.state = JobState.RUNNING
            job.last_sched_t = now
            job.evicted = False
            if job.first_start_t is None:
                job.first_start_t = now
                self.log.emit(now, "start", p.job_id, gpus=p.gpus,
                              first=True)
            else:
                self.log.emit(now, "resume", p.job_id, gpus=p.gpus)

    # ---- backend feedback ---------------------------------------

    def on_progress(self, job_id: str, iters: int, service: float,
                    now: float, iter_time: Optional[float] = None) -> bool:
        """Record progress; returns True when the job completed."""
        job = self.jobs[job_id]
        if not job.is_active:
            # partial gang failure: a member's "fail" report can land
            # before the leader's "ok" report for the same tick — the
            # job is already FAILED (resources released); accepting
            # progress now would double-release and emit complete
            # after fail.
            return False
        job.done_iters += iters
        job.attained_service += service
        job.tokens_done += iters * catalog.job_tokens_per_iter(
            job.spec, max(len(job.placement), 1))
        if iter_time:
            job.iter_time = iter_time
        if job.done_iters >= job.spec.total_iters:
            self.cluster.release(job)
            job.state = JobState.DONE
            job.finish_t = now
            job.placement = []
            self.log.emit(now, "complete", job_id, jct=job.jct,
                          iters=job.done_iters,
                          tokens=int(job.tokens_done),
                          tokens_per_s=round(
                              job.tokens_done / max(job.jct, 1e-9), 1))
            return True
        return False

    def on_fail(self, job_id: str, err: str, now: float) -> None:
        job = self.jobs[job_id]
        if not job.is_active:
            return
        self.cluster.release(job)
        job.state = JobState.FAILED
        job.finish_t = now
        job.placement = []
        self.log.emit(now, "fail", job_id, error=err)
