"""Discrete-event simulator backend (the pure-CPU "plumbing" path).

Shares the scheduler core + policies with the real Torch backend; only
time and job execution are virtual.  Each running job progresses at
``catalog.sim_iter_time(spec, width)`` seconds/iteration; placements
pay modeled one-time overheads:

* cold start (first placement):        ``COLD_START_S``
* resume from HBM-resident state:      ``RESUME_HBM_S``  (Gandiva cheap path)
* restore from a host checkpoint:      mem / ``H2D_BYTES_PER_S``
* migration (resident elsewhere):      D2H + H2D of the checkpoint

The loop advances the virtual clock to the next event: arrival, job
completion, or the periodic re-evaluation boundary (``reeval_s``, which
also bounds LAS demotion / Gandiva quantum latency).
"""
from __future__ import annotations

from typing import Dict, Optional

from .. import catalog
from ..sched.events import EventLog
from ..sched.scheduler import Plan, Scheduler
from ..trace.job import JobState

COLD_START_S = 0.5
RESUME_HBM_S = 0.05
D2H_BYTES_PER_S = 48e9   # pinned hipMemcpyAsync, side stream
H2D_BYTES_PER_S = 48e9
XGMI_BYTES_PER_S = 150e9  # per-link broadcast (overlap-resize path)


class Simulator:
    def __init__(self, scheduler: Scheduler, reeval_s: float = 1.0,
                 max_time: float = 10_000_000.0,
                 warm_pool: bool = False):
        self.sched = scheduler
        self.reeval_s = reeval_s
        # warm_pool models the real backend's prewarmed runner pool
        # (exec/trainer.py RunnerPool): a job whose model family has a
        # pooled runner adopts it in-place instead of paying model
        # construction + hipGraph capture + MIOpen find, so first
        # placement costs ~RESUME_HBM_S, not COLD_START_S.  Measured:
        # scripts/sim_fidelity.py (profiles/sim_fidelity.json).
        self.warm_pool = warm_pool
        self.max_time = max_time
        self.now = 0.0
        self._frac: Dict[str, float] = {}      # fractional iterations
        self._overhead: Dict[str, float] = {}  # startup seconds left

    # -- overhead model -------------------------------------------

    def _placement_overhead(self, p, job) -> float:
        mem = self.sched.cluster.job_mem(p.job_id) or \
            catalog.job_mem_bytes(job.spec)
        ckpt = mem / 3  # checkpoint = weights+optimizer, not activations
        if (p.resize_from_gpus and not p.from_evicted
                and set(p.resize_from_gpus) & set(p.gpus)):
            # overlap resize: state broadcast over xGMI from a
            # surviving rank (exec/backend.py), no host round-trip
            return ckpt / XGMI_BYTES_PER_S
        if p.migrate_from is not None or p.resize_from is not None:
            return ckpt / D2H_BYTES_PER_S + ckpt / H2D_BYTES_PER_S
        if p.from_evicted:
            return ckpt / H2D_BYTES_PER_S
        if p.resume:
            return RESUME_HBM_S
        return RESUME_HBM_S if self.warm_pool else COLD_START_S

    def _apply_plan_overheads(self, plan: Plan) -> None:
        for p in plan.places:
            job = self.sched.jobs[p.job_id]
            self._overhead[p.job_id] = self._placement_overhead(p, job)

    # -- main loop ------------------------------------------------

    def run(self) -> EventLog:
        sched = self.sched
        while not sched.all_done and self.now < self.max_time:
            plan = sched.tick(self.now)
            self._apply_plan_overheads(plan)
            self._advance()
        return sched.log

    def _advance(self) -> None:
        sched = self.sched
        running = [j for j in sched.jobs.values()
                   if j.state == JobState.RUNNING]
        # Next event time.
        horizon = self.now + self.reeval_s
        arr = sched.next_arrival()
        if arr is not None:
            horizon = min(horizon, max(arr, self.now + 1e-9))
        def eff_iter_time(j):
            it = catalog.sim_iter_time(j.spec, len(j.placement))
            if any(len(sched.cluster.gpu(g).running_set) > 1
                   for g in j.placement):
                it *= catalog.PACK_SLOWDOWN
            return it

        for j in running:
            it = eff_iter_time(j)
            left = (j.remaining_iters - self._frac.get(j.job_id, 0.0)) * it
            fin = self.now + self._overhead.get(j.job_id, 0.0) + max(left, 0)
            horizon = min(horizon, fin)
        if not running and arr is None:
            # Active jobs exist but none runnable and no arrivals: just
            # step the re-evaluation clock.
            horizon = self.now + self.reeval_s
        # Floor on dt: float rounding must not let progress converge
        # asymptotically toward an iteration boundary.
        dt = max(horizon - self.now, 1e-6)

        done = []
        for j in running:
            avail = dt
            ov = self._overhead.get(j.job_id, 0.0)
            if ov > 0:
                used = min(ov, avail)
                self._overhead[j.job_id] = ov - used
                avail -= used
            if avail <= 0:
                continue
            it = eff_iter_time(j)
            frac = self._frac.get(j.job_id, 0.0) + avail / it + 1e-9
            whole = min(int(frac), j.remaining_iters)
            self._frac[j.job_id] = max(frac - int(frac) - 1e-9, 0.0)
            service = avail * len(j.placement)
            done.append((j.job_id, whole, service, it))
        self.now += dt
        for job_id, whole, service, it in done:
            job = sched.jobs[job_id]
            fail_at = job.spec.fail_at_iter
            if fail_at and job.done_iters + whole >= fail_at:
                # fault injection: partial progress up to the failure
                # point, then the job dies mid-run.  Attained service
                # is scaled to the fraction of this slice's iterations
                # actually completed before the fault.
                completed = max(fail_at - job.done_iters - 1, 0)
                frac_done = completed / whole if whole > 0 else 0.0
                sched.on_progress(job_id, completed,
                                  service * frac_done, self.now,
                                  iter_time=it)
                sched.on_fail(job_id, "injected fault "
                              f"(fail_at_iter={fail_at})", self.now)
                self._frac.pop(job_id, None)
                self._overhead.pop(job_id, None)
                continue
            finished = sched.on_progress(job_id, whole, service,
                                         self.now, iter_time=it)
            if finished:
                self._frac.pop(job_id, None)
                self._overhead.pop(job_id, None)


def simulate(specs, policy, n_gpus: int = 8, gpus_per_node: int = 8,
             reeval_s: float = 1.0, log: Optional[EventLog] = None,
             warm_pool: bool = False):
    """Convenience: simulate a trace under a policy; returns (log, summary)."""
    from ..sched.cluster import Cluster
    sched = Scheduler(Cluster(n_gpus, gpus_per_node), policy, log=log)
    sched.submit(list(specs))
    sim = Simulator(sched, reeval_s=reeval_s, warm_pool=warm_pool)
    out = sim.run()
    return out, out.summary()
