"""Real execution backend: one rank per GPU, synchronous tick protocol.

Topology (single MI355X node, driver-launched via torchrun):

* default process group: **gloo** (control plane — tiny pickled
  messages; RCCL stays free for gang collectives),
* gang collectives: pre-created contiguous-range subgroups on the
  **nccl** backend (= RCCL over xGMI) from exec/gangs.py,
* rank r owns GPU r (``cuda:LOCAL_RANK``); rank 0 additionally runs the
  Scheduler (sched/scheduler.py) against wall-clock time.

Tick protocol (every rank, lock-step):

1. rank 0: ``plan = scheduler.tick(now)``; broadcast TickMsg
   {stop, plan, per-job iteration quota, job metadata, cleanup list}.
2. apply phase A: preempts (migrate/resize → gang-leader writes a node
   -local checkpoint, runners released; plain suspends stay resident in
   HBM), evictions (checkpoint + free);  gloo barrier (checkpoints must
   be on disk before phase B reads them).
3. apply phase B: placements — reuse the resident runner, or build a
   replica (seeded identical across the gang) and load the checkpoint.
4. run the iteration quota for locally hosted jobs (gang members step
   collectively: bucketed all-reduce inside JobRunner.step).
5. gather per-job progress/failures to rank 0; scheduler accounts
   service, completes/fails jobs.
"""
from __future__ import annotations

import os
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .. import catalog
from ..sched.scheduler import Plan, Scheduler
from .checkpoint import CheckpointStore
from .gangs import GangGroups
from .trainer import RUNNER_POOL, JobRunner


@dataclass
class TickMsg:
    stop: bool = False
    plan: Optional[Plan] = None
    iters: Dict[str, int] = field(default_factory=dict)
    # job_id -> (model, batch, gpus, done_iters)
    meta: Dict[str, tuple] = field(default_factory=dict)
    cleanup: List[str] = field(default_factory=list)


class TorchBackend:
    def __init__(self, scheduler: Optional[Scheduler], rank: int = 0,
                 world: int = 1, device_type: str = "cuda",
                 dtype=torch.bfloat16, iters_per_tick: int = 1,
                 ckpt_root: Optional[str] = None,
                 gang_backend: Optional[str] = None):
        self.sched = scheduler  # only rank 0 holds one
        self.rank = rank
        self.world = world
        self.device_type = device_type
        self.device = (f"cuda:{rank}" if device_type == "cuda" else "cpu")
        if device_type == "cuda":
            torch.cuda.set_device(self.device)
        self.dtype = dtype if device_type == "cuda" else torch.float32
        self.iters_per_tick = iters_per_tick
        self.ckpt = CheckpointStore(ckpt_root)
        self.gangs = GangGroups(world, backend=gang_backend)
        self.runners: Dict[str, JobRunner] = {}
        self._t0 = time.perf_counter()
        self._pending_cleanup: List[str] = []
        self.ticks = 0

    # ---- tiny comm helpers --------------------------------------

    def _bcast(self, obj):
        if self.world == 1:
            return obj
        box = [obj]
        dist.broadcast_object_list(box, src=0)
        return box[0]

    def _gather(self, obj):
        if self.world == 1:
            return [obj]
        out = [None] * self.world if self.rank == 0 else None
        dist.gather_object(obj, out, dst=0)
        return out

    def _barrier(self):
        if self.world > 1:
            dist.barrier()

    def now(self) -> float:
        return time.perf_counter() - self._t0

    # ---- plan application ---------------------------------------

    @staticmethod
    def _bcast_resizes(plan: Plan) -> Dict[str, "Placement"]:
        """Resizes whose old placement overlaps the new one: state
        moves by RCCL broadcast from a surviving rank (no checkpoint
        file) — see parallel/ddp.broadcast_state."""
        out = {}
        for p in plan.places:
            if (p.resize_from_gpus and not p.from_evicted
                    and set(p.resize_from_gpus) & set(p.gpus)):
                out[p.job_id] = p
        return out

    def _apply_phase_a(self, plan: Plan, meta: Dict[str, tuple]) -> None:
        bcast = self._bcast_resizes(plan)
        for job_id, reason in plan.preempts:
            r = self.runners.get(job_id)
            if r is None:
                continue
            if reason == "resize" and job_id in bcast:
                continue  # survivors keep state; joiners get broadcast
            if reason in ("migrate", "resize"):
                if r.gang_rank == 0:
                    self.ckpt.save(job_id, r.model, r.optim, r.done_iters)
                r.release()
                del self.runners[job_id]
            # reason "policy": suspend-in-HBM — runner stays resident.
        # SUSPENDED (HBM-resident) jobs that the scheduler migrates or
        # resizes never appear in plan.preempts (only RUNNING jobs do);
        # they arrive as placements with migrate_from/resize_from set.
        # Their resident runner holds the ONLY copy of the training
        # state — persist it before phase B drops the stale runner,
        # otherwise the destination rank rebuilds from initial weights.
        preempted = {job_id for job_id, _ in plan.preempts}
        for p in plan.places:
            if p.migrate_from is None and p.resize_from is None:
                continue
            if p.job_id in preempted or p.job_id in bcast:
                continue  # already checkpointed / broadcast path
            r = self.runners.get(p.job_id)
            if r is None:
                continue
            if r.gang_rank == 0 and not p.from_evicted:
                self.ckpt.save(p.job_id, r.model, r.optim, r.done_iters)
            r.release()
            del self.runners[p.job_id]
        for job_id in plan.evictions:
            r = self.runners.get(job_id)
            if r is not None:
                if r.gang_rank == 0:
                    self.ckpt.save(job_id, r.model, r.optim, r.done_iters)
                r.release()
                del self.runners[job_id]
                if self.device_type == "cuda":
                    torch.cuda.empty_cache()

    def _apply_phase_b(self, plan: Plan, meta: Dict[str, tuple]) -> None:
        bcast = self._bcast_resizes(plan)
        for p in plan.places:
            if self.rank not in p.gpus:
                # if we host a stale resident copy (migrated away or
                # shrunk off this rank), drop it
                r = self.runners.pop(p.job_id, None)
                if r is not None and not RUNNER_POOL.put(r):
                    r.release()
                continue
            model_name, batch, gpus, done_iters = meta[p.job_id]
            runner = self.runners.get(p.job_id)
            if p.job_id in bcast:
                # overlap resize: swap gang in place on survivors,
                # fresh-build (or adopt) on joiners, then collectively
                # broadcast the full training state from a survivor.
                from ..parallel.ddp import broadcast_state
                group = self.gangs.get(gpus)
                key = (model_name, batch, tuple(gpus))
                old = p.resize_from_gpus
                if runner is None:
                    pooled = RUNNER_POOL.get(key)
                    if pooled is not None:
                        pooled.adopt(p.job_id)
                        runner = pooled
                    else:
                        runner = self._build_runner(
                            p.job_id, model_name, batch, group, gpus,
                            0, None, key)
                    self.runners[p.job_id] = runner
                # every gang member (survivor, joiner, pool-adopted)
                # rebinds to the SAME new group object; the pool key
                # must follow (a later put/adopt under the OLD gang's
                # key would hand out a runner whose reducer still
                # all-reduces with the new gang — phantom partners)
                runner.regang(group, gpus.index(self.rank), len(gpus))
                runner.pool_key = key
                src = next(g for g in gpus if g in old)
                broadcast_state(runner.model, runner.optim, group, src)
                runner.done_iters = done_iters
                continue
            if runner is not None:
                continue  # resume-in-place: state resident in HBM
            state = None
            if self.ckpt.exists(p.job_id):
                state = self.ckpt.load(p.job_id)
            group = self.gangs.get(gpus)
            key = (model_name, batch, tuple(gpus))
            if state is None and done_iters == 0:
                # fresh job: adopt a warm pooled runner (same model /
                # batch / gang ⇒ same graph, optimizer tables, stream)
                pooled = RUNNER_POOL.get(key)
                if pooled is not None:
                    pooled.adopt(p.job_id)
                    self.runners[p.job_id] = pooled
                    continue
            self.runners[p.job_id] = self._build_runner(
                p.job_id, model_name, batch, group, gpus, done_iters,
                state, key)

    def _build_runner(self, job_id, model_name, batch, group, gpus,
                      done_iters, state, key) -> JobRunner:
        try:
            return JobRunner(
                job_id, model_name, batch, self.device, self.dtype,
                group, gpus.index(self.rank), len(gpus),
                done_iters=done_iters, state=state, pool_key=key)
        except torch.cuda.OutOfMemoryError:
            # pooled runners hold HBM the scheduler's cluster model
            # does not account for — reclaim it all and retry once
            RUNNER_POOL.flush()
            torch.cuda.empty_cache()
            return JobRunner(
                job_id, model_name, batch, self.device, self.dtype,
                group, gpus.index(self.rank), len(gpus),
                done_iters=done_iters, state=state, pool_key=key)

    # ---- warm pool ----------------------------------------------

    def prewarm(self, families: List[str], count: int = 1,
                iters: int = 6) -> None:
        """Pre-build `count` single-GPU runners per model family on
        this rank, run `iters` training iterations each (eager warmup
        + hipGraph capture + first-touch MIOpen find + allocator
        growth), and park them in the process-level runner pool.  Jobs
        the scheduler later places on this rank adopt a warm runner
        instead of paying cold construction inside the timed path."""
        import itertools
        for i, fam in itertools.product(range(count), families):
            batch = catalog.model_meta(fam).default_batch
            key = (fam, batch, (self.rank,))
            r = self._build_runner(f"_warm_{fam}_{i}", fam, batch,
                                   None, [self.rank], 0, None, key)
            r.step(iters)
            if not RUNNER_POOL.put(r):
                r.release()

    def prewarm_gang(self, family: str, gpus: List[int],
                     iters: int = 6) -> None:
        """Gang-runner prewarm: every rank in `gpus` must call this at
        the same point (the warmup iterations run the gang's bucketed
        all-reduce collectively, warming the RCCL communicator)."""
        if self.rank not in gpus:
            return
        batch = catalog.model_meta(family).default_batch
        key = (family, batch, tuple(gpus))
        r = self._build_runner(f"_warm_gang_{family}", family, batch,
                               self.gangs.get(gpus), gpus, 0, None, key)
        r.step(iters)
        if not RUNNER_POOL.put(r):
            r.release()

    def retire_all(self) -> None:
        """Release every hosted runner into the pool (end of a replay
        phase; pooled runners stay warm for the next backend)."""
        for r in self.runners.values():
            if not RUNNER_POOL.put(r):
                r.release()
        self.runners.clear()

    # ---- one tick -----------------------------------------------

    def tick_once(self) -> bool:
        """Run one scheduler tick on all ranks; returns stop flag."""
        if self.rank == 0:
            s = self.sched
            now = self.now()
            plan = s.tick(now)
            iters, meta = {}, {}
            for job in s.jobs.values():
                if job.state.name == "RUNNING":
                    iters[job.job_id] = min(self.iters_per_tick,
                                            job.remaining_iters)
                    meta[job.job_id] = (job.spec.model,
                                        catalog.job_batch(job.spec),
                                        list(job.placement),
                                        job.done_iters)
            msg = TickMsg(stop=s.all_done, plan=plan, iters=iters,
                          meta=meta, cleanup=list(self._pending_cleanup))
            self._pending_cleanup.clear()
        else:
            msg = None
        msg = self._bcast(msg)

        for job_id in msg.cleanup:
            r = self.runners.pop(job_id, None)
            if r is not None and not RUNNER_POOL.put(r):
                r.release()
            if self.rank == 0:
                self.ckpt.drop(job_id)
        if msg.stop:
            return True

        self._apply_phase_a(msg.plan, msg.meta)
        self._barrier()
        self._apply_phase_b(msg.plan, msg.meta)

        # run this tick's quotas for locally hosted running jobs.
        # Launch every job's iterations first (per-job HIP streams:
        # co-packed jobs overlap), then sync each.
        report = {}
        launched = []
        for job_id, n in msg.iters.items():
            r = self.runners.get(job_id)
            if r is None or self.rank not in msg.meta[job_id][2]:
                continue
            try:
                r.step_async(n)
                launched.append((job_id, r, n))
            except Exception as e:  # job-level fault isolation
                r.poolable = False  # possibly corrupt: never reuse
                report[job_id] = ("fail", 0, 0.0, f"{type(e).__name__}: {e}")
        for job_id, r, n in launched:
            try:
                done, secs = r.step_finish()
                if r.gang_rank == 0:
                    mem = 0
                    if r.measured_mem and not r._mem_reported:
                        mem = r.measured_mem
                        r._mem_reported = True
                    report[job_id] = ("ok", done, secs, r.last_loss, mem)
            except Exception as e:
                r.poolable = False  # possibly corrupt: never reuse
                report[job_id] = ("fail", 0, 0.0, f"{type(e).__name__}: {e}")

        reports = self._gather(report)
        if self.rank == 0:
            now = self.now()
            for rep in reports:
                for job_id, r in rep.items():
                    if r[0] == "fail":
                        self.sched.on_fail(job_id, r[3], now)
                        self._pending_cleanup.append(job_id)
                    else:
                        _, done, secs, _, mem = r
                        if mem:
                            self.sched.cluster.update_mem(job_id, mem)
                        width = len(self.sched.jobs[job_id].placement)
                        fin = self.sched.on_progress(
                            job_id, done, secs * width, now,
                            iter_time=secs / max(done, 1))
                        if fin:
                            self._pending_cleanup.append(job_id)
        self.ticks += 1
        return False

    # ---- main loop ----------------------------------------------

    def run(self, max_ticks: Optional[int] = None,
            max_seconds: Optional[float] = None):
        while True:
            if self.tick_once():
                break
            if max_ticks is not None and self.ticks >= max_ticks:
                break
            if max_seconds is not None and self.now() > max_seconds:
                break
        self._barrier()
        if self.rank == 0:
            self.ckpt.cleanup()
        return self.sched.log if self.rank == 0 else None


def init_distributed(backend: str = "gloo") -> tuple:
    """torchrun entry: returns (rank, world).  Single-process when no
    torchrun env is present."""
    if "RANK" not in os.environ:
        return 0, 1
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world)
    return rank, world
