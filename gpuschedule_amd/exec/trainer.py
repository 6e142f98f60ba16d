"""Per-rank job runner: owns a job's replica and executes iterations.

One JobRunner per (job, rank).  The model lives on this rank's GPU (or
CPU in tests); gang members share a pre-created RCCL group and reduce
grads through parallel/ddp.GradReducer.  Weights are seeded by MODEL
FAMILY so every gang member constructs an identical replica without
any broadcast AND a pooled runner re-seeded in place is bit-identical
to a fresh build; data is seeded by (job, rank, iteration) so DP
shards differ and a migrated/restored job replays a deterministic
stream.

Warm runner pool: building a runner on-GPU costs model construction +
eager warmup + hipGraph capture + first-touch MIOpen find — hundreds
of ms that would otherwise land inside the scheduler's timed path for
every arriving job.  RUNNER_POOL keeps released runners alive, keyed
by (model, batch, gang); :meth:`JobRunner.adopt` rebinds one to a new
job by copying the family's pristine-init template back into the SAME
parameter/optimizer-state tensors, so the captured hipGraph (whose
baked device pointers reference those tensors) replays unchanged.
"""
from __future__ import annotations

import os
import time
import zlib
from typing import Optional

import torch

from ..models import build_model, compute_loss
from ..ops.optim import FusedAdamW, FusedSGD
from ..parallel.ddp import GradReducer


# MIOpen solver selection for the library conv path (ResNet): let
# find-mode benchmark solvers per problem instead of immediate-mode
# heuristics (A/B'd: gate to disable).
if os.environ.get("GSAMD_MIOPEN_BENCHMARK", "1") != "0":
    torch.backends.cudnn.benchmark = True


def _job_seed(job_id: str) -> int:
    return zlib.crc32(job_id.encode()) & 0x7FFFFFFF


# Pristine-init templates per (family, device, dtype): a clone of the
# state_dict right after seeded construction.  Because construction is
# seeded by FAMILY, the template equals what any fresh build produces,
# so JobRunner.adopt(template copy) == fresh JobRunner, bit for bit.
# Families above the size gate (Llama-8B) skip caching (and pooling).
_TEMPLATE_MAX_BYTES = 4 << 30
_INIT_TEMPLATES: dict = {}


def _template_key(model_name: str, device, dtype) -> tuple:
    return (model_name, str(device), str(dtype))


class RunnerPool:
    """Process-level cache of released JobRunners, keyed by
    (model, batch, gang-gpus).  Budgeted by the runners' measured
    per-job memory (the scheduler's cluster model does not account for
    pooled memory, so the budget keeps hoarding bounded; the backend
    flushes the pool and retries on OOM)."""

    def __init__(self, max_bytes: Optional[int] = None):
        if max_bytes is None:
            max_bytes = int(os.environ.get("GSAMD_POOL_BYTES",
                                           32 << 30))
        self.max_bytes = max_bytes
        self._pool: dict = {}      # key -> list[JobRunner]
        self._order: list = []     # LRU of (key, runner)
        self._bytes = 0

    def _cost(self, r) -> int:
        return max(r.measured_mem, 1 << 20)

    def put(self, runner) -> bool:
        if runner is None or runner.pool_key is None \
                or runner.model is None or not runner.poolable:
            return False
        cost = self._cost(runner)
        if cost > self.max_bytes:
            return False
        while self._bytes + cost > self.max_bytes and self._order:
            k, old = self._order.pop(0)
            self._pool[k].remove(old)
            self._bytes -= self._cost(old)
            old.release()
        self._pool.setdefault(runner.pool_key, []).append(runner)
        self._order.append((runner.pool_key, runner))
        self._bytes += cost
        return True

    def get(self, key):
        lst = self._pool.get(key)
        if not lst:
            return None
        r = lst.pop()
        self._order.remove((key, r))
        self._bytes -= self._cost(r)
        return r

    def flush(self) -> None:
        for k, r in self._order:
            r.release()
        self._pool.clear()
        self._order.clear()
        self._bytes = 0


RUNNER_POOL = RunnerPool()


class JobRunner:
    def __init__(self, job_id: str, model_name: str, batch_size: int,
                 device, dtype, gang_group, gang_rank: int,
                 gang_world: int, done_iters: int = 0,
                 state: Optional[dict] = None,
                 pool_key: Optional[tuple] = None):
        self.job_id = job_id
        self.model_name = model_name
        self.batch_size = batch_size
        self.device = device
        self.dtype = dtype
        self.gang_group = gang_group
        self.gang_rank = gang_rank
        self.gang_world = gang_world
        self.done_iters = done_iters
        self.pool_key = pool_key

        # Seeded by FAMILY (not job) so a pooled runner re-seeded from
        # the family template is identical to a fresh build on every
        # gang member — see module docstring.
        self.model, self.make_batch, default_bs = build_model(
            model_name, device=device, dtype=dtype,
            seed=_job_seed(model_name))
        if device == "cpu" or str(device) == "cpu":
            self.model = self.model.float()
        self.batch_size = batch_size or default_bs
        eff_dtype = (torch.float32 if str(device) == "cpu" else dtype)
        tkey = _template_key(model_name, device, eff_dtype)
        if tkey not in _INIT_TEMPLATES:
            sd = self.model.state_dict()
            nbytes = sum(v.numel() * v.element_size()
                         for v in sd.values())
            _INIT_TEMPLATES[tkey] = (
                {k: v.detach().clone() for k, v in sd.items()}
                if nbytes <= _TEMPLATE_MAX_BYTES else None)
        self.poolable = _INIT_TEMPLATES[tkey] is not None
        self._tkey = tkey
        if model_name.startswith("resnet"):
            self.optim = FusedSGD(self.model.parameters(), lr=0.05,
                                  momentum=0.9, weight_decay=5e-4)
        else:
            self.optim = FusedAdamW(self.model.parameters(), lr=1e-4,
                                    weight_decay=0.01)
        if state is not None:
            self.load_state(state)
        self.reducer = GradReducer(self.model, gang_group)
        self.last_loss = float("nan")
        # hipGraph capture of the whole training step (fwd+bwd+fused
        # optimizer): per-job static shapes make the step replayable;
        # launch-bound models (small CNNs) gain the most.  Gang steps
        # (RCCL collectives are hipGraph-capturable on ROCm) are gated
        # behind GSAMD_GANG_GRAPHS=1: the path cannot be validated on a
        # 1-GPU lease — RCCL refuses same-device 2-rank communicators
        # ("Duplicate GPU detected", scripts/rccl_1gpu_probe.py) — so
        # it stays off by default until a multi-GPU run proves it.
        self.use_graph = (str(device) != "cpu"
                          and not os.environ.get("GSAMD_NO_GRAPHS")
                          and (gang_world == 1
                               or bool(os.environ.get("GSAMD_GANG_GRAPHS"))))
        self._graph = None
        self._graph_warmup = 3  # eager iters before capture (lazy opt
        #                         state init + allocator warmup)
        self._static = None     # (x, y, loss)
        # Per-job HIP stream: all of this job's work (including eager
        # warmup) runs on it, so autograd's AccumulateGrad nodes bind
        # to the SAME stream the hipGraph captures on (the default-
        # stream mismatch breaks capture), and concurrently-resident
        # jobs' kernels can overlap.
        self.stream = (torch.cuda.Stream(device)
                       if str(device) != "cpu" else None)
        self._gen = None  # device RNG for synthetic batches
        self._loss_ref = None
        self._pending = 0
        self._t0 = 0.0
        # measured per-GPU footprint (reported once to the scheduler
        # for bin-packing refinement)
        self._mem_base = (torch.cuda.memory_allocated(device)
                          if self.stream is not None else 0)
        self.measured_mem = 0
        self._mem_reported = False

    # ---- state movement -----------------------------------------

    def state_dict(self) -> dict:
        return {"model": self.model.state_dict(),
                "optim": self.optim.state_dict(),
                "done_iters": self.done_iters}

    def load_state(self, state: dict) -> None:
        self.model.load_state_dict(state["model"])
        self.optim.load_state_dict(state["optim"])
        self.done_iters = state.get("done_iters", self.done_iters)
        # Restore the saver's global RNG streams so stochastic ops
        # (dropout) resume where the checkpoint left off; synthetic
        # data streams are independently (job, rank, iter)-seeded.
        rng = state.get("torch_rng")
        if rng is not None:
            torch.set_rng_state(rng)
        crng = state.get("cuda_rng")
        if crng is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(crng, self.device)

    def regang(self, group, gang_rank: int, gang_world: int) -> None:
        """Rebind a SURVIVING runner to its job's resized gang: swap
        the collective group + reducer in place (model/optimizer state
        tensors stay — that is the point: no checkpoint round-trip).
        The hipGraph captured for the old gang width is dropped; width
        1 recaptures after one eager iteration (allocator/MIOpen are
        already warm)."""
        self.reducer.detach()
        self.gang_group = group
        self.gang_rank = gang_rank
        self.gang_world = gang_world
        self.reducer = GradReducer(self.model, group)
        self._graph = None
        self._static = None
        self.use_graph = (str(self.device) != "cpu"
                          and not os.environ.get("GSAMD_NO_GRAPHS")
                          and (gang_world == 1
                               or bool(os.environ.get("GSAMD_GANG_GRAPHS"))))
        self._graph_warmup = 1

    @torch.no_grad()
    def adopt(self, job_id: str, done_iters: int = 0) -> None:
        """Rebind this (pooled) runner to a new job: copy the family's
        pristine-init template back into the SAME parameter tensors
        and reset the optimizer state in place, so the captured
        hipGraph and the multi-tensor optimizer pointer tables stay
        valid.  The result is bit-identical to a fresh build (family-
        seeded init); data streams re-seed from the new job_id."""
        tpl = _INIT_TEMPLATES[self._tkey]
        sd = self.model.state_dict()
        for k, v in tpl.items():
            sd[k].copy_(v)
        for p in self.model.parameters():
            if p.grad is not None:
                p.grad.zero_()
        self.optim.reset_state()
        self.job_id = job_id
        self.done_iters = done_iters
        self.last_loss = float("nan")
        self._loss_ref = None
        self._pending = 0
        self._mem_reported = False

    # ---- execution ----------------------------------------------

    def _batch(self, it: int):
        seed = (_job_seed(self.job_id) * 1000003 + it * 8191 +
                self.gang_rank) & 0x7FFFFFFF
        dev = torch.device(self.device)
        nb = str(dev) != "cpu"
        if nb:
            # generate synthetic data ON the GPU: a ResNet-50 batch is
            # ~38 MB — host-side randn + H2D would dominate the tick
            if self._gen is None:
                self._gen = torch.Generator(device=dev)
            self._gen.manual_seed(seed)
            x, y = self.make_batch(self.batch_size, dev, self._gen)
            if x.is_floating_point():
                x = x.to(self.dtype)
            if x.dim() == 4:
                x = x.contiguous(memory_format=torch.channels_last)
            return x, y
        gen = torch.Generator(device="cpu").manual_seed(seed)
        return self.make_batch(self.batch_size, "cpu", gen)

    def _eager_iter(self, x, y):
        if not getattr(self.optim, "fused_zero_grad", False):
            self.optim.zero_grad(set_to_none=True)
        loss = compute_loss(self.model_name, self.model, x, y)
        loss.backward()
        self.reducer.finalize()
        self.optim.step()
        return loss

    def _try_capture(self, x, y) -> bool:
        try:
            sx = x.clone()
            sy = y.clone()
            self.stream.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, stream=self.stream):
                sloss = self._eager_iter(sx, sy)
            self._graph = g
            self._static = (sx, sy, sloss)
            return True
        except Exception:
            # capture-unsafe model (CPU syncs, dynamic shapes): stay
            # eager for this job
            self._graph = None
            self.use_graph = False
            return False

    def _one_iter(self, x, y):
        if self._graph is None and self.use_graph \
                and self._graph_warmup <= 0:
            self._try_capture(x, y)
        if self._graph is not None:
            sx, sy, sloss = self._static
            sx.copy_(x, non_blocking=True)
            sy.copy_(y, non_blocking=True)
            self._graph.replay()
            return sloss
        self._graph_warmup -= 1
        return self._eager_iter(x, y)

    def step_async(self, n_iters: int = 1) -> None:
        """Launch n_iters iterations on this job's stream WITHOUT the
        final sync (lets co-packed jobs' kernels overlap);
        :meth:`step_finish` completes the accounting."""
        self._t0 = time.perf_counter()
        self._pending = n_iters
        loss = None
        if self.stream is None:
            for _ in range(n_iters):
                x, y = self._batch(self.done_iters)
                loss = self._one_iter(x, y)
                self.done_iters += 1
        else:
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                for _ in range(n_iters):
                    x, y = self._batch(self.done_iters)
                    loss = None  # drop autograd refs before capture
                    loss = self._one_iter(x, y)
                    self.done_iters += 1
        self._loss_ref = loss

    def step_finish(self) -> tuple:
        if self.stream is not None:
            self.stream.synchronize()
            if self.measured_mem == 0 and self.done_iters >= 4:
                used = (torch.cuda.memory_allocated(self.device)
                        - self._mem_base)
                self.measured_mem = max(used, 1)
        if self._loss_ref is not None:
            self.last_loss = float(self._loss_ref.detach())
            self._loss_ref = None
        n = self._pending
        self._pending = 0
        return n, time.perf_counter() - self._t0

    def step(self, n_iters: int = 1) -> tuple:
        """Run up to n_iters training iterations; returns
        (iters_done, wall_seconds)."""
        self.step_async(n_iters)
        return self.step_finish()

    def release(self) -> None:
        """Drop GPU state (eviction or completion)."""
        self.reducer.detach()
        self._graph = None
        self._static = None
        self.model = None
        self.optim = None
