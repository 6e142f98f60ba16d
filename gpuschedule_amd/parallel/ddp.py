"""Bucketed data-parallel gradient all-reduce, overlapped with backward.

Own implementation (no torch.nn.parallel.DistributedDataParallel): the
scheduler creates/destroys gangs constantly, and DDP's reducer keeps
process-group state that doesn't survive that churn cheaply.  This
reducer is a light object over an existing process group:

* parameters are grouped into ~``bucket_mb`` buckets in reverse
  registration order (the order backward produces grads),
* each parameter's ``post_accumulate_grad_hook`` marks it ready; when a
  bucket completes, its grads are flattened, pre-scaled by 1/world and
  an async ``all_reduce`` is launched on the gang's RCCL group — so
  communication for early buckets overlaps the rest of backward
  (ring all-reduce over xGMI, per-link bound: ~25 MB buckets keep the
  pipeline full without serializing on launch latency),
* ``finalize()`` waits on the in-flight works and scatters the reduced
  flats back into ``param.grad``.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class GradReducer:
    def __init__(self, model: torch.nn.Module,
                 group: Optional[dist.ProcessGroup],
                 bucket_mb: int = 25):
        self.group = group
        self.world = dist.get_world_size(group) if group is not None else 1
        params = [p for p in model.parameters() if p.requires_grad]
        self.buckets: List[List[torch.nn.Parameter]] = []
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(params):
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= bucket_mb << 20:
                self.buckets.append(cur)
                cur, size = [], 0
        if cur:
            self.buckets.append(cur)
        self._pidx = {}
        for bi, bkt in enumerate(self.buckets):
            for p in bkt:
                self._pidx[p] = bi
        self._ready = [0] * len(self.buckets)
        self._inflight = []  # (work, flat, bucket_index)
        self._hooks = []
        if self.world > 1:
            for p in params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad))

    def _launch(self, bi: int) -> None:
        bkt = self.buckets[bi]
        flat = torch.cat([
            (p.grad if p.grad is not None
             else torch.zeros_like(p)).reshape(-1) for p in bkt])
        flat.div_(self.world)
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM,
                               group=self.group, async_op=True)
        self._inflight.append((work, flat, bi))

    def _on_grad(self, p: torch.nn.Parameter) -> None:
        bi = self._pidx[p]
        self._ready[bi] += 1
        if self._ready[bi] == len(self.buckets[bi]):
            self._launch(bi)

    def finalize(self) -> None:
        """Call after backward(), before optimizer.step()."""
        if self.world <= 1:
            return
        # launch any bucket that never completed (unused params)
        launched = {bi for _, _, bi in self._inflight}
        for bi in range(len(self.buckets)):
            if bi not in launched and self._ready[bi] > 0:
                self._launch(bi)
        for work, flat, bi in self._inflight:
            work.wait()
            off = 0
            for p in self.buckets[bi]:
                n = p.numel()
                if p.grad is not None:
                    p.grad.copy_(flat[off:off + n].view_as(p))
                off += n
        self._inflight.clear()
        self._ready = [0] * len(self.buckets)

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()


def broadcast_module(model: torch.nn.Module,
                     group: Optional[dist.ProcessGroup], src_rank: int):
    """Broadcast params+buffers from global rank ``src_rank`` (joiners
    in an elastic resize sync from an existing replica).  Iterates the
    state_dict, NOT parameters()+buffers(): non-persistent buffers
    (e.g. lazily-built RoPE tables) may exist on trained ranks but not
    on joiners, which would desynchronize the collective sequence."""
    if group is None or dist.get_world_size(group) <= 1:
        return
    with torch.no_grad():
        for t in model.state_dict().values():
            if isinstance(t, torch.Tensor) and t.numel():
                dist.broadcast(t.data, src=src_rank, group=group)


def broadcast_state(model: torch.nn.Module, optim,
                    group: Optional[dist.ProcessGroup], src_rank: int):
    """Move a job's full training state to elastic joiners over the
    gang's collective backend (RCCL over xGMI on GPU): params+buffers,
    optimizer state (fp32 masters/moments), and the step counter.  All
    ranks of the new gang must call this collectively; joiners first
    materialize empty state tensors as broadcast destinations
    (optim.init_state), so the tensor sequence is identical on every
    rank.  Replaces the /dev/shm checkpoint round-trip for pure-grow
    (and overlap) resizes."""
    if group is None or dist.get_world_size(group) <= 1:
        return
    broadcast_module(model, group, src_rank)
    if optim is None:
        return
    with torch.no_grad():
        optim.init_state()
        for g in optim.param_groups:
            for p in g["params"]:
                st = optim.state.get(p)
                if not st:
                    continue
                for k in sorted(st.keys()):
                    v = st[k]
                    if isinstance(v, torch.Tensor):
                        dist.broadcast(v.data, src=src_rank, group=group)
        if hasattr(optim, "true_step"):
            dev = next(model.parameters()).device
            step = torch.tensor([optim.true_step()], dtype=torch.long,
                                device=dev)
            dist.broadcast(step, src=src_rank, group=group)
            optim.set_step(int(step.item()))
