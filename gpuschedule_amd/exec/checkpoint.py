"""Checkpoint / suspend / migration state movement.

MI355X-first design:

* **Suspend (Gandiva fast path)** keeps the job's tensors RESIDENT in
  the GPU's 288 GB HBM — no copy at all; resume is free.  The
  scheduler's cluster model decides when residency must end.
* **Eviction / migration** stages every GPU tensor into a reusable
  PINNED host buffer with ``copy_(non_blocking=True)`` issued on a
  dedicated side HIP stream (= ``hipMemcpyAsync`` on a non-default
  stream, overlapping any compute still queued on the job stream),
  then serializes to a node-local file (/dev/shm by default) that the
  destination rank maps back in.  The pinned pool is reused across
  checkpoints to avoid re-registering host memory.

Checkpoint file format (documented, stable): ``torch.save`` of
``{"job_id", "done_iters", "model": state_dict, "optim": state_dict,
"torch_rng", "cuda_rng"}`` — all tensors on CPU.  The RNG states are
the saving (gang-leader) rank's global torch CPU/device generator
states, restored on load so stochastic ops (dropout) resume their
stream; per-rank synthetic-data streams are independently seeded by
(job, rank, iteration) and need no saved state.
"""
from __future__ import annotations

import os
from typing import Dict, Optional

import torch


class PinnedPool:
    """Reusable pinned staging buffers, keyed by size class.

    ``acquire`` hands out a free buffer (allocating when none is
    cached) so several D2H copies can be in flight at once; the saver
    calls ``release`` after the single final sync, returning buffers
    for reuse by the next checkpoint."""

    def __init__(self):
        self._free: Dict[int, list] = {}

    def acquire(self, nbytes: int):
        # round to 1 MiB size classes
        size = max(1 << 20, 1 << max((nbytes - 1).bit_length(), 0))
        lst = self._free.get(size)
        if lst:
            return size, lst.pop()
        return size, torch.empty(size, dtype=torch.uint8, pin_memory=True)

    def release(self, size: int, buf: torch.Tensor) -> None:
        self._free.setdefault(size, []).append(buf)


class _Staged:
    """A tensor whose D2H copy is in flight on the side stream.  Holds
    a reference to the SOURCE (which may be a reshape temporary) so the
    allocator cannot recycle its memory before the copy lands — the
    per-tensor sync that used to provide that guarantee is gone."""
    __slots__ = ("view", "shape", "size", "buf", "src")

    def __init__(self, view, shape, size, buf, src):
        self.view = view
        self.shape = shape
        self.size = size
        self.buf = buf
        self.src = src


class CheckpointStore:
    def __init__(self, root: Optional[str] = None):
        if root is None:
            base = "/dev/shm" if os.path.isdir("/dev/shm") else "/tmp"
            root = os.path.join(base, f"gpuschedule_ckpt_{os.getppid()}")
        self.root = root
        os.makedirs(root, exist_ok=True)
        self._pool = PinnedPool()
        self._side: Optional[torch.cuda.Stream] = None

    def path(self, job_id: str) -> str:
        return os.path.join(self.root, f"{job_id}.pt")

    def exists(self, job_id: str) -> bool:
        return os.path.exists(self.path(job_id))

    def _stage_to_cpu(self, t: torch.Tensor):
        """Issue the D2H copy on the side stream; NO per-tensor sync.
        Returns a :class:`_Staged` placeholder that :meth:`_finalize`
        resolves after ONE sync covering every in-flight copy — the
        copies of a multi-GB state_dict pipeline back-to-back instead
        of paying launch+sync latency per tensor."""
        if not t.is_cuda:
            return t.detach().clone()
        if self._side is None:
            self._side = torch.cuda.Stream()
        flat = t.detach().reshape(-1)
        nbytes = flat.numel() * flat.element_size()
        size, buf = self._pool.acquire(nbytes)
        view = buf[:nbytes].view(t.dtype)[:flat.numel()]
        with torch.cuda.stream(self._side):
            self._side.wait_stream(torch.cuda.current_stream())
            view.copy_(flat, non_blocking=True)
        return _Staged(view, t.shape, size, buf, flat)

    def _tree_to_cpu(self, obj):
        if isinstance(obj, torch.Tensor):
            return self._stage_to_cpu(obj)
        if isinstance(obj, dict):
            return {k: self._tree_to_cpu(v) for k, v in obj.items()}
        if isinstance(obj, (list, tuple)):
            t = [self._tree_to_cpu(v) for v in obj]
            return type(obj)(t) if not isinstance(obj, tuple) else tuple(t)
        return obj

    def _finalize(self, obj):
        """Resolve _Staged placeholders after the single side-stream
        sync: clone out of the pinned buffer and return it to the
        pool."""
        if isinstance(obj, _Staged):
            out = obj.view.clone().reshape(obj.shape)
            self._pool.release(obj.size, obj.buf)
            return out
        if isinstance(obj, dict):
            return {k: self._finalize(v) for k, v in obj.items()}
        if isinstance(obj, (list, tuple)):
            t = [self._finalize(v) for v in obj]
            return type(obj)(t) if not isinstance(obj, tuple) else tuple(t)
        return obj

    def save(self, job_id: str, model: torch.nn.Module,
             optim: Optional[torch.optim.Optimizer],
             done_iters: int) -> str:
        state = {
            "job_id": job_id,
            "done_iters": done_iters,
            "model": self._tree_to_cpu(model.state_dict()),
            "optim": self._tree_to_cpu(optim.state_dict())
            if optim is not None else None,
            "torch_rng": torch.get_rng_state(),
            "cuda_rng": (torch.cuda.get_rng_state()
                         if torch.cuda.is_available() else None),
        }
        if self._side is not None:
            self._side.synchronize()  # one sync for ALL staged copies
        state = self._finalize(state)
        p = self.path(job_id)
        tmp = p + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, p)  # atomic: readers never see partial files
        return p

    def load(self, job_id: str) -> dict:
        return torch.load(self.path(job_id), map_location="cpu",
                          weights_only=False)

    def drop(self, job_id: str) -> None:
        try:
            os.remove(self.path(job_id))
        except FileNotFoundError:
            pass

    def cleanup(self) -> None:
        for f in os.listdir(self.root):
            try:
                os.remove(os.path.join(self.root, f))
            except OSError:
                pass
