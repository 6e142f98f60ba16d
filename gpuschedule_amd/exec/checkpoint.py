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
"torch_rng"}`` — all tensors on CPU.
"""
from __future__ import annotations

import os
from typing import Dict, Optional

import torch


class PinnedPool:
    """Reusable pinned staging buffers, keyed by size class."""

    def __init__(self):
        self._bufs: Dict[int, torch.Tensor] = {}

    def get(self, nbytes: int) -> torch.Tensor:
        # round to 1 MiB size classes
        size = max(1 << 20, 1 << (nbytes - 1).bit_length())
        buf = self._bufs.get(size)
        if buf is None:
            buf = torch.empty(size, dtype=torch.uint8, pin_memory=True)
            self._bufs[size] = buf
        return buf


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

    def _stage_to_cpu(self, t: torch.Tensor) -> torch.Tensor:
        """D2H via pinned buffer on the side stream."""
        if not t.is_cuda:
            return t.detach().clone()
        if self._side is None:
            self._side = torch.cuda.Stream()
        flat = t.detach().reshape(-1)
        nbytes = flat.numel() * flat.element_size()
        pinned = self._pool.get(nbytes)
        view = pinned[:nbytes].view(t.dtype)[:flat.numel()]
        with torch.cuda.stream(self._side):
            self._side.wait_stream(torch.cuda.current_stream())
            view.copy_(flat, non_blocking=True)
        self._side.synchronize()
        return view.clone().reshape(t.shape)

    def _tree_to_cpu(self, obj):
        if isinstance(obj, torch.Tensor):
            return self._stage_to_cpu(obj)
        if isinstance(obj, dict):
            return {k: self._tree_to_cpu(v) for k, v in obj.items()}
        if isinstance(obj, (list, tuple)):
            t = [self._tree_to_cpu(v) for v in obj]
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
        }
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
