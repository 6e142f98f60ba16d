"""Pre-created RCCL process groups for every contiguous rank range.

Placement is contiguous (sched/cluster.py), so every gang a job can
ever get is one of the O(N^2) contiguous ranges.  Creating ALL of them
up-front (collectively, same order on every rank) removes dynamic
``new_group`` creation from the scheduling path — group creation is a
collective and racing it against per-tick placement deadlocks.

RCCL communicators are lazily initialized by the first collective on a
group, so unused groups cost nothing.
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch.distributed as dist


class GangGroups:
    def __init__(self, world_size: int, backend: Optional[str] = None):
        self.world_size = world_size
        self._groups: Dict[Tuple[int, int], object] = {}
        if dist.is_initialized():
            for n in range(2, world_size + 1):
                for s in range(0, world_size - n + 1):
                    ranks = list(range(s, s + n))
                    self._groups[(s, n)] = dist.new_group(
                        ranks, backend=backend)

    def get(self, gpus) -> Optional[object]:
        """Group for the contiguous gpu/rank list; None for width 1."""
        gpus = sorted(gpus)
        n = len(gpus)
        if n <= 1:
            return None
        s = gpus[0]
        assert gpus == list(range(s, s + n)), \
            f"gang must be contiguous, got {gpus}"
        return self._groups[(s, n)]
