"""Fused embedding lookup — ops/hip/embedding.hip.

Replaces ``nn.Embedding`` in the transformer work units.  Besides the
one-pass vectorized gather, the point is the BACKWARD: torch's dense
embedding backward runs rocprim radix-sort/partition kernels whose
hipGraph replay proved fragile on ROCm 7 when many job graphs coexist
(HSA aperture faults; see scripts/repro_prewarm*.py).  The HIP path is
two plain kernels (fp32 atomic scatter + cast) that capture and replay
trivially.  CPU path: fp32 torch reference (tests)."""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from ._ext import get_ext


class _EmbedFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight, idx):
        ext = get_ext(required=True)
        ctx.save_for_backward(idx)
        ctx.V = weight.shape[0]
        return ext.embed_fwd(weight, idx)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        (idx,) = ctx.saved_tensors
        dw = ext.embed_bwd(dy, idx, ctx.V)
        return dw, None


class FusedEmbedding(nn.Module):
    def __init__(self, num_embeddings: int, embedding_dim: int):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.weight = nn.Parameter(
            torch.empty(num_embeddings, embedding_dim))
        nn.init.normal_(self.weight)

    def forward(self, idx: torch.Tensor) -> torch.Tensor:
        if self.weight.is_cuda and self.weight.dtype == torch.bfloat16:
            return _EmbedFn.apply(self.weight, idx)
        return F.embedding(idx, self.weight)

    def extra_repr(self) -> str:
        return f"{self.num_embeddings}, {self.embedding_dim}"
