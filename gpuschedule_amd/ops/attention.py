"""Attention dispatch: hand-written flash-forward kernel on GPU.

``flash_attention(q, k, v, causal)`` takes [B, H, S, D] bf16 tensors.
GPU forward runs the gfx950 flash kernel (ops/hip/attention.hip,
online-softmax MFMA); backward recomputes through the exact composite
(matmul-softmax-matmul) on the MFMA GEMM path via torch matmuls.
CPU path: fp32 reference composite (tests).
"""
from __future__ import annotations

import math

import torch

from ._ext import get_ext


def _ref_attention(q, k, v, causal: bool):
    # fp32 reference composite; used on CPU and as the numerics oracle.
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, v.float()).to(q.dtype)


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        ext = get_ext(required=True)
        o, lse = ext.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        # hand-written FA2-style backward (ops/hip/attention.hip):
        # no S x S activation ever materialized.
        ext = get_ext(required=True)
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = ext.attn_bwd(q, k, v, o, do.contiguous(), lse,
                                  ctx.causal)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = False) -> torch.Tensor:
    """q,k,v: [B, H, S, D] bf16 (k/v may have fewer heads: GQA)."""
    if q.is_cuda:
        ext = get_ext(required=True)
        if hasattr(ext, "attn_fwd"):
            if k.shape[1] != q.shape[1]:  # GQA: expand KV heads
                rep = q.shape[1] // k.shape[1]
                k = k.repeat_interleave(rep, dim=1)
                v = v.repeat_interleave(rep, dim=1)
            return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), causal)
        raise RuntimeError("attn_fwd missing from HIP extension")
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return _ref_attention(q, k, v, causal)
