"""Attention dispatch: hand-written flash kernels on GPU.

``flash_attention(q, k, v, causal)`` takes [B, H, S, D] bf16 tensors
(k/v may have fewer heads: GQA).  ``flash_attention_qkv(qkv, causal)``
takes the packed [B, S, 3, H, D] view of a fused self-attention
projection and runs the zero-copy strided path.  GPU forward runs the
gfx950 flash kernel (ops/hip/attention.hip, online-softmax MFMA);
backward runs the FA2-style two-kernel HIP backward (dK/dV then dQ —
no S x S activation is ever materialized).  CPU path: fp32 reference
composite (tests).
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


class _FlashAttnQKVFn(torch.autograd.Function):
    """Packed-QKV self-attention: takes the [B,S,3,H,D] view of one
    fused projection.  The kernels read the q/k/v slices through their
    strides and write the three grads into ONE [B,S,3,H,D] buffer, so
    the whole attention block adds ZERO layout copies (no per-slice
    contiguous(), no select-backward zero+copy, no output re-pack)."""

    @staticmethod
    def forward(ctx, qkv, causal):
        ext = get_ext(required=True)
        q = qkv[:, :, 0].transpose(1, 2)  # [B,H,S,D] strided views
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        o, lse = ext.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(qkv, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        ext = get_ext(required=True)
        qkv, o, lse = ctx.saved_tensors
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        if do.stride() != o.stride():  # usually a free view already
            do = do.permute(0, 2, 1, 3).contiguous().permute(0, 2, 1, 3)
        out = ext.attn_bwd(q, k, v, o, do, lse, ctx.causal)
        return out[3], None  # the packed [B,S,3,H,D] grad


def flash_attention_qkv(qkv: torch.Tensor,
                        causal: bool = False) -> torch.Tensor:
    """qkv: [B, S, 3, H, D] (contiguous, e.g. a fused-projection view).
    Returns O as logical [B, H, S, D]; ``o.transpose(1, 2)`` is a free
    view for the head-merge reshape."""
    if qkv.is_cuda:
        return _FlashAttnQKVFn.apply(qkv.contiguous(), causal)
    q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
    return _ref_attention(q, k, v, causal)


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = False) -> torch.Tensor:
    """q,k,v: [B, H, S, D] bf16 (k/v may have fewer heads: GQA)."""
    if q.is_cuda:
        ext = get_ext(required=True)
        if hasattr(ext, "attn_fwd"):
            # GQA is kernel-native: k/v keep their (fewer) heads; the
            # kernels read kv head h/(H/HK) directly and the backward
            # group-accumulates dK/dV over each head's query group.
            return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), causal)
        raise RuntimeError("attn_fwd missing from HIP extension")
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return _ref_attention(q, k, v, causal)
