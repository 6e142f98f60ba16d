"""Fused RoPE and SwiGLU-gate wrappers (ops/hip/elementwise.hip)."""
from __future__ import annotations

import torch

from ._ext import get_ext


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cosb, sinb):
        ext = get_ext(required=True)
        ctx.save_for_backward(cosb, sinb)
        return ext.rope_apply(x, cosb, sinb, False)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        cosb, sinb = ctx.saved_tensors
        return ext.rope_apply(dy.contiguous(), cosb, sinb, True), \
            None, None


def fused_rope(x: torch.Tensor, cosb: torch.Tensor,
               sinb: torch.Tensor) -> torch.Tensor:
    """Rotate-half RoPE on [B, H, S, D] bf16 with fp32 [S, D/2] tables."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _RopeFn.apply(x, cosb, sinb)
    S = x.shape[-2]
    c = cosb[:S].to(x.dtype)[None, None]
    s = sinb[:S].to(x.dtype)[None, None]
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)


class _SiluMulFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, u):
        ext = get_ext(required=True)
        ctx.save_for_backward(g, u)
        return ext.silu_mul_fwd(g, u)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        g, u = ctx.saved_tensors
        dg, du = ext.silu_mul_bwd(dy.contiguous(), g, u)
        return dg, du


def silu_mul(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    if g.is_cuda and g.dtype == torch.bfloat16:
        return _SiluMulFn.apply(g.contiguous(), u.contiguous())
    return torch.nn.functional.silu(g) * u


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = get_ext(required=True)
        ctx.save_for_backward(x)
        return ext.gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        (x,) = ctx.saved_tensors
        return ext.gelu_bwd(dy.contiguous(), x)


def fused_gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approximate GELU: one fused bf16 pass each way on GPU
    (torch's Gelu + GeluBackward elementwise pair otherwise); fp32
    torch reference on CPU."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _GeluFn.apply(x.contiguous())
    return torch.nn.functional.gelu(x, approximate="tanh")


class _SwigluPackedFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        ext = get_ext(required=True)
        ctx.save_for_backward(gu)
        return ext.swiglu_fwd(gu)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        (gu,) = ctx.saved_tensors
        return ext.swiglu_bwd(dy.contiguous(), gu)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    """SwiGLU on the PACKED [.., 2F] gate_up projection: the kernels
    read both halves through the row stride and the backward writes one
    packed grad buffer — no contiguous() copies of the halves and no
    chunk-backward cat.  CPU: fp32 torch reference."""
    if gu.is_cuda and gu.dtype == torch.bfloat16:
        return _SwigluPackedFn.apply(gu.contiguous())
    g, u = gu.chunk(2, dim=-1)
    return torch.nn.functional.silu(g) * u
