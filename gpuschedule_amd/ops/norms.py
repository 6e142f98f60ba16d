"""Fused LayerNorm / RMSNorm modules backed by the gfx950 kernels.

GPU path: hand-written HIP kernels (ops/hip/norms.hip) — bf16 in/out,
fp32 statistics, one fused pass each direction.  CPU path (tests): the
same math via plain fp32 torch ops under autograd.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import get_ext


class _NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps, rms):
        ext = get_ext(required=True)
        x = x.contiguous()
        y, mean, rstd = ext.norm_fwd(x, weight, bias, eps, rms)
        ctx.save_for_backward(x, weight, mean, rstd)
        ctx.rms = rms
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.norm_bwd(dy.contiguous(), x,
                                  None if ctx.rms else mean, rstd,
                                  weight, ctx.rms)
        return dx, dw, (None if ctx.rms else db), None, None


def _cpu_layernorm(x, w, b, eps):
    xf = x.float()
    y = torch.nn.functional.layer_norm(xf, (x.shape[-1],), w.float(),
                                       b.float(), eps)
    return y.to(x.dtype)


def _cpu_rmsnorm(x, w, eps):
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * w.float()).to(x.dtype)


class FusedLayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.dim, self.eps = dim, eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        if x.is_cuda:
            return _NormFn.apply(x, self.weight, self.bias, self.eps, False)
        return _cpu_layernorm(x, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.dim}, eps={self.eps}"


class FusedRMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.dim, self.eps = dim, eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x):
        if x.is_cuda:
            return _NormFn.apply(x, self.weight, None, self.eps, True)
        return _cpu_rmsnorm(x, self.weight, self.eps)

    def extra_repr(self):
        return f"{self.dim}, eps={self.eps}"
