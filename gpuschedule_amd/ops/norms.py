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
    """Fused norm with optional residual-add: y = norm(x + res).
    Returns (y, summed): ``summed`` is x+res (the residual stream) when
    a residual is given, else x."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps, rms):
        ext = get_ext(required=True)
        ctx.set_materialize_grads(False)
        x = x.contiguous()
        y, summed, mean, rstd = ext.norm_fwd(x, res, weight, bias, eps,
                                             rms)
        has_res = res is not None
        ctx.save_for_backward(summed, weight, mean, rstd)
        ctx.rms = rms
        ctx.has_res = has_res
        # second output: the residual stream (x+res); dummy when no res
        return y, (summed if has_res else x.new_empty(0))

    @staticmethod
    def backward(ctx, dy, dsum):
        ext = get_ext(required=True)
        summed, weight, mean, rstd = ctx.saved_tensors
        if dy is None:
            dy = torch.zeros_like(summed)
        # downstream residual-stream grad (dsum) is added into dx
        # INSIDE the bwd kernel — no separate elementwise add pass
        dx, dw, db = ext.norm_bwd(dy.contiguous(), summed,
                                  None if ctx.rms else mean, rstd,
                                  weight, ctx.rms,
                                  dsum.contiguous()
                                  if dsum is not None else None)
        dres = dx if ctx.has_res else None
        return dx, dres, dw, (None if ctx.rms else db), None, None


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

    def forward(self, x, residual=None, return_sum=False):
        if x.is_cuda:
            y, summed = _NormFn.apply(
                x, residual.contiguous() if residual is not None
                else None, self.weight, self.bias, self.eps, False)
            return (y, summed) if return_sum else y
        y = _cpu_layernorm(x + residual if residual is not None else x,
                           self.weight, self.bias, self.eps)
        if return_sum:
            return y, (x + residual if residual is not None else x)
        return y

    def extra_repr(self):
        return f"{self.dim}, eps={self.eps}"


class FusedRMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.dim, self.eps = dim, eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x, residual=None, return_sum=False):
        if x.is_cuda:
            y, summed = _NormFn.apply(
                x, residual.contiguous() if residual is not None
                else None, self.weight, None, self.eps, True)
            return (y, summed) if return_sum else y
        xs = x + residual if residual is not None else x
        y = _cpu_rmsnorm(xs, self.weight, self.eps)
        return (y, xs) if return_sum else y

    def extra_repr(self):
        return f"{self.dim}, eps={self.eps}"
