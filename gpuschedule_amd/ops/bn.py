"""Fused BatchNorm2d (+ReLU) module on the gfx950 kernels.

GPU path expects channels_last activations ([N,H,W,C] memory) — the
registry puts ResNet models in channels_last.  Parameters and running
stats stay fp32 (standard practice for BN in bf16 training).  The ReLU
(and its backward mask) is fused into the apply kernels.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import get_ext


class _BNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, res2d, weight, bias, mean, rstd, rm, rv,
                momentum, eps, relu, training):
        ext = get_ext(required=True)
        if training:
            y, mean, rstd = ext.bn_fwd(x2d, weight, bias, None, None,
                                       rm, rv, res2d, momentum, eps,
                                       relu)
        else:
            y, _, _ = ext.bn_fwd(x2d, weight, bias, mean, rstd, None,
                                 None, res2d, momentum, eps, relu)
        # relu WITHOUT residual: backward recomputes the mask from
        # x*scale+shift (needs beta) instead of re-reading y — one
        # fewer activation stream in both backward passes
        has_res = res2d is not None
        ctx.save_for_backward(x2d, y if (relu and has_res) else None,
                              mean, rstd, weight, bias)
        ctx.relu = relu
        ctx.has_res = has_res
        # y ONLY: returning mean/rstd made autograd materialize two
        # zero-grad tensors per BN backward (106 fp32 fills/iter on
        # ResNet-50)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x2d, y, mean, rstd, weight, bias = ctx.saved_tensors
        out = ext.bn_bwd(dy.contiguous(), x2d, y, mean,
                         rstd, weight, ctx.relu, ctx.has_res,
                         beta=None if y is not None else bias)
        dx, dbeta, dgamma = out[0], out[1], out[2]
        dres = out[3] if ctx.has_res else None
        return (dx, dres, dgamma, dbeta, None, None, None, None,
                None, None, None, None)


class FusedBN2d(nn.Module):
    def __init__(self, channels: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = False):
        super().__init__()
        self.channels = channels
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))
        self.register_buffer("running_mean", torch.zeros(channels))
        self.register_buffer("running_var", torch.ones(channels))

    def _cpu(self, x, residual=None):
        y = F.batch_norm(x.float(), self.running_mean, self.running_var,
                         self.weight, self.bias, self.training,
                         self.momentum, self.eps).to(x.dtype)
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.relu else y

    def forward(self, x, residual=None):
        if not x.is_cuda:
            return self._cpu(x, residual)
        N, C, H, W = x.shape
        x2d = x.contiguous(memory_format=torch.channels_last) \
               .permute(0, 2, 3, 1).reshape(-1, C)
        res2d = None
        if residual is not None:
            res2d = residual.contiguous(
                memory_format=torch.channels_last) \
                .permute(0, 2, 3, 1).reshape(-1, C)
        if self.training:
            mean = rstd = self.running_mean  # placeholders (unused)
        else:
            mean = self.running_mean
            rstd = (self.running_var + self.eps).rsqrt()
        y2d = _BNFn.apply(
            x2d, res2d, self.weight, self.bias, mean, rstd,
            self.running_mean, self.running_var, self.momentum,
            self.eps, self.relu, self.training)
        return y2d.view(N, H, W, C).permute(0, 3, 1, 2)

    def extra_repr(self):
        return f"{self.channels}, relu={self.relu}"
