"""Linear layer with dual GEMM dispatch (see _use_lib):

* GSAMD_GEMM_LIB=1 (opt-in): plain projection GEMMs via hipBLASLt
  (F.linear) — native NN/TN backward, measured 10-13% faster end to
  end (profiles/gemm_mode_compare.json).
* GSAMD_GEMM_LIB=0 (default): the hand-written bf16 MFMA GEMM
  (ops/hip/gemm_mfma.hip, NT layout) with the fused
  bias+residual-addend epilogue.  Backward reuses the same NT kernel
  with fast LDS-tiled transposes:

      dx[M,K] = dy[M,N] @ W[N,K]        = gemm_nt(dy, W^T)
      dW[N,K] = dy^T[N,M] @ x[M,K]      = gemm_nt(dy^T, x^T)
      db      = dy.sum over rows

  The reduction dim of every gemm_nt call must be a multiple of 64;
  the wrapper zero-pads the token dim when needed (zero rows
  contribute nothing to dW / db).
"""
from __future__ import annotations

import math
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import get_ext


def _use_lib() -> bool:
    """GSAMD_GEMM_LIB=1 routes MFMALinear's plain projection GEMMs
    through hipBLASLt (F.linear): its native NN/TN operand paths make
    backward transpose-free, and measured end-to-end it is faster —
    llama3-8b 183.3 -> 162.4 ms/iter (1.13x), bert-base 11.7 -> 10.6
    (1.10x) (scripts/gemm_mode_compare.py, profiles/).  It is an
    OPT-IN deployment knob: the DEFAULT (=0) is the hand-written
    gfx950 MFMA kernel (0.77-0.94x hipBLASLt depending on shape, PMC
    evidence in profiles/) with the fused epilogue — bias + residual
    addend with d(add)=dy identity, zero extra kernels at residual
    junctions — because this framework's contract (BASELINE.json
    north_star) specifies hand-written CDNA4 GEMM kernels in the
    training step, visible in rocprof.  Read per call so tests/benches
    can toggle; a hipGraph capture bakes whichever path was active at
    capture time."""
    return os.environ.get("GSAMD_GEMM_LIB", "0") == "1"


def _pad_rows64(t: torch.Tensor) -> torch.Tensor:
    m = t.shape[0]
    pad = (-m) % 64
    if pad == 0:
        return t
    return F.pad(t, (0, 0, 0, pad))


def _gemm_nt(ext, a, b, bias=None, relu=False, add=None):
    """gemm_nt with zero-padding of the reduction dim when K % 64 != 0
    (zero columns contribute nothing to the product)."""
    k = a.shape[1]
    pad = (-k) % 64
    if pad:
        a = F.pad(a, (0, pad))
        b = F.pad(b, (0, pad))
    return ext.gemm_nt(a.contiguous(), b.contiguous(), bias, relu,
                       add=add)


class _MFMALinearFn(torch.autograd.Function):
    """relu and the fused elementwise addend (residual stream) are
    mutually exclusive here: the epilogue applies add BEFORE relu and
    no model composes both.  d(add) = dy exactly (identity), so the
    residual junction costs ZERO kernels in both directions."""

    @staticmethod
    def forward(ctx, x2d, weight, bias, relu, add):
        ext = get_ext(required=True)
        y = _gemm_nt(ext, x2d, weight, bias, relu, add=add)
        ctx.save_for_backward(x2d, weight, y if relu else None)
        ctx.relu = relu
        ctx.has_bias = bias is not None
        ctx.has_add = add is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x2d, weight, y = ctx.saved_tensors
        dy = dy.contiguous()
        if ctx.relu:
            dy = dy * (y > 0).to(dy.dtype)
        M, N = dy.shape
        K = weight.shape[1]
        # dx = dy @ W: gemm_nn consumes W exactly as stored ([N,K] =
        # red-major) via ds_read_b64_tr_b16 fragment reads — no W^T
        # pass.  MEASURED: ties the transpose+NT composite (the
        # transpose pass was ~2-4%, and the tr-read path gives the
        # same back in fragment-read overhead), so the proven
        # composite stays default; GSAMD_GEMM_NN=1 enables the native
        # path (numerics-tested either way).
        import os
        blocks = ((M + 255) // 256) * ((K + 255) // 256)
        if (os.environ.get("GSAMD_GEMM_NN") == "1"
                and N % 64 == 0 and K % 8 == 0 and M >= 256 and K >= 256
                and blocks >= 200):
            dx = ext.gemm_nn(dy, weight, None, False)
        else:
            wt = ext.transpose2d(weight)        # [K, N]
            dx = _gemm_nt(ext, dy, wt)          # [M, K]
        dyp = _pad_rows64(dy)
        xp = _pad_rows64(x2d)
        if ctx.has_bias:
            dyt, db = ext.transpose2d_colsum(dyp)  # [N, Mp] + bf16 db
        else:
            dyt = ext.transpose2d(dyp)
            db = None
        xt = ext.transpose2d(xp)                # [K, Mp]
        dw = _gemm_nt(ext, dyt, xt)             # [N, K]
        return dx, dw, db, None, (dy if ctx.has_add else None)


class MFMALinear(nn.Module):
    """Drop-in nn.Linear running on the gfx950 MFMA GEMM.

    ``relu=True`` fuses the activation into the GEMM epilogue (and its
    mask into backward).
    """

    def __init__(self, in_features: int, out_features: int,
                 bias: bool = True, relu: bool = False):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.relu = relu
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    def forward(self, x, add=None):
        """``add``: optional residual stream fused into the GEMM
        epilogue (y = x @ W^T + b + add); its grad is dy (identity)."""
        assert not (self.relu and add is not None), \
            "MFMALinear: relu and fused addend are exclusive"
        shape = x.shape
        x2d = x.reshape(-1, shape[-1])
        if x.is_cuda:
            x2d = x2d.contiguous()
            add2d = (add.reshape(-1, self.out_features).contiguous()
                     if add is not None else None)
            if _use_lib():
                y = F.linear(x2d, self.weight, self.bias)
                if self.relu:
                    y = F.relu(y)
                if add2d is not None:
                    y = y + add2d
            else:
                y = _MFMALinearFn.apply(x2d, self.weight, self.bias,
                                        self.relu, add2d)
        else:
            y = F.linear(x2d, self.weight, self.bias)
            if self.relu:
                y = F.relu(y)
            if add is not None:
                y = y + add.reshape(-1, self.out_features)
        return y.reshape(*shape[:-1], self.out_features)

    def extra_repr(self):
        return (f"in={self.in_features}, out={self.out_features}, "
                f"bias={self.bias is not None}, relu={self.relu}")
