"""Fused cross-entropy (bf16 logits, fp32 loss) — ops/hip/ce.hip."""
from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import get_ext


class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits2d, labels):
        ext = get_ext(required=True)
        loss, lse = ext.ce_fwd(logits2d, labels)
        ctx.save_for_backward(logits2d, labels, lse)
        return loss

    @staticmethod
    def backward(ctx, gout):
        ext = get_ext(required=True)
        logits2d, labels, lse = ctx.saved_tensors
        # gscale folds d(mean)/d(row) into the elementwise pass
        gscale = (gout.reshape(1).float() / logits2d.shape[0])
        dx = ext.ce_bwd(logits2d, labels, lse, gscale.contiguous())
        return dx, None


def fused_cross_entropy(logits: torch.Tensor,
                        labels: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over flattened rows.  GPU: one fused pass
    over bf16 logits; CPU: fp32 torch reference."""
    logits2d = logits.reshape(-1, logits.shape[-1])
    labels1d = labels.reshape(-1)
    if logits.is_cuda and logits.dtype == torch.bfloat16:
        return _CEFn.apply(logits2d.contiguous(), labels1d.contiguous())
    return F.cross_entropy(logits2d.float(), labels1d)
