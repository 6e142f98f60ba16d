"""Fused optimizers (bf16 params + fp32 master weights and state).

One HIP kernel launch per parameter per step (ops/hip/optim.hip),
reading/writing every state tensor exactly once — instead of torch's
multi-kernel unfused update.  CPU fallback does the same math in torch
(used by the no-GPU tests and for fp32 parameters).
"""
from __future__ import annotations

import math

import torch

from ._ext import get_ext


class _MasterStateMixin:
    """Keep fp32 master/state fp32 across load_state_dict (torch's
    default casts floating state to the param dtype, which would
    quantize the masters to bf16 on restore)."""

    _FP32_KEYS = ("p32", "m", "v", "buf")

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if not st:
                    continue
                for k in self._FP32_KEYS:
                    if k in st and isinstance(st[k], torch.Tensor):
                        st[k] = st[k].to(device=p.device,
                                         dtype=torch.float32)


class FusedAdamW(_MasterStateMixin, torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state["step"] = 0
                    state["p32"] = p.detach().float().clone()
                    state["m"] = torch.zeros_like(state["p32"])
                    state["v"] = torch.zeros_like(state["p32"])
                state["step"] += 1
                t = state["step"]
                if p.is_cuda and p.dtype == torch.bfloat16:
                    ext = get_ext(required=True)
                    ext.adamw_step(p.data, state["p32"],
                                   p.grad.contiguous(), state["m"],
                                   state["v"], lr, beta1, beta2, eps, wd, t)
                else:
                    g = p.grad.float()
                    m, v, p32 = state["m"], state["v"], state["p32"]
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    mhat = m / (1 - beta1 ** t)
                    vhat = v / (1 - beta2 ** t)
                    p32.add_(mhat / (vhat.sqrt() + eps) + wd * p32,
                             alpha=-lr)
                    p.data.copy_(p32.to(p.dtype))
        return loss


class FusedSGD(_MasterStateMixin, torch.optim.Optimizer):
    def __init__(self, params, lr=0.1, momentum=0.9, weight_decay=5e-4):
        defaults = dict(lr=lr, momentum=momentum,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr, mom, wd = (group["lr"], group["momentum"],
                           group["weight_decay"])
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state["p32"] = p.detach().float().clone()
                    state["buf"] = torch.zeros_like(state["p32"])
                if p.is_cuda and p.dtype == torch.bfloat16:
                    ext = get_ext(required=True)
                    ext.sgd_step(state["p32"], p.data, p.grad.contiguous(),
                                 state["buf"], lr, mom, wd)
                elif p.is_cuda and p.dtype == torch.float32:
                    ext = get_ext(required=True)
                    ext.sgd_step_fp32(p.data, p.grad.contiguous(),
                                      state["buf"], lr, mom, wd)
                else:
                    g = p.grad.float() + wd * state["p32"]
                    state["buf"].mul_(mom).add_(g)
                    state["p32"].add_(state["buf"], alpha=-lr)
                    p.data.copy_(state["p32"].to(p.dtype))
        return loss
