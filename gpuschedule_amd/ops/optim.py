"""Fused optimizers (bf16 params + fp32 master weights and state).

One HIP kernel launch per parameter per step (ops/hip/optim.hip),
reading/writing every state tensor exactly once — instead of torch's
multi-kernel unfused update.  CPU fallback does the same math in torch
(used by the no-GPU tests and for fp32 parameters).
"""
from __future__ import annotations


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

    @torch.no_grad()
    def reset_state(self):
        """Reset optimizer state IN PLACE (fresh-job semantics) without
        replacing any state tensor: the multi-tensor pointer tables and
        any hipGraph that baked them stay valid.  Masters re-snapshot
        the (re-initialized) params; moments/momentum zero; the device
        step buffer rewinds to 0."""
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if not st:
                    continue
                if "step" in st:
                    st["step"] = 0
                if "p32" in st:
                    st["p32"].copy_(p.detach().float())
                for k in ("m", "v", "buf"):
                    if k in st:
                        st[k].zero_()
        mt = getattr(self, "_mt", None)
        if isinstance(mt, tuple):   # AdamW: (..., stepbuf, params)
            mt[4].zero_()

    @torch.no_grad()
    def init_state(self):
        """Materialize state for every trainable param (the step()
        lazy-init requires grads).  Elastic joiners call this so the
        state tensors exist as broadcast destinations."""
        for group in self.param_groups:
            for p in group["params"]:
                if not p.requires_grad or self.state.get(p):
                    continue
                st = self.state[p]
                if isinstance(self, FusedAdamW):
                    st["step"] = 0
                st["p32"] = p.detach().float().clone()
                if isinstance(self, FusedAdamW):
                    st["m"] = torch.zeros_like(st["p32"])
                    st["v"] = torch.zeros_like(st["p32"])
                else:
                    st["buf"] = torch.zeros_like(st["p32"])

    def true_step(self) -> int:
        """Current optimizer step count, reading the device counter
        when the multi-tensor path (possibly under graph replay) is
        active."""
        mt = getattr(self, "_mt", None)
        if isinstance(mt, tuple):
            return int(mt[4][0].item())
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if st and "step" in st:
                    return st["step"]
        return 0

    def set_step(self, step: int) -> None:
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if st and "step" in st:
                    st["step"] = step
        mt = getattr(self, "_mt", None)
        if isinstance(mt, tuple):
            mt[4][0] = float(step)


def _mt_eligible(p) -> bool:
    """Param qualifies for the flat multi-tensor kernels: any DENSE
    layout works (the kernels are elementwise over data_ptr+numel) as
    long as param and grad share identical strides — channels_last
    conv weights included (``is_contiguous()`` alone rejected every
    channels_last ResNet weight, silently forcing the per-tensor
    path: ~160 optimizer launches/iter instead of 2)."""
    g = p.grad
    if g is None or not p.is_cuda:
        return False
    if g.stride() != p.stride():
        return False
    return (p.is_contiguous() or
            p.is_contiguous(memory_format=torch.channels_last))


class FusedAdamW(_MasterStateMixin, torch.optim.Optimizer):
    """Fused AdamW.  When every param of a group is a bf16 CUDA tensor
    with a grad, the MULTI-TENSOR path runs: ONE kernel updates all
    params AND zeroes the grads in place (``fused_zero_grad`` tells
    the trainer to skip zero_grad, which keeps grad tensors stable —
    required for the chunk table and for hipGraph capture).  The step
    counter/bias corrections live in a device buffer so graph replays
    keep advancing them."""

    MT_CHUNK = 8192

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.fused_zero_grad = False
        self._mt = None  # (ptrs, sizes, chunk_t, chunk_off, stepbuf, params)

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        self._mt = None  # state tensors replaced: table is stale
        self.fused_zero_grad = False

    def state_dict(self):
        # sync host step mirrors from the device counter (python-side
        # step() does not run under hipGraph replay)
        if self._mt is not None:
            step = int(self._mt[4][0].item())
            for p in self._mt[5]:
                self.state[p]["step"] = step
        return super().state_dict()

    def _try_build_mt(self, group):
        params = [p for p in group["params"] if p.requires_grad]
        if not params or not all(
                p.dtype == torch.bfloat16 and _mt_eligible(p)
                for p in params):
            return None
        dev = params[0].device
        ptrs, sizes, ct, co = [], [], [], []
        step0 = None
        for ti, p in enumerate(params):
            st = self.state[p]
            ptrs.append([p.data.data_ptr(), st["p32"].data_ptr(),
                         p.grad.data_ptr(), st["m"].data_ptr(),
                         st["v"].data_ptr()])
            n = p.numel()
            sizes.append(n)
            step0 = st["step"] if step0 is None else step0
            if st["step"] != step0:
                return None  # inconsistent restore; stay per-tensor
            off = 0
            while off < n:
                ct.append(ti)
                co.append(off)
                off += self.MT_CHUNK
        sb = torch.zeros(3, dtype=torch.float32, device=dev)
        sb[0] = float(step0)
        return (torch.tensor(ptrs, dtype=torch.int64, device=dev),
                torch.tensor(sizes, dtype=torch.int64, device=dev),
                torch.tensor(ct, dtype=torch.int32, device=dev),
                torch.tensor(co, dtype=torch.int64, device=dev),
                sb, params)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            # lazy state init for all params with grads
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state["step"] = 0
                    state["p32"] = p.detach().float().clone()
                    state["m"] = torch.zeros_like(state["p32"])
                    state["v"] = torch.zeros_like(state["p32"])
            # multi-tensor fast path
            if self._mt is None and len(self.param_groups) == 1:
                self._mt = self._try_build_mt(group)
                if self._mt is not None:
                    self.fused_zero_grad = True
            if self._mt is not None:
                ext = get_ext(required=True)
                ptrs, sizes, ct, co, sb, params = self._mt
                ext.adamw_multi(ptrs, sizes, ct, co, sb, lr, beta1,
                                beta2, eps, wd)
                return loss
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
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
    """Fused SGD-momentum with a multi-tensor fast path: one launch
    per dtype class (bf16 params w/ fp32 masters, fp32 params) that
    also zeroes the grads in place (see FusedAdamW)."""

    MT_CHUNK = 8192

    def __init__(self, params, lr=0.1, momentum=0.9, weight_decay=5e-4):
        defaults = dict(lr=lr, momentum=momentum,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.fused_zero_grad = False
        self._mt = None  # list of (ptrs, sizes, ct, co, bf16)

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        self._mt = None
        self.fused_zero_grad = False

    def _try_build_mt(self, group):
        params = [p for p in group["params"] if p.requires_grad]
        if not params or not all(
                p.dtype in (torch.bfloat16, torch.float32)
                and _mt_eligible(p) for p in params):
            return None
        dev = params[0].device
        tables = []
        for want_bf16 in (True, False):
            sel = [p for p in params
                   if (p.dtype == torch.bfloat16) == want_bf16]
            if not sel:
                continue
            ptrs, sizes, ct, co = [], [], [], []
            for ti, p in enumerate(sel):
                st = self.state[p]
                p32 = st["p32"].data_ptr() if want_bf16 else 0
                ptrs.append([p.data.data_ptr(), p32,
                             p.grad.data_ptr(), st["buf"].data_ptr()])
                n = p.numel()
                sizes.append(n)
                off = 0
                while off < n:
                    ct.append(ti)
                    co.append(off)
                    off += self.MT_CHUNK
            tables.append((
                torch.tensor(ptrs, dtype=torch.int64, device=dev),
                torch.tensor(sizes, dtype=torch.int64, device=dev),
                torch.tensor(ct, dtype=torch.int32, device=dev),
                torch.tensor(co, dtype=torch.int64, device=dev),
                want_bf16))
        return tables

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
            if self._mt is None and len(self.param_groups) == 1:
                self._mt = self._try_build_mt(group)
                if self._mt:
                    self.fused_zero_grad = True
            if self._mt:
                ext = get_ext(required=True)
                for ptrs, sizes, ct, co, bf16 in self._mt:
                    ext.sgd_multi(ptrs, sizes, ct, co, lr, mom, wd,
                                  bf16)
                return loss
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
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
