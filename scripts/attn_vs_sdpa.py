#!/usr/bin/env python3
"""Measure our HIP flash-attention (ops/hip/attention.hip) against
PyTorch's stock scaled_dot_product_attention on MI355X, at the two
shapes the framework's models actually run:

  llama:  B1  H32 HKV8 S2048 D128 causal (GQA)   [llama3-8b]
  bert:   B32 H12        S128  D64  non-causal    [bert-base]

SDPA dispatches to whatever the wheel ships for ROCm (AOTriton flash /
MIOpen / math); we report which backend it picked. GQA for SDPA uses
enable_gqa=True when available, else expanded K/V (whichever is faster
is the fair stock baseline).
"""
import json
import time

import torch
import torch.nn.functional as F

from gpuschedule_amd.ops.attention import flash_attention


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) * 1e3 / iters


def bench_shape(name, B, H, HK, S, D, causal):
    torch.manual_seed(0)
    dt = torch.bfloat16
    q = torch.randn(B, H, S, D, device="cuda", dtype=dt)
    k = torch.randn(B, HK, S, D, device="cuda", dtype=dt)
    v = torch.randn(B, HK, S, D, device="cuda", dtype=dt)

    def ours_fwd():
        return flash_attention(q, k, v, causal=causal)

    def sdpa_fwd():
        try:
            return F.scaled_dot_product_attention(
                q, k, v, is_causal=causal, enable_gqa=(HK != H))
        except TypeError:
            ke = k.repeat_interleave(H // HK, dim=1)
            ve = v.repeat_interleave(H // HK, dim=1)
            return F.scaled_dot_product_attention(q, ke, ve,
                                                  is_causal=causal)

    def make_bwd(f):
        qa = q.clone().requires_grad_(True)
        ka = k.clone().requires_grad_(True)
        va = v.clone().requires_grad_(True)
        if f is ours_fwd:
            def run():
                o = flash_attention(qa, ka, va, causal=causal)
                o.backward(torch.ones_like(o))
                qa.grad = ka.grad = va.grad = None
        else:
            def run():
                try:
                    o = F.scaled_dot_product_attention(
                        qa, ka, va, is_causal=causal,
                        enable_gqa=(HK != H))
                except TypeError:
                    o = F.scaled_dot_product_attention(
                        qa, ka.repeat_interleave(H // HK, dim=1),
                        va.repeat_interleave(H // HK, dim=1),
                        is_causal=causal)
                o.backward(torch.ones_like(o))
                qa.grad = ka.grad = va.grad = None
        return run

    # numerics cross-check vs fp32 math reference
    with torch.no_grad():
        ref = F.scaled_dot_product_attention(
            q.float(), k.repeat_interleave(H // HK, dim=1).float(),
            v.repeat_interleave(H // HK, dim=1).float(), is_causal=causal)
        err = (ours_fwd().float() - ref).abs().max().item()

    res = {
        "shape": name, "B": B, "H": H, "HK": HK, "S": S, "D": D,
        "causal": causal, "max_abs_err_vs_fp32": round(err, 4),
        "ours_fwd_ms": round(timeit(ours_fwd), 3),
        "sdpa_fwd_ms": round(timeit(sdpa_fwd), 3),
        "ours_fwdbwd_ms": round(timeit(make_bwd(ours_fwd)), 3),
        "sdpa_fwdbwd_ms": round(timeit(make_bwd(sdpa_fwd)), 3),
    }
    res["fwd_speedup"] = round(res["sdpa_fwd_ms"] / res["ours_fwd_ms"], 3)
    res["fwdbwd_speedup"] = round(
        res["sdpa_fwdbwd_ms"] / res["ours_fwdbwd_ms"], 3)
    return res


def main():
    torch.cuda.init()
    out = {"torch": torch.__version__,
           "sdpa_flash_enabled": torch.backends.cuda.flash_sdp_enabled(),
           "results": [
               bench_shape("llama3-8b", 1, 32, 8, 2048, 128, True),
               bench_shape("bert-base", 32, 12, 12, 128, 64, False),
           ]}
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
