#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: custom HIP kernels vs torch.

Prints one line per case: achieved TFLOP/s (GEMM/attention) or TB/s
(norms/optim) and the torch (hipBLASLt / native) equivalent for
context.  Used to drive tuning; summaries are committed under
profiles/.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.ops import get_ext  # noqa: E402


def timeit(fn, warmup=10, iters=50):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemm(ext, out):
    print("== GEMM (bf16, C = A @ B^T) ==")
    shapes = [
        ("bert qkv", 4096, 2304, 768),
        ("bert ffn1", 4096, 3072, 768),
        ("bert head", 4096, 30528, 768),
        ("llama qkv", 4096, 6144, 4096),
        ("llama ffn", 4096, 28672, 4096),
        ("square 4k", 4096, 4096, 4096),
        ("square 8k", 8192, 8192, 8192),
    ]
    for name, M, N, K in shapes:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t_ours = timeit(lambda: ext.gemm_nt(a, b, None, False))
        t_torch = timeit(lambda: a @ b.t())
        tf = 2 * M * N * K / t_ours / 1e12
        tf_t = 2 * M * N * K / t_torch / 1e12
        print(f"  {name:12s} M{M} N{N} K{K}: ours {tf:7.1f} TF | "
              f"hipblaslt {tf_t:7.1f} TF | ratio {tf/tf_t:.2f}")
        out[f"gemm/{name}"] = {"ours_tf": round(tf, 1),
                               "torch_tf": round(tf_t, 1)}


def bench_norms(ext, out):
    print("== LayerNorm / RMSNorm (bf16) ==")
    for name, N, D in [("bert", 4096, 768), ("llama", 8192, 4096)]:
        x = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
        g = torch.ones(D, device="cuda", dtype=torch.bfloat16)
        b = torch.zeros(D, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn_like(x)
        t_f = timeit(lambda: ext.norm_fwd(x, None, g, b, 1e-5, False))
        y, _, mean, rstd = ext.norm_fwd(x, None, g, b, 1e-5, False)
        t_b = timeit(lambda: ext.norm_bwd(dy, x, mean, rstd, g, False))
        t_t = timeit(lambda: torch.nn.functional.layer_norm(
            x, (D,), g, b, 1e-5))
        bytes_f = 2 * N * D * 2  # read x + write y
        print(f"  LN {name:6s} [{N}x{D}] fwd {bytes_f/t_f/1e12:5.2f} TB/s "
              f"(torch fwd {bytes_f/t_t/1e12:5.2f}) bwd "
              f"{4*N*D*2/t_b/1e12:5.2f} TB/s")
        out[f"ln/{name}"] = {
            "fwd_tbs": round(bytes_f / t_f / 1e12, 2),
            "torch_fwd_tbs": round(bytes_f / t_t / 1e12, 2),
            "bwd_tbs": round(4 * N * D * 2 / t_b / 1e12, 2)}


def bench_attn(ext, out):
    print("== Flash attention forward (bf16) ==")
    shapes = [("bert", 32, 12, 128, 64, False),
              ("llama2k", 2, 32, 2048, 128, True),
              ("gqa4k", 1, 32, 4096, 128, True)]
    for name, B, H, S, D, causal in shapes:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k, v = torch.randn_like(q), torch.randn_like(q)
        t_ours = timeit(lambda: ext.attn_fwd(q, k, v, causal), iters=30)
        t_sdpa = timeit(lambda: torch.nn.functional.
                        scaled_dot_product_attention(q, k, v,
                                                     is_causal=causal),
                        iters=30)
        flops = 4 * B * H * S * S * D * (0.5 if causal else 1.0)
        tf, tf_s = flops / t_ours / 1e12, flops / t_sdpa / 1e12
        print(f"  {name:8s} B{B} H{H} S{S} D{D} causal={causal}: "
              f"ours {tf:6.1f} TF | sdpa {tf_s:6.1f} TF | "
              f"ratio {tf/tf_s:.2f}")
        out[f"attn/{name}"] = {"ours_tf": round(tf, 1),
                               "sdpa_tf": round(tf_s, 1)}


def bench_optim(ext, out):
    print("== Fused AdamW ==")
    n = 100 * 1024 * 1024
    p32 = torch.randn(n, device="cuda")
    p = p32.bfloat16()
    g = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    t = timeit(lambda: ext.adamw_step(p, p32, g, m, v, 1e-3, 0.9, 0.999,
                                      1e-8, 0.01, 5), iters=20)
    # traffic: read g(2)+m(4)+v(4)+p32(4), write m(4)+v(4)+p32(4)+p(2)
    tbs = n * 28 / t / 1e12
    print(f"  adamw 100M params: {t*1e3:.2f} ms = {tbs:.2f} TB/s")
    out["adamw"] = {"tbs": round(tbs, 2), "ms_100m": round(t * 1e3, 2)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--only", default=None,
                    choices=[None, "gemm", "norms", "attn", "optim"])
    ap.add_argument("--json", default=None)
    args = ap.parse_args()
    ext = get_ext(required=True)
    out = {}
    if args.only in (None, "gemm"):
        bench_gemm(ext, out)
    if args.only in (None, "norms"):
        bench_norms(ext, out)
    if args.only in (None, "attn"):
        bench_attn(ext, out)
    if args.only in (None, "optim"):
        bench_optim(ext, out)
    if args.json:
        with open(args.json, "w") as f:
            json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
