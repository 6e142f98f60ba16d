"""Numerics + perf A/B for the v2 (3-buffer pipelined) GEMM kernel.

Usage on a GPU box:  PYTHONPATH=. python scripts/gemm_v2_check.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.ops import get_ext  # noqa: E402


def timeit(fn, warmup=10, iters=30):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def check(ext, M, N, K, bias=False, relu=False, tag=""):
    torch.manual_seed(1)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bt = (torch.randn(N, device="cuda", dtype=torch.bfloat16)
          if bias else None)
    c = ext.gemm_nt(a, b, bt, relu)
    ref = a.float() @ b.float().t()
    if bias:
        ref = ref + bt.float()
    if relu:
        ref = ref.relu()
    err = (c.float() - ref).abs()
    scale = ref.abs().max().item() + 1e-3
    rel = err.max().item() / scale
    ok = rel < 1.5e-2
    print(f"  {tag} M{M} N{N} K{K} bias={bias} relu={relu}: "
          f"maxrel={rel:.2e} {'OK' if ok else 'FAIL'}", flush=True)
    return ok


def main():
    ext = get_ext(required=True)
    print("== numerics (GSAMD_GEMM_V2 in env decides the kernel) ==",
          flush=True)
    ok = True
    ok &= check(ext, 4096, 4096, 4096)
    ok &= check(ext, 4096, 4096, 4096, bias=True, relu=True)
    ok &= check(ext, 4096, 28672, 4096)
    ok &= check(ext, 8192, 8192, 8192)
    ok &= check(ext, 4000, 4060, 4096, tag="tails")  # M/N tails
    assert ok, "NUMERICS FAILED"

    print("== perf ==", flush=True)
    for name, M, N, K in [("square4k", 4096, 4096, 4096),
                          ("square8k", 8192, 8192, 8192),
                          ("llamaffn", 4096, 28672, 4096)]:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t = timeit(lambda: ext.gemm_nt(a, b, None, False))
        tt = timeit(lambda: a @ b.t())
        print(f"  {name}: ours {2*M*N*K/t/1e12:.0f} TF | "
              f"hipBLASLt {2*M*N*K/tt/1e12:.0f} TF | "
              f"ratio {tt/t:.3f}", flush=True)


if __name__ == "__main__":
    main()
