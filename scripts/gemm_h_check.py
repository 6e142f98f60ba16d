"""Numerics + perf for the p8h (256x128) mid-grid GEMM path."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.ops import get_ext  # noqa: E402


def timeit(fn, w=10, it=30):
    for _ in range(w):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it


def main():
    ext = get_ext(required=True)
    shapes = [(2048, 4096, 4096), (2048, 4096, 14336),
              (2048, 6144, 4096), (4096, 2304, 768), (4096, 3072, 768),
              (2000, 4096, 4096)]
    for M, N, K in shapes:
        torch.manual_seed(1)
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        c = ext.gemm_nt(a, b, bias, True)
        ref = (a.float() @ b.float().t() + bias.float()).relu()
        rel = (c.float() - ref).abs().max().item() / \
            (ref.abs().max().item() + 1e-3)
        t = timeit(lambda: ext.gemm_nt(a, b, None, False))
        tt = timeit(lambda: a @ b.t())
        f = 2 * M * N * K / 1e12
        print(f"M{M} N{N} K{K}: rel={rel:.1e} "
              f"{'OK' if rel < 1.5e-2 else 'FAIL'} | ours {f/t:.0f} TF | "
              f"blas {f/tt:.0f} TF | ratio {tt/t:.2f}", flush=True)


if __name__ == "__main__":
    main()
