"""Numerics + perf for gemm_nn (B red-major, tr_b16 fragment reads)
vs the transpose2d + gemm_nt composite it replaces."""
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


def check(ext, M, R, N, tag=""):
    torch.manual_seed(2)
    a = torch.randn(M, R, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(R, N, device="cuda", dtype=torch.bfloat16)
    c = ext.gemm_nn(a, b, None, False)
    ref = a.float() @ b.float()
    rel = (c.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    print(f"  {tag} M{M} R{R} N{N}: maxrel={rel:.2e} "
          f"{'OK' if rel < 1.5e-2 else 'FAIL'}", flush=True)
    return rel < 1.5e-2


def main():
    ext = get_ext(required=True)
    ok = True
    ok &= check(ext, 4096, 4096, 4096)
    ok &= check(ext, 4096, 28672, 4096, "ffn-dx")
    ok &= check(ext, 4096, 2304, 768 * 8, "wide")
    ok &= check(ext, 4000, 4096, 4056, "tails")
    assert ok, "NUMERICS FAILED"
    for tag, M, R, N in [("dx4k", 4096, 4096, 4096),
                         ("ffn-dx", 4096, 28672, 4096)]:
        a = torch.randn(M, R, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(R, N, device="cuda", dtype=torch.bfloat16)
        t_nn = timeit(lambda: ext.gemm_nn(a, b, None, False))
        t_tr = timeit(lambda: ext.gemm_nt(a, ext.transpose2d(b), None,
                                          False))
        t_bl = timeit(lambda: a @ b)
        f = 2 * M * N * R / 1e12
        print(f"  {tag}: nn {f/t_nn:.0f} TF | transpose+nt {f/t_tr:.0f}"
              f" TF | hipBLASLt {f/t_bl:.0f} TF", flush=True)


if __name__ == "__main__":
    main()
