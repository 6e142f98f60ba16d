"""Drive the p8 GEMM at the 3 hot shapes for a PMC counter run."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from gpuschedule_amd.ops import get_ext
ext = get_ext(required=True)
for M, N, K in [(4096, 4096, 4096), (8192, 8192, 8192),
                (4096, 28672, 4096)]:
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        ext.gemm_nt(a, b, None, False)
    torch.cuda.synchronize()
print("done")
