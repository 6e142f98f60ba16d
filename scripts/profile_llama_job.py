"""Profile workload: 30 iterations of a llama3-tiny training job (used
under rocprofv3 to show RoPE/SwiGLU/attention/RMSNorm/GEMM kernels)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.exec import JobRunner  # noqa: E402

r = JobRunner("lt", "llama3-tiny", 16, "cuda:0", torch.bfloat16,
              None, 0, 1)
r.step(30)
print("loss", r.last_loss)
