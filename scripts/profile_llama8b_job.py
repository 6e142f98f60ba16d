"""rocprofv3 workload: 3 iterations of the Llama-3-8B job."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.exec import JobRunner  # noqa: E402

r = JobRunner("l8b", "llama3-8b", 1, "cuda:0", torch.bfloat16,
              None, 0, 1)
import time
r.step(4)  # warmup + capture
torch.cuda.synchronize()
t0 = time.perf_counter()
n, secs = r.step(3)  # profiled replays
print(f"ms/iter {secs / n * 1000:.1f}")
print("loss", r.last_loss)
