"""rocprofv3 workload: 3 iterations of the Llama-3-8B job."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.exec import JobRunner  # noqa: E402

r = JobRunner("l8b", "llama3-8b", 1, "cuda:0", torch.bfloat16,
              None, 0, 1)
r.step(4)  # warmup + capture
r.step(3)  # profiled replays
print("loss", r.last_loss)
