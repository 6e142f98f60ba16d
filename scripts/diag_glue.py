"""Attribute residual torch-glue kernels (fills/copies/adds) to op sites.

Runs a few eager iterations (graphs disabled so the profiler keeps op
correlation) of one job per family and prints the op-level aggregation
of every kernel whose name matches the glue patterns, with source
stacks.  Diagnostic only — not part of the test suite.
"""
import os
import sys

os.environ["GSAMD_NO_GRAPHS"] = "1"

import torch  # noqa: E402
from torch.profiler import ProfilerActivity, profile  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.exec.trainer import JobRunner  # noqa: E402

GLUE = ("fill", "copy", "Functor_add", "direct_copy")


def run(model, iters=4):
    r = JobRunner(f"diag-{model}", model, 0, "cuda:0", torch.bfloat16,
                  None, 0, 1)
    for _ in range(3):
        r.step(1)
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU,
                             ProfilerActivity.CUDA],
                 with_stack=True) as p:
        for _ in range(iters):
            r.step(1)
    print(f"\n===== {model} ({iters} iters) =====")
    ka = p.key_averages(group_by_stack_n=5)
    rows = []
    for e in ka:
        dev_us = e.self_device_time_total
        if dev_us <= 0:
            continue
        rows.append((dev_us, e))
    rows.sort(key=lambda t: -t[0])
    total = sum(d for d, _ in rows)
    print(f"total device time {total/1e3:.2f} ms; ops touching glue:")
    for dev_us, e in rows:
        name = e.key
        if not any(g in name for g in
                   ("zero", "fill", "copy", "add", "cat", "to")):
            continue
        if dev_us / total < 0.002:
            continue
        print(f"  {dev_us/1e3:8.2f} ms  n={e.count:5d}  {name[:70]}")
        if e.stack:
            for fr in e.stack[:4]:
                print(f"      {fr[-100:]}")
    r.release()
    del r
    torch.cuda.empty_cache()


if __name__ == "__main__":
    for m in (sys.argv[1:] or ["bert-base", "resnet50"]):
        run(m)
