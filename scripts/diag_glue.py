"""Attribute residual torch-glue kernels (fills/copies/adds) to the
training-step phase that launches them.

Runs eager iterations (graphs disabled so op correlation survives)
with record_function brackets around each phase, then attributes every
aten::fill_/copy_/add_ call to its enclosing phase by CPU-interval
containment.  Diagnostic only — not part of the test suite.
"""
import os
import sys
from collections import defaultdict

os.environ["GSAMD_NO_GRAPHS"] = "1"

import torch  # noqa: E402
from torch.profiler import ProfilerActivity, profile, record_function  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd.exec.trainer import JobRunner  # noqa: E402
from gpuschedule_amd.models import compute_loss  # noqa: E402

GLUE_OPS = ("aten::fill_", "aten::zero_", "aten::zeros", "aten::copy_",
            "aten::add_", "aten::add", "aten::contiguous", "aten::cat",
            "aten::to", "aten::_to_copy")


def run(model, iters=4):
    r = JobRunner(f"diag-{model}", model, 0, "cuda:0", torch.bfloat16,
                  None, 0, 1)
    for _ in range(3):
        r.step(1)
    torch.cuda.synchronize()

    def one_iter():
        with record_function("PH_datagen"):
            x, y = r._batch(r.done_iters)
        with record_function("PH_forward"):
            loss = compute_loss(r.model_name, r.model, x, y)
        with record_function("PH_backward"):
            loss.backward()
        with record_function("PH_optim"):
            r.optim.step()
        r.done_iters += 1

    with profile(activities=[ProfilerActivity.CPU,
                             ProfilerActivity.CUDA]) as p:
        for _ in range(iters):
            one_iter()
        torch.cuda.synchronize()

    # phase intervals on each CPU thread
    phases = []
    for e in p.events():
        if e.name.startswith("PH_"):
            phases.append((e.time_range.start, e.time_range.end,
                           e.name, e.thread))
    agg = defaultdict(lambda: [0, 0.0])

    def phase_of(e):
        for s, en, name, tid in phases:
            if tid == e.thread and s <= e.time_range.start <= en:
                return name
        return "other"

    for e in p.events():
        if e.name in GLUE_OPS and not e.cpu_parent:
            k = (phase_of(e), e.name)
            agg[k][0] += 1
            agg[k][1] += e.device_time_total
        elif e.name in GLUE_OPS and e.cpu_parent \
                and e.cpu_parent.name not in GLUE_OPS \
                and not e.cpu_parent.name.startswith(("autograd",)):
            k = (phase_of(e), f"{e.cpu_parent.name} -> {e.name}")
            agg[k][0] += 1
            agg[k][1] += e.device_time_total

    print(f"\n===== {model} ({iters} iters) =====")
    for (ph, op), (n, us) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
        if us < 50:
            continue
        print(f"  {us/1e3:8.2f} ms  n={n:5d}  {ph:12s} {op[:70]}")
    r.release()
    del r
    torch.cuda.empty_cache()


if __name__ == "__main__":
    for m in (sys.argv[1:] or ["bert-base", "resnet50"]):
        run(m)
