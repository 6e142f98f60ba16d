"""Tick-time breakdown for the trace-replay bench (debug tool)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from gpuschedule_amd import Cluster, Scheduler, make_policy  # noqa: E402
from gpuschedule_amd.exec import TorchBackend  # noqa: E402
from gpuschedule_amd.exec.trainer import JobRunner  # noqa: E402
from bench import build_trace, run_phase  # noqa: E402


def main():
    steps, ipt = 40, 2
    arrivals, fill = build_trace(8, steps, 1, ipt)
    sched = Scheduler(Cluster(1, 8),
                      make_policy("las", thresholds=(1.0, 4.0)))
    sched.submit(fill)
    be = TorchBackend(sched, device_type="cuda", iters_per_tick=ipt)

    stats = {"batch": 0.0, "iter": 0.0, "n": 0}
    per_job = {}
    ob, oi = JobRunner._batch, JobRunner._one_iter

    def wb(self, it):
        t0 = time.perf_counter()
        r = ob(self, it)
        stats["batch"] += time.perf_counter() - t0
        return r

    def wi(self, x, y):
        t0 = time.perf_counter()
        r = oi(self, x, y)
        dt = time.perf_counter() - t0
        stats["iter"] += dt
        stats["n"] += 1
        k = (self.job_id, self.model_name, self._graph is not None)
        a = per_job.setdefault(k, [0.0, 0])
        a[0] += dt
        a[1] += 1
        return r

    JobRunner._batch, JobRunner._one_iter = wb, wi
    run_phase(be, 6)
    torch.cuda.synchronize()
    stats.update(batch=0.0, iter=0.0, n=0)
    per_job.clear()
    t0 = time.perf_counter()
    run_phase(be, steps, arrivals, 0)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    n = stats["n"]
    print(f"total {el/steps*1e3:.1f} ms/tick, {n} iters "
          f"({n/steps:.1f}/tick)")
    print(f"  _batch {stats['batch']/steps*1e3:.2f} ms/tick, "
          f"_one_iter {stats['iter']/steps*1e3:.2f} ms/tick "
          f"({stats['iter']/max(n,1)*1e3:.2f} ms/iter)")
    for k, (t, c) in sorted(per_job.items(), key=lambda x: -x[1][0]):
        print(f"  {k[0]:8s} {k[1]:10s} graphed={k[2]} "
              f"{t/c*1e3:7.2f} ms/iter x{c}")


if __name__ == "__main__":
    main()
