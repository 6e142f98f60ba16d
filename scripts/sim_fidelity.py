#!/usr/bin/env python3
"""Simulator fidelity check: replay the SAME trace through (a) the
discrete-event simulator and (b) the real Torch backend on the GPU, and
compare avg/p50/p95 JCT and makespan.

The simulator's per-model iteration times are calibrated to measured
MI355X hipGraph-replay times (catalog.py), so with a warm runner pool
the two should agree closely; this script quantifies how closely and is
the evidence behind the calibration claim in BASELINE.md.

Usage (GPU box):  python scripts/sim_fidelity.py
       (CPU dbg): python scripts/sim_fidelity.py --device cpu
"""
from __future__ import annotations

import argparse
import json
import time

import torch

from gpuschedule_amd import Cluster, JobSpec, Scheduler, make_policy
from gpuschedule_amd.exec import TorchBackend
from gpuschedule_amd.sim.simulator import simulate


def make_trace(scale: float = 1.0):
    """10 single-GPU jobs, resnet18/bert-base mix, staggered arrivals.
    Sized so each job's virtual service time is ~1-2 s (resnet18 b128
    = 5 ms/iter, bert-base b32 = 12.5 ms/iter measured)."""
    specs = []
    for i in range(10):
        if i % 2 == 0:
            model, iters = "resnet18", int(300 * scale)   # ~1.5 s
        else:
            model, iters = "bert-base", int(120 * scale)  # ~1.5 s
        specs.append(JobSpec(job_id=f"f{i:02d}", submit_time=0.4 * i,
                             n_gpus=1, model=model, total_iters=iters))
    return specs


def pick(s):
    return {k: round(s[k], 3) for k in
            ("avg_jct", "p50_jct", "p95_jct", "makespan",
             "avg_queue_delay", "n_complete")}


def per_job_jct(log):
    return {e["job"]: round(e["jct"], 3) for e in log.events
            if e["event"] == "complete"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--policy", default="srtf")
    ap.add_argument("--scale", type=float, default=1.0)
    ap.add_argument("--iters-per-tick", type=int, default=10)
    args = ap.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")

    # ---- (a) simulator (cold-pool and warm-pool variants) ---------
    slog, ssum = simulate(make_trace(args.scale), make_policy(args.policy),
                          n_gpus=1, reeval_s=0.05, warm_pool=True)
    _, csum = simulate(make_trace(args.scale), make_policy(args.policy),
                       n_gpus=1, reeval_s=0.05, warm_pool=False)

    # ---- (b) real backend ----------------------------------------
    sched = Scheduler(Cluster(1, 8), make_policy(args.policy))
    sched.submit(make_trace(args.scale))
    be = TorchBackend(sched, rank=0, world=1, device_type=device,
                      dtype=torch.bfloat16,
                      iters_per_tick=args.iters_per_tick)
    if device == "cuda":
        be.prewarm(["resnet18", "bert-base"], count=1)
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100_000):
        if be.tick_once():
            break
    wall = time.perf_counter() - t0
    rsum = be.sched.log.summary()

    ratios = {k: (round(rsum[k] / ssum[k], 3) if ssum[k] else None)
              for k in ("avg_jct", "p50_jct", "p95_jct", "makespan",
                        "avg_queue_delay")}
    out = {
        "policy": args.policy, "device": device,
        "n_jobs": 10, "wall_s": round(wall, 2),
        "sim_warm_pool": pick(ssum), "sim_cold": pick(csum),
        "real": pick(rsum),
        "real_over_sim": ratios,
        "per_job_jct_sim": per_job_jct(slog),
        "per_job_jct_real": per_job_jct(be.sched.log),
    }
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
