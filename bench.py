#!/usr/bin/env python3
"""Flagship benchmark: Philly-style trace replay through the scheduler
on N MI355X GPUs — the BASELINE.json headline metric
(avg JCT + makespan on a Philly-trace subset, per policy).

Driver contract:
  python bench.py --gpus N --steps K --warmup W
(N>1 is launched via torch.distributed.run, one rank per GPU, RCCL.)

A "step" is one scheduler tick: every placed job runs a fixed
iteration quota (real fwd+bwd+fused-optimizer training on the gfx950
HIP kernel path), then the policy re-evaluates.  The timed region is
EXACTLY K ticks, bracketed by barrier + torch.cuda.synchronize on both
sides.  The trace's job budgets are sized in *tick-slot* space so every
trace job completes inside the window independent of wall-clock
iteration times (sum over jobs of running-ticks <= 70% of K slots —
ONE GPU's window, so the identical trace replays at every N and the
strong-scaling metric is honest); per-GPU best-effort fill jobs
(excluded from JCT/makespan, real training work) keep every GPU busy
for all K ticks, so no timed tick does less than N GPUs of real work.

value = avg JCT + makespan over the trace jobs, in wall seconds
(lower is better).  Strong scaling: the SAME trace (single-GPU jobs,
ResNet-50 + BERT-base mix, Tiresias-LAS policy) replays at every N.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

from gpuschedule_amd import Cluster, JobSpec, Scheduler, make_policy
from gpuschedule_amd.exec import TorchBackend
from gpuschedule_amd.exec.backend import init_distributed


def build_trace(n_jobs: int, steps: int, n_gpus: int, iters_per_tick: int,
                seed: int = 1234):
    """Single-GPU job mix whose total tick-slot demand fits 70% of the
    K x N slot budget (identical job set at every N for strong
    scaling).  Job SERVICE DEMANDS and ARRIVAL ORDER come from the
    Philly-trace sample (data/philly_sample.json, the msr-fiddle
    cluster_job_log schema — see trace/philly_log.py): each trace job's
    logged GPU-time becomes its share of the tick-slot budget, and its
    submit offset lands proportionally inside the first 25% of the K
    ticks, so the replayed subset keeps the trace's heavy-tailed size
    distribution and burstiness at any hardware speed.

    Returns (arrivals, fill_specs): arrivals is a list of
    (arrival_tick, JobSpec)."""
    from gpuschedule_amd.trace.philly_log import load_philly_log
    fixture = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "data", "philly_sample.json")
    singles = [s for s in load_philly_log(fixture, time_scale=1 / 3600,
                                          seed=seed) if s.n_gpus == 1]
    philly = [singles[i % len(singles)] for i in range(n_jobs)]
    # STRONG scaling: the trace is IDENTICAL at every N, sized so the
    # total tick-slot demand fits one GPU's window (the smallest
    # configuration) — at larger N the same jobs spread across GPUs and
    # avg JCT + makespan genuinely shrink.
    budget = int(0.70 * steps) * iters_per_tick
    models = ["resnet50", "bert-base"]
    raw = [max(s.total_iters, 1) for s in philly]      # logged demand
    subs = [s.submit_time for s in philly]
    span = (max(subs) - min(subs)) or 1.0
    scale = budget / sum(raw)
    arrivals = []
    for i, s in enumerate(philly):
        iters = max(iters_per_tick, int(raw[i] * scale))
        tick = min(int((subs[i] - min(subs)) / span * 0.25 * steps),
                   max(int(0.25 * steps) - 1, 0))
        arrivals.append((tick, JobSpec(
            job_id=f"job{i:03d}", submit_time=0.0, n_gpus=1,
            total_iters=iters, model=models[i % 2])))
    arrivals.sort(key=lambda a: a[0])
    # per-GPU fill jobs: best-effort, never complete inside the window.
    # At N>=2 the first fill is a 2-GPU DDP gang so the scale bench
    # exercises gang scheduling + bucketed RCCL all-reduce over xGMI
    # (fill jobs are excluded from JCT/makespan, so the strong-scaling
    # metric over the 1-GPU trace jobs is unaffected).
    fill = []
    g = 0
    while g < n_gpus:
        width = 2 if (g == 0 and n_gpus >= 2) else 1
        fill.append(JobSpec(
            job_id=f"bg{g}", submit_time=0.0, n_gpus=width,
            total_iters=10**9,
            model=("bert-base" if g % 2 == 0 else "resnet50"),
            background=True))
        g += width
    return arrivals, fill


def run_phase(backend, ticks: int, arrivals=None, rank: int = 0):
    for t in range(ticks):
        if rank == 0 and arrivals:
            for at, spec in arrivals:
                if at == t:
                    backend.sched.submit_now(spec, backend.now())
        if backend.tick_once():
            break


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=40)
    ap.add_argument("--warmup", type=int, default=6)
    ap.add_argument("--policy", default="las")
    ap.add_argument("--jobs", type=int, default=0,
                    help="trace jobs (default 8; same trace at every "
                         "N for strong scaling)")
    ap.add_argument("--iters-per-tick", type=int, default=2)
    ap.add_argument("--device", default=None,
                    help="cuda (default) or cpu (debug)")
    args = ap.parse_args()

    rank, world = init_distributed("gloo")
    n_gpus = max(args.gpus, world)
    assert world in (1, n_gpus), \
        f"launched with world={world} but --gpus={args.gpus}"
    device_type = args.device or ("cuda" if torch.cuda.is_available()
                                  else "cpu")
    if device_type == "cuda":
        torch.cuda.set_device(rank)
    n_jobs = args.jobs or 8

    def make_backend(specs, policy, policy_kw=None):
        sched = None
        if rank == 0:
            sched = Scheduler(Cluster(n_gpus, max(n_gpus, 8)),
                              make_policy(policy, **(policy_kw or {})))
            sched.submit(specs)
        return TorchBackend(
            sched, rank=rank, world=world, device_type=device_type,
            dtype=torch.bfloat16, iters_per_tick=args.iters_per_tick,
            gang_backend=("nccl" if device_type == "cuda" else "gloo"))

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if device_type == "cuda":
            torch.cuda.synchronize()

    # ---- warmup: prewarm the runner pool, then a fill-only replay.
    # Prewarming pre-builds enough runners per model family to cover
    # the peak number of concurrently-resident jobs on this rank
    # (model construction + eager warmup + hipGraph capture + MIOpen
    # find all happen HERE, untimed); jobs arriving inside the timed
    # window adopt a warm runner, so the measured ms/step at the
    # driver's short --steps equals the long-replay steady state.
    from gpuschedule_amd.exec.trainer import RUNNER_POOL
    RUNNER_POOL.max_bytes = 160 << 30
    warm_specs = [JobSpec(job_id=f"warm{g}", submit_time=0.0, n_gpus=1,
                          total_iters=10**9,
                          model=("bert-base" if g % 2 else "resnet50"),
                          background=True) for g in range(n_gpus)]
    wb = make_backend(warm_specs, "fifo")
    if device_type == "cuda":
        per_fam = -(-n_jobs // (2 * n_gpus)) + 1  # ceil + fill job
        wb.prewarm(["resnet50", "bert-base"], count=per_fam)
        if world >= 2:
            # the scale bench's 2-GPU DDP gang fill job (bg0)
            wb.prewarm_gang("bert-base", [0, 1])
        torch.cuda.synchronize()
    tw0 = time.perf_counter()
    run_phase(wb, max(args.warmup, 1))
    if device_type == "cuda":
        torch.cuda.synchronize()
    # calibrated steady-state tick time (runners come pre-warmed from
    # the pool, so no cold-start bias)
    tick_s = (time.perf_counter() - tw0) / max(args.warmup, 1)
    if world > 1:
        t = torch.tensor([tick_s], dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        tick_s = float(t.item())
    wb.retire_all()
    if device_type == "cuda":
        torch.cuda.empty_cache()

    # ---- timed region: EXACTLY K ticks of the real trace
    arrivals, fill = build_trace(n_jobs, args.steps, n_gpus,
                                 args.iters_per_tick)
    policy_kw = {}
    if args.policy == "las":
        # LAS thresholds in GPU-seconds, scaled to the workload window
        horizon = args.steps * tick_s
        policy_kw = {"thresholds": (horizon / 8, horizon / 2),
                     "starve_limit": 4.0}
    elif args.policy == "gandiva":
        policy_kw = {"quantum": max(4 * tick_s, 1e-3)}
    be = make_backend(fill, args.policy, policy_kw)
    barrier_sync()
    t0 = time.perf_counter()
    run_phase(be, args.steps, arrivals, rank)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        log = be.sched.log
        trace_ids = {spec.job_id for _, spec in arrivals}
        completes = [e for e in log.events if e["event"] == "complete"
                     and e["job"] in trace_ids]
        submits = {e["job"]: e["t"] for e in log.events
                   if e["event"] == "submit" and e["job"] in trace_ids}
        n_done = len(completes)
        jcts = [e["jct"] for e in completes]
        avg_jct = sum(jcts) / max(1, n_done)
        makespan = (max(e["t"] for e in completes) - min(submits.values())
                    if completes else elapsed)
        value = avg_jct + makespan
        out = {
            "metric": "avg JCT + makespan on Philly-trace subset "
                      "(seconds; avg_jct + makespan)",
            "value": round(value, 3),
            "unit": "s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000 / args.steps, 3),
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic (random tokens/images, random-init "
                    "weights; no datasets in env)",
            "config": {
                "model": "mixed ResNet-50 + BERT-base single-GPU jobs",
                "policy": args.policy,
                "trace_jobs": n_jobs,
                "jobs_completed": n_done,
                "avg_jct_s": round(avg_jct, 3),
                "makespan_s": round(makespan, 3),
                "global_batch": f"64/img-job, 32/seq-job per GPU",
                "seq_len": 128,
                "iters_per_tick": args.iters_per_tick,
                "parallelism": f"scheduler over {n_gpus} GPUs, "
                               f"1-GPU jobs, RCCL control+gangs",
            },
        }
        print(json.dumps(out))
        if n_done < len(trace_ids):
            print(f"WARNING: {len(trace_ids) - n_done} trace jobs did "
                  f"not complete within {args.steps} ticks",
                  file=sys.stderr)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
