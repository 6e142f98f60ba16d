"""Seeded synthetic trace generation (no network; no real datasets).

Generates Philly-shaped job mixes over the model families the framework
ships. Used by the eval configs and by ``bench.py`` (the driver's
benchmark runs on synthetic traces with random-init weights).
"""
from __future__ import annotations

import random
from typing import List, Optional, Sequence, Tuple

from .job import JobSpec

# (model, gang-size choices, weight) mixes per named workload.
MIXES = {
    # config 2: 16 ResNet-18 CIFAR-shape single-GPU jobs
    "resnet18-cifar": [("resnet18", (1,), 1.0)],
    # config 3: mixed ResNet-50 + BERT-base, 1-4 GPU gangs
    "mixed-cv-nlp": [("resnet50", (1, 2, 4), 0.5),
                     ("bert-base", (1, 2, 4), 0.5)],
    # config 4: 64-job Philly-like slice, mostly small jobs, a few gangs
    "philly-slice": [("resnet18", (1,), 0.35),
                     ("resnet50", (1, 2), 0.3),
                     ("bert-base", (1, 2, 4), 0.3),
                     ("llama3-tiny", (1,), 0.05)],
}


def make_trace(n_jobs: int,
               mix: str = "mixed-cv-nlp",
               seed: int = 0,
               mean_interarrival: float = 10.0,
               iters_range: Tuple[int, int] = (50, 400),
               gang_sizes: Optional[Sequence[int]] = None) -> List[JobSpec]:
    """Generate ``n_jobs`` jobs with Poisson arrivals and log-uniform
    iteration budgets (heavy-tailed durations, like Philly)."""
    rng = random.Random(seed)
    entries = MIXES[mix]
    weights = [w for _, _, w in entries]
    t = 0.0
    specs = []
    for i in range(n_jobs):
        model, sizes, _ = rng.choices(entries, weights=weights, k=1)[0]
        if gang_sizes is not None:
            sizes = tuple(gang_sizes)
        lo, hi = iters_range
        # log-uniform iteration count => heavy-tailed job durations
        iters = int(lo * (hi / lo) ** rng.random())
        specs.append(JobSpec(
            job_id=f"job{i:04d}",
            submit_time=round(t, 3),
            n_gpus=rng.choice(sizes),
            total_iters=iters,
            model=model,
        ))
        t += rng.expovariate(1.0 / mean_interarrival)
    return specs


def make_elastic_trace(seed: int = 0, total_iters: int = 400) -> List[JobSpec]:
    """Config-5 shape: one elastic Llama DP job that can grow 1->8 GPUs,
    plus a few small jobs that come and go to free GPUs over time."""
    rng = random.Random(seed)
    specs = [JobSpec(job_id="elastic0", submit_time=0.0, n_gpus=1,
                     total_iters=total_iters, model="llama3-8b",
                     elastic=True, min_gpus=1, max_gpus=8)]
    t = 0.0
    for i in range(6):
        t += rng.expovariate(1.0 / 20.0)
        specs.append(JobSpec(
            job_id=f"small{i}", submit_time=round(t, 3), n_gpus=1,
            total_iters=rng.randint(20, 80), model="resnet18"))
    return specs
