"""Philly-style trace loading and saving.

Trace schema (CSV, header required; this is the framework's documented
input format — a subset of the fields in Microsoft's Philly cluster
trace, named compatibly)::

    job_id,submit_time,num_gpus,iterations,model,batch_size,mem_bytes,elastic,min_gpus,max_gpus,iter_time_hint,background,fail_at_iter

Required columns: ``job_id, submit_time, num_gpus, iterations, model``.
Optional columns default to 0/false and model defaults.
``submit_time`` is in seconds relative to trace start.  Rows need not be
sorted; the loader sorts by (submit_time, job_id).
"""
from __future__ import annotations

import csv
import io
from typing import Iterable, List, Union

from .job import JobSpec

REQUIRED = ("job_id", "submit_time", "num_gpus", "iterations", "model")
OPTIONAL = ("batch_size", "mem_bytes", "elastic", "min_gpus", "max_gpus",
            "iter_time_hint", "background", "fail_at_iter")


def _parse_bool(s: str) -> bool:
    return s.strip().lower() in ("1", "true", "yes", "y")


def load_trace(src: Union[str, io.TextIOBase]) -> List[JobSpec]:
    """Load a trace from a CSV path or open text stream."""
    close = False
    if isinstance(src, str):
        src = open(src, "r", newline="")
        close = True
    try:
        reader = csv.DictReader(src)
        if reader.fieldnames is None:
            raise ValueError("empty trace file")
        missing = [c for c in REQUIRED if c not in reader.fieldnames]
        if missing:
            raise ValueError(f"trace missing required columns: {missing}")
        specs = []
        for row in reader:
            if not row.get("job_id"):
                continue
            specs.append(JobSpec(
                job_id=row["job_id"].strip(),
                submit_time=float(row["submit_time"]),
                n_gpus=int(row["num_gpus"]),
                total_iters=int(row["iterations"]),
                model=row["model"].strip(),
                batch_size=int(row.get("batch_size") or 0),
                mem_bytes=int(row.get("mem_bytes") or 0),
                elastic=_parse_bool(row.get("elastic") or ""),
                min_gpus=int(row.get("min_gpus") or 1),
                max_gpus=int(row.get("max_gpus") or 8),
                iter_time_hint=float(row.get("iter_time_hint") or 0.0),
                background=_parse_bool(row.get("background") or ""),
                fail_at_iter=int(row.get("fail_at_iter") or 0),
            ))
        specs.sort(key=lambda s: (s.submit_time, s.job_id))
        ids = [s.job_id for s in specs]
        if len(set(ids)) != len(ids):
            raise ValueError("duplicate job_id in trace")
        return specs
    finally:
        if close:
            src.close()


def save_trace(specs: Iterable[JobSpec], dst: Union[str, io.TextIOBase]) -> None:
    """Write a trace CSV (inverse of :func:`load_trace`)."""
    close = False
    if isinstance(dst, str):
        dst = open(dst, "w", newline="")
        close = True
    try:
        w = csv.writer(dst)
        w.writerow(list(REQUIRED) + list(OPTIONAL))
        for s in sorted(specs, key=lambda s: (s.submit_time, s.job_id)):
            w.writerow([s.job_id, s.submit_time, s.n_gpus, s.total_iters,
                        s.model, s.batch_size, s.mem_bytes,
                        int(s.elastic), s.min_gpus, s.max_gpus,
                        s.iter_time_hint, int(s.background),
                        s.fail_at_iter])
    finally:
        if close:
            dst.close()
