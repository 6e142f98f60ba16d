"""Scheduler event log (JSONL) and JCT/makespan metrics.

Output-log format (the framework's documented, stable format — one JSON
object per line)::

    {"t": <float seconds>, "event": "<kind>", "job": "<job_id>", ...}

Event kinds and extra fields:

* ``submit``    — n_gpus, model, iters, ideal_s (contention-free solo
  runtime, used by the finish-time-slowdown metric)
* ``start``     — gpus (list), first (bool: first ever start)
* ``preempt``   — reason ("policy"|"resize"|"migrate")
* ``evict``     — job's state left HBM for a host checkpoint
  (standalone event; follows a preempt when the job was running)
* ``resume``    — gpus
* ``migrate``   — src, dst (lists)
* ``resize``    — old_gpus, new_gpus (GPU counts before/after)
* ``complete``  — jct, iters, tokens, tokens_per_s (tokens for sequence models, samples for image models; whole-gang)
* ``fail``      — error
* ``tick``      — util, running, queued (cluster samples; optional)

Summary metrics are computed from completes: average / p50 / p95 JCT,
makespan (last completion − first submit), mean queueing delay (first
start − submit), total preemptions/migrations.
"""
from __future__ import annotations

import json
from typing import Dict, IO, List, Optional


class EventLog:
    def __init__(self, sink: Optional[IO[str]] = None):
        self.events: List[dict] = []
        self.sink = sink

    def emit(self, t: float, event: str, job: str = "", **kw) -> None:
        rec = {"t": round(float(t), 6), "event": event}
        if job:
            rec["job"] = job
        rec.update(kw)
        self.events.append(rec)
        if self.sink is not None:
            self.sink.write(json.dumps(rec) + "\n")

    def save(self, path: str) -> None:
        with open(path, "w") as f:
            for rec in self.events:
                f.write(json.dumps(rec) + "\n")

    @staticmethod
    def load(path: str) -> "EventLog":
        log = EventLog()
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line:
                    log.events.append(json.loads(line))
        return log

    # ---- visualization -------------------------------------------

    def to_chrome_trace(self) -> dict:
        """Render the schedule as a Chrome/Perfetto trace (load in
        chrome://tracing or ui.perfetto.dev): one track (tid) per GPU,
        one slice per contiguous occupancy interval of a job, instant
        markers for submits/evictions/migrations/resizes.  Times are
        microseconds as the format requires."""
        US = 1e6
        ev: List[dict] = [
            {"ph": "M", "name": "process_name", "pid": 0,
             "args": {"name": "cluster"}},
        ]
        open_runs: Dict[str, tuple] = {}   # job -> (t0, gpus)
        meta: Dict[str, dict] = {}
        gpus_seen = set()
        t_max = 0.0

        def close(job: str, t1: float) -> None:
            t0, gpus = open_runs.pop(job)
            for g in gpus:
                ev.append({"ph": "X", "name": job, "pid": 0, "tid": g,
                           "ts": t0 * US, "dur": max(t1 - t0, 0) * US,
                           "args": meta.get(job, {})})

        for e in self.events:
            t = e["t"]
            t_max = max(t_max, t)
            kind = e["event"]
            job = e.get("job", "")
            if kind == "submit":
                meta[job] = {"model": e.get("model", ""),
                             "n_gpus": e.get("n_gpus", 0),
                             "iters": e.get("iters", 0)}
                ev.append({"ph": "i", "name": f"submit {job}", "pid": 0,
                           "tid": "sched", "ts": t * US, "s": "g"})
            elif kind in ("start", "resume"):
                if job in open_runs:
                    close(job, t)
                gpus = e.get("gpus", [])
                gpus_seen.update(gpus)
                open_runs[job] = (t, gpus)
            elif kind in ("preempt", "complete", "fail"):
                if job in open_runs:
                    close(job, t)
            elif kind == "evict":
                ev.append({"ph": "i", "name": f"evict {job}",
                           "pid": 0, "tid": "sched", "ts": t * US,
                           "s": "g"})
            elif kind in ("migrate", "resize"):
                ev.append({"ph": "i", "name": f"{kind} {job}", "pid": 0,
                           "tid": "sched", "ts": t * US, "s": "g"})
        for job in list(open_runs):
            close(job, t_max)
        for g in sorted(gpus_seen):
            ev.append({"ph": "M", "name": "thread_name", "pid": 0,
                       "tid": g, "args": {"name": f"GPU {g}"}})
        return {"traceEvents": ev, "displayTimeUnit": "ms"}

    def save_chrome_trace(self, path: str) -> None:
        with open(path, "w") as f:
            json.dump(self.to_chrome_trace(), f)

    # ---- metrics -------------------------------------------------

    def summary(self) -> Dict[str, float]:
        submits = {e["job"]: e["t"] for e in self.events
                   if e["event"] == "submit"}
        starts: Dict[str, float] = {}
        for e in self.events:
            if e["event"] == "start" and e["job"] not in starts:
                starts[e["job"]] = e["t"]
        completes = {e["job"]: e for e in self.events
                     if e["event"] == "complete"}
        jcts = sorted(e["jct"] for e in completes.values())
        n = len(jcts)

        def pct(p: float) -> float:
            if not jcts:
                return 0.0
            return jcts[min(n - 1, int(p * n))]

        makespan = 0.0
        if completes and submits:
            makespan = (max(e["t"] for e in completes.values())
                        - min(submits.values()))
        qdelays = [starts[j] - submits[j] for j in starts if j in submits]
        # finish-time slowdown = JCT / contention-free solo runtime
        # (Tiresias/Themis-style fairness metric); ideal_s comes from
        # the submit events
        ideals = {e["job"]: e["ideal_s"] for e in self.events
                  if e["event"] == "submit" and e.get("ideal_s")}
        slowdowns = sorted(e["jct"] / ideals[j]
                           for j, e in completes.items() if j in ideals)
        ns = len(slowdowns)
        return {
            "avg_slowdown": sum(slowdowns) / ns if ns else 0.0,
            "p95_slowdown": (slowdowns[min(ns - 1, int(0.95 * ns))]
                             if ns else 0.0),
            "max_slowdown": slowdowns[-1] if ns else 0.0,
            "n_jobs": float(len(submits)),
            "n_complete": float(n),
            "n_failed": float(sum(1 for e in self.events
                                  if e["event"] == "fail")),
            "avg_jct": sum(jcts) / n if n else 0.0,
            "p50_jct": pct(0.50),
            "p95_jct": pct(0.95),
            "makespan": makespan,
            "avg_queue_delay": sum(qdelays) / len(qdelays) if qdelays else 0.0,
            "tokens_per_s_sum": round(sum(
                e.get("tokens_per_s", 0.0)
                for e in completes.values()), 1),
            "preemptions": float(sum(1 for e in self.events
                                     if e["event"] == "preempt")),
            "migrations": float(sum(1 for e in self.events
                                    if e["event"] == "migrate")),
        }
