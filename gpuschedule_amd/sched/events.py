"""Scheduler event log (JSONL) and JCT/makespan metrics.

Output-log format (the framework's documented, stable format — one JSON
object per line)::

    {"t": <float seconds>, "event": "<kind>", "job": "<job_id>", ...}

Event kinds and extra fields:

* ``submit``    — n_gpus, model, iters
* ``start``     — gpus (list), first (bool: first ever start)
* ``preempt``   — reason ("policy"|"quantum"|"resize"|"migrate"), evicted
* ``resume``    — gpus
* ``migrate``   — src, dst (lists)
* ``resize``    — old_gpus, new_gpus (lists)
* ``complete``  — jct, iters
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
        return {
            "n_jobs": float(len(submits)),
            "n_complete": float(n),
            "avg_jct": sum(jcts) / n if n else 0.0,
            "p50_jct": pct(0.50),
            "p95_jct": pct(0.95),
            "makespan": makespan,
            "avg_queue_delay": sum(qdelays) / len(qdelays) if qdelays else 0.0,
            "preemptions": float(sum(1 for e in self.events
                                     if e["event"] == "preempt")),
            "migrations": float(sum(1 for e in self.events
                                    if e["event"] == "migrate")),
        }
