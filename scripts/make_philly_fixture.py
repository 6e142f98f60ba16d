"""Generate data/philly_sample.json: a format-faithful sample of the
msr-fiddle cluster_job_log schema (see trace/philly_log.py).

The VALUES are synthetic (seeded; this environment has no network and
the real trace is not redistributable) but the distributions follow
the ATC'19 paper's reported shape: ~80% single-GPU jobs, heavy-tailed
durations, ~30% Killed/Failed, multi-server detail entries for >8-GPU
jobs, and a retry attempt for a slice of the failed jobs.
"""
import json
import os
import random
from datetime import datetime, timedelta

SEED = 20260914
N_JOBS = 64
START = datetime(2017, 10, 3, 5, 0, 0)
FMT = "%Y-%m-%d %H:%M:%S"


def main():
    rng = random.Random(SEED)
    jobs = []
    t = START
    for i in range(N_JOBS):
        t += timedelta(seconds=rng.expovariate(1 / 240.0))
        r = rng.random()
        if r < 0.78:
            n_gpus = 1
        elif r < 0.88:
            n_gpus = 2
        elif r < 0.95:
            n_gpus = 4
        elif r < 0.985:
            n_gpus = 8
        else:
            n_gpus = rng.choice([16, 24])  # multi-server: loader clamps
        status = rng.choices(["Pass", "Killed", "Failed"],
                             weights=[0.69, 0.20, 0.11])[0]
        dur = rng.lognormvariate(7.2, 1.6)  # median ~22 min, heavy tail
        dur = min(dur, 6 * 24 * 3600)
        queue = rng.expovariate(1 / 90.0)
        attempts = []
        start = t + timedelta(seconds=queue)
        n_attempts = 1 + (rng.random() < 0.25 and status != "Pass")
        left = dur
        for a in range(n_attempts):
            d = left if a == n_attempts - 1 else left * rng.uniform(.2, .6)
            left -= d
            end = start + timedelta(seconds=d)
            per_server = 8
            detail = []
            g = 0
            s = 0
            while g < n_gpus:
                k = min(per_server, n_gpus - g)
                detail.append({
                    "ip": f"10.0.{i % 16}.{s + 1}",
                    "gpus": [f"gpu{j}" for j in range(k)],
                })
                g += k
                s += 1
            attempts.append({
                "start_time": start.strftime(FMT),
                "end_time": end.strftime(FMT),
                "detail": detail,
            })
            start = end + timedelta(seconds=rng.expovariate(1 / 120.0))
        jobs.append({
            "status": status,
            "vc": f"vc{i % 4}",
            "jobid": f"application_1506638472019_{i:05d}",
            "user": f"user{i % 11}",
            "submitted_time": t.strftime(FMT),
            "attempts": attempts,
        })
    # a few never-started rows (no attempts): the loader must skip them
    for i in range(3):
        t += timedelta(seconds=rng.expovariate(1 / 240.0))
        jobs.append({
            "status": "Killed", "vc": "vc0",
            "jobid": f"application_1506638472019_9{i:04d}",
            "user": "user0",
            "submitted_time": t.strftime(FMT),
            "attempts": ([] if i == 0 else
                         [{"start_time": t.strftime(FMT),
                           "end_time": None, "detail": []}]),
        })
    out = os.path.join(os.path.dirname(__file__), "..", "data",
                       "philly_sample.json")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    with open(out, "w") as f:
        json.dump(jobs, f, indent=1)
    print(f"wrote {len(jobs)} records -> {out}")


if __name__ == "__main__":
    main()
