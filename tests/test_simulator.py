"""Simulator correctness on hand-solvable traces + event-log format."""
import json

from gpuschedule_amd import EventLog, JobSpec, make_policy, make_trace, simulate


def J(jid, t=0.0, n=1, iters=10, hint=1.0, model="resnet18", **kw):
    return JobSpec(job_id=jid, submit_time=t, n_gpus=n, total_iters=iters,
                   model=model, iter_time_hint=hint, **kw)


def test_single_job_jct():
    # 10 iters x 1s + 0.5s cold start ~= 10.5s JCT
    log, s = simulate([J("a", iters=10)], make_policy("fifo"), n_gpus=1)
    assert abs(s["avg_jct"] - 10.5) < 0.3
    assert abs(s["makespan"] - s["avg_jct"]) < 1e-6


def test_two_sequential_jobs_makespan():
    specs = [J("a", t=0.0, iters=10), J("b", t=0.0, iters=10)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=1)
    # a: ~10.5, b waits then runs ~10.5 => makespan ~21, avg ~15.75
    assert 20.0 < s["makespan"] < 22.5
    assert 14.5 < s["avg_jct"] < 17.0


def test_parallel_jobs_independent():
    specs = [J("a", iters=10), J("b", iters=10)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=2)
    assert s["makespan"] < 12.0


def test_gang_allreduce_overhead():
    one, _ = simulate([J("a", n=1, iters=50, hint=0.0)],
                      make_policy("fifo"), n_gpus=8)
    four, _ = simulate([J("a", n=4, iters=50, hint=0.0)],
                       make_policy("fifo"), n_gpus=8)
    t1 = one.summary()["makespan"]
    t4 = four.summary()["makespan"]
    assert t4 > t1  # gang pays all-reduce overhead (weak scaling)
    assert t4 < 1.3 * t1


def test_event_log_jsonl_roundtrip(tmp_path):
    specs = make_trace(8, seed=2, mean_interarrival=2.0,
                       iters_range=(5, 20))
    log, s = simulate(specs, make_policy("srtf"), n_gpus=4)
    assert int(s["n_complete"]) == 8
    p = tmp_path / "events.jsonl"
    log.save(str(p))
    with open(p) as f:
        for line in f:
            rec = json.loads(line)
            assert "t" in rec and "event" in rec
    loaded = EventLog.load(str(p))
    assert loaded.summary() == s


def test_slowdown_metrics():
    # solo job on an idle cluster: slowdown ~1; a queued job behind it
    # on 1 GPU has slowdown > 1
    specs = [J("a", t=0.0, iters=20, hint=1.0),
             J("b", t=0.0, iters=20, hint=1.0)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=1)
    assert 0.9 < s["avg_slowdown"] and s["max_slowdown"] >= 1.8
    _, s1 = simulate(specs[:1], make_policy("fifo"), n_gpus=1)
    assert 0.9 < s1["avg_slowdown"] < 1.2


def test_chrome_trace_export(tmp_path):
    specs = make_trace(6, seed=3, mean_interarrival=2.0,
                       iters_range=(5, 30), gang_sizes=[1, 1, 2])
    log, s = simulate(specs, make_policy("gandiva", quantum=3.0),
                      n_gpus=2)
    assert int(s["n_complete"]) == 6
    trace = log.to_chrome_trace()
    ev = trace["traceEvents"]
    slices = [e for e in ev if e["ph"] == "X"]
    # every job appears as at least one occupancy slice on a GPU track
    assert {e["name"] for e in slices} == {sp.job_id for sp in specs}
    for e in slices:
        assert e["dur"] >= 0 and e["tid"] in (0, 1)
    # instant submit markers on the sched track
    assert sum(1 for e in ev
               if e["ph"] == "i" and e["name"].startswith("submit")) == 6
    p = tmp_path / "trace.json"
    log.save_chrome_trace(str(p))
    with open(p) as f:
        assert json.load(f)["traceEvents"]


def test_all_jobs_complete_all_policies():
    specs = make_trace(24, mix="philly-slice", seed=11,
                       mean_interarrival=1.0, iters_range=(5, 60))
    for pol in ["fifo", "srtf", "las", "gandiva", "elastic"]:
        log, s = simulate(specs, make_policy(pol), n_gpus=4)
        assert int(s["n_complete"]) == 24, f"{pol} lost jobs"


def test_contention_separates_policies():
    # Heavy contention: SRTF should beat FIFO on avg JCT.
    specs = [J("big", t=0.0, iters=200)] + \
        [J(f"s{i}", t=1.0 + i, iters=5) for i in range(8)]
    _, fifo = simulate(specs, make_policy("fifo"), n_gpus=1)
    _, srtf = simulate(specs, make_policy("srtf"), n_gpus=1)
    assert srtf["avg_jct"] < 0.5 * fifo["avg_jct"]


def test_fault_injection_mid_run():
    # Job "dies" at iteration 5 of 50: it fails, frees its GPU, and the
    # queued job behind it still completes.
    specs = [J("dies", iters=50, fail_at_iter=5),
             J("ok", t=0.5, iters=5)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=1)
    fails = [e for e in log.events if e["event"] == "fail"]
    assert len(fails) == 1 and fails[0]["job"] == "dies"
    assert "injected fault" in fails[0]["error"]
    assert int(s["n_complete"]) == 1
    completes = [e["job"] for e in log.events if e["event"] == "complete"]
    assert completes == ["ok"]


def test_unsatisfiable_job_fails_at_admission():
    specs = [J("wide", n=4, iters=5), J("ok", n=1, iters=5)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=2)
    fails = [e for e in log.events if e["event"] == "fail"]
    assert len(fails) == 1 and fails[0]["job"] == "wide"
    assert int(s["n_complete"]) == 1


def test_complete_event_reports_tokens():
    from gpuschedule_amd import JobSpec, make_policy
    from gpuschedule_amd.sim.simulator import simulate
    specs = [JobSpec(job_id="a", submit_time=0.0, n_gpus=2,
                     total_iters=10, model="bert-base", batch_size=4)]
    log, s = simulate(specs, make_policy("fifo"), n_gpus=2)
    done = [e for e in log.events if e["event"] == "complete"][0]
    # 10 iters x batch 4 x 128 tokens x 2 GPUs
    assert done["tokens"] == 10 * 4 * 128 * 2
    assert done["tokens_per_s"] > 0
    assert s["tokens_per_s_sum"] > 0


def test_warm_pool_skips_cold_start():
    # warm_pool models the backend's prewarmed RunnerPool: first
    # placement costs RESUME_HBM_S (0.05s) instead of COLD_START_S
    # (0.5s); a 10x1s job's JCT drops by ~0.45s and nothing else moves.
    _, cold = simulate([J("a", iters=10)], make_policy("fifo"), n_gpus=1)
    _, warm = simulate([J("a", iters=10)], make_policy("fifo"), n_gpus=1,
                       warm_pool=True)
    assert abs(cold["avg_jct"] - 10.5) < 0.3
    assert abs(warm["avg_jct"] - 10.05) < 0.3
    assert warm["avg_jct"] < cold["avg_jct"] - 0.3


def test_placement_overhead_model():
    """Unit-test the documented per-placement overhead model
    (sim/simulator.py docstring): resume-in-HBM is cheapest, overlap
    resizes ride xGMI broadcast (150 GB/s), migrations pay D2H+H2D
    (48 GB/s each way), evicted restores pay H2D only."""
    from gpuschedule_amd import Cluster, JobSpec, Scheduler
    from gpuschedule_amd.sched.scheduler import Placement
    from gpuschedule_amd.sim.simulator import (
        COLD_START_S, D2H_BYTES_PER_S, H2D_BYTES_PER_S, RESUME_HBM_S,
        XGMI_BYTES_PER_S, Simulator)

    sched = Scheduler(Cluster(4, 4), make_policy("fifo"))
    spec = JobSpec(job_id="a", submit_time=0.0, n_gpus=1,
                   model="resnet50", total_iters=10)
    sched.submit([spec])
    sched.tick(0.0)
    sim = Simulator(sched)
    job = sched.jobs["a"]
    from gpuschedule_amd import catalog
    ckpt = catalog.job_mem_bytes(spec) / 3

    def ov(**kw):
        return sim._placement_overhead(Placement(job_id="a", gpus=[0],
                                                 **kw), job)

    assert ov() == COLD_START_S
    assert ov(resume=True) == RESUME_HBM_S
    assert abs(ov(resume=True, resize_from=2, resize_from_gpus=[0, 1])
               - ckpt / XGMI_BYTES_PER_S) < 1e-9      # overlap: xGMI bcast
    assert abs(ov(resume=True, migrate_from=[1])
               - (ckpt / D2H_BYTES_PER_S + ckpt / H2D_BYTES_PER_S)) < 1e-9
    assert abs(ov(resume=True, from_evicted=True)
               - ckpt / H2D_BYTES_PER_S) < 1e-9
    # disjoint resize (no surviving rank) = checkpoint round-trip
    assert abs(ov(resume=True, resize_from=1, resize_from_gpus=[3])
               - (ckpt / D2H_BYTES_PER_S + ckpt / H2D_BYTES_PER_S)) < 1e-9
    # warm pool: fresh placements cost RESUME_HBM_S instead
    assert Simulator(sched, warm_pool=True)._placement_overhead(
        Placement(job_id="a", gpus=[0]), job) == RESUME_HBM_S


def test_summary_counts_failures_and_tokens():
    from gpuschedule_amd import EventLog
    log = EventLog()
    log.emit(0.0, "submit", "a", n_gpus=1, model="bert-base", iters=4,
             ideal_s=2.0)
    log.emit(0.0, "submit", "b", n_gpus=1, model="bert-base", iters=4,
             ideal_s=2.0)
    log.emit(0.1, "start", "a", gpus=[0], first=True)
    log.emit(0.1, "start", "b", gpus=[1], first=True)
    log.emit(1.0, "fail", "b", error="boom")
    log.emit(4.0, "complete", "a", jct=4.0, iters=4, tokens=16384,
             tokens_per_s=4096.0)
    s = log.summary()
    assert s["n_jobs"] == 2 and s["n_complete"] == 1 and s["n_failed"] == 1
    assert s["tokens_per_s_sum"] == 4096.0
    assert abs(s["avg_slowdown"] - 2.0) < 1e-9   # jct 4 / ideal 2
    assert s["makespan"] == 4.0
