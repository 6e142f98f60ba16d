"""Unit coverage: cluster memory accounting, catalog, event metrics."""
import pytest

from gpuschedule_amd import Cluster, EventLog, JobSpec
from gpuschedule_amd import catalog
from gpuschedule_amd.trace.job import Job, MI355X_HBM_BYTES

GB = 1024**3


def J(jid, model="resnet18", n=1):
    return JobSpec(job_id=jid, submit_time=0, n_gpus=n, total_iters=10,
                   model=model)


def test_hbm_default_is_288gb():
    assert MI355X_HBM_BYTES == 288 * GB
    assert Cluster(1).gpu(0).hbm_bytes == 288 * GB


def test_residency_and_release():
    c = Cluster(2, 2)
    a = Job(spec=J("a"))
    c.place(a, [0], 10 * GB)
    a.placement = [0]
    assert c.gpu(0).resident_bytes == 10 * GB
    c.stop(a)  # suspend: stays resident
    assert c.gpu(0).running is None
    assert "a" in c.gpu(0).resident
    c.release(a)
    assert c.gpu(0).resident_bytes == 0


def test_update_mem_adjusts_residency():
    c = Cluster(1)
    a = Job(spec=J("a"))
    c.place(a, [0], 10 * GB)
    a.placement = [0]
    c.update_mem("a", 4 * GB)
    assert c.gpu(0).resident_bytes == 4 * GB
    assert c.job_mem("a") == 4 * GB
    # floor at 256 MB
    c.update_mem("a", 1)
    assert c.job_mem("a") == 256 * 1024**2
    # unknown job: no-op
    c.update_mem("nope", 1 * GB)


def test_contiguous_window_within_node():
    c = Cluster(16, 8)
    # a 4-gang must not straddle the node boundary (gpus 6..9 invalid)
    for i in (5, 6, 7, 10, 11, 12):
        c.gpu(i).running_set.add("x")
    got = c.find_placement(4, 1 * GB)
    assert got is not None
    s = got[0]
    assert got == list(range(s, s + 4))
    assert s // 8 == (s + 3) // 8, "gang straddles node boundary"


def test_catalog_sim_iter_time_scales():
    spec = J("a", model="resnet50", n=4)
    t1 = catalog.sim_iter_time(spec, 1)
    t4 = catalog.sim_iter_time(spec, 4)
    assert t4 > t1  # all-reduce overhead
    assert t4 < 1.5 * t1


def test_catalog_unknown_model():
    with pytest.raises(ValueError):
        catalog.model_meta("nope")


def test_summary_percentiles():
    log = EventLog()
    for i, jct in enumerate([1.0, 2.0, 3.0, 4.0, 100.0]):
        jid = f"j{i}"
        log.emit(0.0, "submit", jid)
        log.emit(0.1, "start", jid, gpus=[0])
        log.emit(jct, "complete", jid, jct=jct, iters=1)
    s = log.summary()
    assert s["avg_jct"] == pytest.approx(22.0)
    assert s["p50_jct"] == 3.0
    assert s["p95_jct"] == 100.0
    assert s["makespan"] == 100.0


def test_chrome_trace_closes_open_runs_at_eof():
    # a job still running when the log ends must get a slice closed at
    # the last event time (not be dropped from the timeline)
    from gpuschedule_amd import EventLog
    log = EventLog()
    log.emit(0.0, "submit", "a", n_gpus=1, model="resnet18", iters=10)
    log.emit(1.0, "start", "a", gpus=[0], first=True)
    log.emit(5.0, "tick", util=1.0, running=1, queued=0)
    tr = log.to_chrome_trace()
    slices = [e for e in tr["traceEvents"] if e.get("ph") == "X"]
    assert len(slices) == 1
    s = slices[0]
    assert s["name"] == "a" and s["tid"] == 0
    assert s["ts"] == 1.0e6 and s["dur"] == 4.0e6


def test_job_tokens_per_iter_width_and_families():
    from gpuschedule_amd.catalog import job_tokens_per_iter
    from gpuschedule_amd.trace.job import JobSpec

    def J(model, batch=0):
        return JobSpec(job_id="x", submit_time=0.0, n_gpus=1,
                       model=model, total_iters=1, batch_size=batch)
    # image models count samples; sequence models count tokens
    assert job_tokens_per_iter(J("resnet50", 64)) == 64
    assert job_tokens_per_iter(J("bert-base", 32)) == 32 * 128
    assert job_tokens_per_iter(J("llama3-8b", 1)) == 2048
    # DP width multiplies whole-gang throughput
    assert job_tokens_per_iter(J("bert-base", 32), width=4) == 4 * 32 * 128
