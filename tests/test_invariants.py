"""Black-box schedule invariants, reconstructed purely from event logs.

For randomized traces across every policy, replay the event stream and
assert cluster-level safety properties the scheduler must never
violate, regardless of policy:

* GPU occupancy never exceeds capacity (1 job/GPU; 2 when the policy
  packs),
* gang placements are contiguous GPU windows,
* a job is never running twice concurrently,
* every completed job ran exactly its specified iterations,
* preempted/evicted jobs were actually running when preempted.
"""
import pytest

from gpuschedule_amd import make_policy, make_trace, simulate


POLICIES = {
    "fifo": lambda: make_policy("fifo"),
    "srtf": lambda: make_policy("srtf"),
    "las": lambda: make_policy("las", thresholds=(20.0, 80.0)),
    "gandiva": lambda: make_policy("gandiva", quantum=5.0),
    "gandiva-pack": lambda: make_policy("gandiva", quantum=5.0,
                                        packing=True),
    "elastic": lambda: make_policy("elastic"),
}


def replay_and_check(log, capacity):
    running = {}        # job -> list of gpus
    occupancy = {}      # gpu -> set of jobs
    for e in log.events:
        kind, job = e["event"], e.get("job")
        if kind in ("start", "resume"):
            assert job not in running, \
                f"{job} started while already running @ {e['t']}"
            gpus = e["gpus"]
            assert gpus == sorted(gpus)
            # gang contiguity (single-node windows)
            if len(gpus) > 1:
                assert gpus == list(range(gpus[0], gpus[0] + len(gpus))), \
                    f"non-contiguous gang {gpus} for {job}"
            for g in gpus:
                occ = occupancy.setdefault(g, set())
                occ.add(job)
                assert len(occ) <= capacity, \
                    f"GPU {g} oversubscribed: {occ} @ {e['t']}"
            running[job] = gpus
        elif kind in ("preempt", "complete", "fail"):
            if kind == "preempt":
                assert job in running, \
                    f"preempt of non-running {job} @ {e['t']}"
            if job in running:
                for g in running.pop(job):
                    occupancy[g].discard(job)
        elif kind == "migrate":
            # emitted alongside preempt/resume pairs; nothing to track
            pass
    return running


@pytest.mark.parametrize("policy", sorted(POLICIES))
@pytest.mark.parametrize("seed", [0, 7])
def test_schedule_invariants(policy, seed):
    specs = make_trace(20, mix="philly-slice", seed=seed,
                       mean_interarrival=2.0, iters_range=(5, 60),
                       gang_sizes=[1, 1, 1, 2, 4])
    pol = POLICIES[policy]()
    log, s = simulate(specs, pol, n_gpus=4)
    capacity = 2 if getattr(pol, "packing", False) else 1
    leftover = replay_and_check(log, capacity)
    assert not leftover, f"jobs still 'running' at end: {leftover}"
    # all completions ran exactly the specified iterations
    iters = {sp.job_id: sp.total_iters for sp in specs}
    completes = [e for e in log.events if e["event"] == "complete"]
    for e in completes:
        assert e["iters"] == iters[e["job"]]
    # nothing lost: every admitted job either completed or failed
    done = {e["job"] for e in log.events
            if e["event"] in ("complete", "fail")}
    assert done == set(iters)


def test_gangs_never_straddle_nodes():
    """16-GPU, 2-node cluster: every gang start in the actual schedule
    stays within one 8-GPU node (xGMI is intra-node)."""
    from gpuschedule_amd import Cluster, Scheduler
    from gpuschedule_amd.sim.simulator import Simulator
    specs = make_trace(24, mix="philly-slice", seed=5,
                       mean_interarrival=1.0, iters_range=(5, 40),
                       gang_sizes=[1, 2, 4, 8])
    sched = Scheduler(Cluster(16, 8), make_policy("las"))
    sched.submit(specs)
    sim = Simulator(sched)
    sim.run()
    s = sched.log.summary()
    assert int(s["n_complete"]) == 24
    for e in sched.log.events:
        if e["event"] in ("start", "resume"):
            nodes = {g // 8 for g in e["gpus"]}
            assert len(nodes) == 1, f"gang straddles nodes: {e}"


def test_simulator_deterministic_replay():
    """Same trace + policy twice => bitwise-identical event streams."""
    specs = make_trace(16, mix="philly-slice", seed=9,
                       mean_interarrival=1.5, iters_range=(5, 50),
                       gang_sizes=[1, 2])
    log1, s1 = simulate(specs, make_policy("las"), n_gpus=4)
    log2, s2 = simulate(specs, make_policy("las"), n_gpus=4)
    assert log1.events == log2.events
    assert s1 == s2


def test_elastic_resize_events_consistent():
    from gpuschedule_amd.trace.synth import make_elastic_trace
    specs = make_elastic_trace(seed=3, total_iters=200)
    log, s = simulate(specs, make_policy("elastic"), n_gpus=4)
    assert int(s["n_complete"]) == len(specs)
    ids = {sp.job_id for sp in specs}
    resizes = [e for e in log.events if e["event"] == "resize"]
    assert resizes, "elastic trace on idle capacity must resize"
    for e in resizes:  # old_gpus/new_gpus are GPU counts
        assert e["job"] in ids
        assert e["new_gpus"] >= 1 and e["old_gpus"] >= 1
        assert e["new_gpus"] != e["old_gpus"]


def test_chrome_trace_evict_marker():
    """The standalone `evict` event renders as an instant marker in the
    Chrome/Perfetto export (the old dead preempt-evicted branch)."""
    from gpuschedule_amd.sched.events import EventLog
    log = EventLog()
    log.emit(0.0, "submit", "a", n_gpus=1, model="m", iters=2, ideal_s=1)
    log.emit(0.1, "start", "a", gpus=[0], first=True)
    log.emit(0.5, "preempt", "a", reason="policy")
    log.emit(0.5, "evict", "a")
    log.emit(1.0, "complete", "a", jct=1.0, iters=2)
    tr = log.to_chrome_trace()
    names = [e.get("name") for e in tr["traceEvents"]]
    assert "evict a" in names


def test_partial_gang_failure_report_order():
    """Partial gang failure: one member's fail report can be processed
    BEFORE the leader's ok report for the same tick.  The job must end
    FAILED (no resurrection to DONE, no complete-after-fail event, no
    double release)."""
    from gpuschedule_amd import Cluster, JobSpec, Scheduler, make_policy
    sched = Scheduler(Cluster(2, 2), make_policy("fifo"))
    sched.submit([JobSpec(job_id="g", submit_time=0.0, n_gpus=2,
                          model="llama3-tiny", total_iters=4)])
    sched.tick(0.0)
    assert sched.jobs["g"].state.name == "RUNNING"
    sched.on_fail("g", "RuntimeError: boom", 1.0)
    # leader's ok report arrives after — enough iters to "complete"
    fin = sched.on_progress("g", 4, 2.0, 1.0)
    assert fin is False
    assert sched.jobs["g"].state.name == "FAILED"
    kinds = [e["event"] for e in sched.log.events if e.get("job") == "g"]
    assert "complete" not in kinds and kinds.count("fail") == 1
    # cluster fully freed exactly once: both GPUs placeable again
    sched.submit([JobSpec(job_id="h", submit_time=1.0, n_gpus=2,
                          model="llama3-tiny", total_iters=1)])
    sched.tick(2.0)
    assert sched.jobs["h"].state.name == "RUNNING"
