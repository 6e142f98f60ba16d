"""Execution-backend tests on CPU: single process and gloo world-2."""
import os
import sys

import pytest
import torch

from gpuschedule_amd import Cluster, JobSpec, Scheduler, make_policy
from gpuschedule_amd.exec import CheckpointStore, JobRunner, TorchBackend


def J(jid, t=0.0, n=1, iters=4, model="llama3-tiny", batch=2, **kw):
    return JobSpec(job_id=jid, submit_time=t, n_gpus=n, total_iters=iters,
                   model=model, batch_size=batch, **kw)


def make_backend(specs, policy_name="fifo", rank=0, world=1,
                 policy_kw=None, **kw):
    sched = None
    if rank == 0:
        sched = Scheduler(Cluster(world, world),
                          make_policy(policy_name, **(policy_kw or {})))
        sched.submit(specs)
    return TorchBackend(sched, rank=rank, world=world, device_type="cpu",
                        iters_per_tick=2, **kw)


# ---- single process -----------------------------------------------

def test_single_rank_runs_trace(tmp_path):
    specs = [J("a", iters=4), J("b", t=0.0, iters=3, model="resnet18",
               batch=4)]
    be = make_backend(specs, ckpt_root=str(tmp_path / "ck"))
    log = be.run(max_seconds=300)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    assert be.sched.jobs["a"].done_iters == 4


def test_preemption_resume_single_rank(tmp_path):
    # SRTF: short job preempts the long one mid-flight; both finish.
    specs = [J("long", iters=12, iter_time_hint=1.0),
             J("short", t=0.5, iters=2, iter_time_hint=0.001)]
    be = make_backend(specs, "srtf", ckpt_root=str(tmp_path / "ck"))
    log = be.run(max_seconds=300)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    assert s["preemptions"] >= 1


def test_unknown_model_fails_at_admission(tmp_path):
    specs = [J("bad", iters=3, model="nosuchmodel"), J("ok", iters=2)]
    be = make_backend(specs, ckpt_root=str(tmp_path / "ck"))
    log = be.run(max_seconds=300)
    events = {e["job"]: e["event"] for e in log.events
              if e["event"] in ("complete", "fail")}
    assert events["bad"] == "fail" and events["ok"] == "complete"


def test_runtime_failure_isolated(tmp_path):
    # Inject a model family that crashes mid-training; the other job
    # must still complete and the cluster must stay healthy.
    from gpuschedule_amd.models import registry
    from gpuschedule_amd import catalog as cat

    class Exploding(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = torch.nn.Linear(8, 8)
            self.n = 0

        def forward(self, x):
            self.n += 1
            if self.n >= 2:
                raise RuntimeError("boom")
            return self.lin(x)

    registry._REGISTRY["crash"] = (
        Exploding, registry._img_batch((8,), 8))
    cat.CATALOG["crash"] = cat.ModelMeta("crash", 4, 1 << 30, 0.01, 0.1)
    try:
        specs = [J("bad", iters=5, model="crash", batch=4),
                 J("ok", iters=2)]
        be = make_backend(specs, "gandiva", ckpt_root=str(tmp_path / "ck"))
        log = be.run(max_seconds=300)
        events = {e["job"]: e["event"] for e in log.events
                  if e["event"] in ("complete", "fail")}
        assert events["bad"] == "fail" and events["ok"] == "complete"
    finally:
        registry._REGISTRY.pop("crash")
        cat.CATALOG.pop("crash")


def test_runner_checkpoint_determinism(tmp_path):
    ck = CheckpointStore(str(tmp_path / "ck"))
    r1 = JobRunner("j", "llama3-tiny", 2, "cpu", torch.float32, None, 0, 1)
    r1.step(3)
    ck.save("j", r1.model, r1.optim, r1.done_iters)
    r1.step(2)

    state = ck.load("j")
    r2 = JobRunner("j", "llama3-tiny", 2, "cpu", torch.float32, None, 0, 1,
                   state=state)
    assert r2.done_iters == 3
    r2.step(2)
    for (n1, p1), (n2, p2) in zip(r1.model.named_parameters(),
                                  r2.model.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), \
            f"restore diverged at {n1}"


def test_runner_pool_adopt_matches_fresh_build():
    """An adopted pooled runner must be bit-identical to a fresh build
    (same family-seeded init, same optimizer trajectory)."""
    from gpuschedule_amd.exec.trainer import RUNNER_POOL

    r1 = JobRunner("first", "llama3-tiny", 2, "cpu", torch.float32,
                   None, 0, 1, pool_key=("llama3-tiny", 2, (0,)))
    r1.step(3)  # diverge from init
    assert RUNNER_POOL.put(r1)
    pooled = RUNNER_POOL.get(("llama3-tiny", 2, (0,)))
    assert pooled is r1
    pooled.adopt("second")
    assert pooled.done_iters == 0

    fresh = JobRunner("second", "llama3-tiny", 2, "cpu", torch.float32,
                      None, 0, 1)
    pooled.step(3)
    fresh.step(3)
    assert pooled.last_loss == pytest.approx(fresh.last_loss, rel=1e-6)
    for (n1, p1), (n2, p2) in zip(pooled.model.named_parameters(),
                                  fresh.model.named_parameters()):
        assert torch.equal(p1, p2), f"adopt diverged from fresh at {n1}"


def test_backend_reuses_pooled_runner(tmp_path):
    """Two same-family jobs in sequence: the second must adopt the
    first's runner instead of building cold."""
    specs = [J("a", iters=2), J("b", t=0.1, iters=2)]
    be = make_backend(specs, ckpt_root=str(tmp_path / "ck"))
    ran = {}
    orig_phase_b = be._apply_phase_b

    def spy(plan, meta):
        orig_phase_b(plan, meta)
        for p in plan.places:
            if p.job_id in be.runners:
                ran[p.job_id] = id(be.runners[p.job_id])
    be._apply_phase_b = spy
    log = be.run(max_seconds=300)
    assert int(log.summary()["n_complete"]) == 2
    assert ran["a"] == ran["b"], "second job should adopt pooled runner"


def test_suspended_job_migration_preserves_state(tmp_path):
    """A SUSPENDED (HBM-resident, never-evicted) job that the scheduler
    migrates appears only as a Placement with migrate_from set (no
    preempt entry).  Phase A must persist its state before the stale
    runner is dropped, or training silently restarts from initial
    weights on the destination rank."""
    from gpuschedule_amd.sched.scheduler import Plan, Placement

    specs = [J("j", iters=100)]
    be = make_backend(specs, ckpt_root=str(tmp_path / "ck"))
    # Place + step the job so its state diverges from the seeded init.
    be.tick_once()
    assert "j" in be.runners
    r = be.runners["j"]
    stepped = {k: v.clone() for k, v in r.model.state_dict().items()}
    done = r.done_iters
    assert done > 0

    # Scheduler suspends it (reason "policy": stays HBM-resident) ...
    be._apply_phase_a(Plan(preempts=[("j", "policy")]), {})
    assert "j" in be.runners and not be.ckpt.exists("j")

    # ... then migrates it to another GPU.  This rank (0) is NOT in the
    # destination; the only state copy is the resident runner.
    plan = Plan(places=[Placement(job_id="j", gpus=[1], resume=True,
                                  migrate_from=[0])])
    meta = {"j": ("llama3-tiny", 2, [1], done)}
    be._apply_phase_a(plan, meta)
    assert "j" not in be.runners, "stale runner must be dropped"
    assert be.ckpt.exists("j"), "state must be checkpointed in phase A"
    state = be.ckpt.load("j")
    assert state["done_iters"] == done
    for k, v in stepped.items():
        assert torch.allclose(state["model"][k].float(), v.float(),
                              atol=1e-6), f"state lost at {k}"


def test_packing_never_overcommits_memory():
    """Two llama3-8b jobs (150 GB each) must not be packed onto one
    288 GB GPU: the shadow memory map has to see the first job's
    footprint when the second is considered in the same tick."""
    from gpuschedule_amd.sched.policies.gandiva import GandivaPolicy

    sched = Scheduler(Cluster(1, 1),
                      GandivaPolicy(quantum=60.0, packing=True,
                                    pack_capacity=2))
    sched.submit([J("a", model="llama3-8b", iters=10, batch=2),
                  J("b", model="llama3-8b", iters=10, batch=2)])
    plan = sched.tick(0.0)  # must not raise (Cluster.place assert)
    placed = {p.job_id for p in plan.places}
    assert placed == {"a"}, placed
    running = [j.job_id for j in sched.jobs.values()
               if j.state.name == "RUNNING"]
    assert running == ["a"]


# ---- gloo world-2 ---------------------------------------------------

def _worker(rank, world, port, scenario, out_file):
    import torch.distributed as dist
    torch.set_num_threads(max(1, (os.cpu_count() or 4) // world))
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    pol_kw = None
    try:
        if scenario == "gang":
            specs = [J("gang", n=2, iters=4)]
            pol = "fifo"
        elif scenario == "mix":
            specs = [J("g", n=2, iters=3), J("s1", t=0.2, iters=2),
                     J("s2", t=0.2, iters=2)]
            pol = "srtf"
        elif scenario == "las4":
            # LAS gangs on 4 ranks: a 4-gang, a 2-gang and singles
            # contend; LAS preempts long-served jobs for arrivals
            specs = [J("g4", n=4, iters=6), J("g2", t=0.1, n=2, iters=4),
                     J("s1", t=0.1, iters=3), J("s2", t=0.2, iters=3)]
            pol = "las"
            pol_kw = {"thresholds": (0.05, 0.4), "starve_limit": 2.0}
        elif scenario == "gandiva4":
            # quantum rotation + migration across 4 ranks; the 2-gang
            # forces defrag when singles fragment the window
            specs = ([J(f"s{i}", t=0.05 * i, iters=4) for i in range(5)]
                     + [J("g2", t=0.3, n=2, iters=4)])
            pol = "gandiva"
            pol_kw = {"quantum": 0.2}
        elif scenario == "elastic4":
            specs = [JobSpec(job_id="el", submit_time=0.0, n_gpus=1,
                             total_iters=10, model="llama3-tiny",
                             batch_size=2, elastic=True, min_gpus=1,
                             max_gpus=4)]
            pol = "elastic"
        elif scenario == "elastic8":
            # BASELINE config-5 shape: one elastic job growing 1->8
            specs = [JobSpec(job_id="el", submit_time=0.0, n_gpus=1,
                             total_iters=12, model="llama3-tiny",
                             batch_size=2, elastic=True, min_gpus=1,
                             max_gpus=8)]
            pol = "elastic"
        else:  # elastic grow
            specs = [JobSpec(job_id="el", submit_time=0.0, n_gpus=1,
                             total_iters=6, model="llama3-tiny",
                             batch_size=2, elastic=True, min_gpus=1,
                             max_gpus=2)]
            pol = "elastic"
        ck = f"/tmp/gpuschedule_test_ck_{port}"
        be = make_backend(specs, pol, rank=rank, world=world,
                          policy_kw=pol_kw, ckpt_root=ck)
        saves = []
        orig_save = be.ckpt.save
        be.ckpt.save = lambda jid, *a, **kw: (saves.append(jid),
                                              orig_save(jid, *a, **kw))[1]
        log = be.run(max_seconds=300)
        if rank == 0:
            import json
            with open(out_file, "w") as f:
                json.dump({"summary": log.summary(),
                           "events": log.events, "saves": saves}, f)
    finally:
        dist.destroy_process_group()


def _bcast_worker(rank, world, port, out_file):
    import torch.distributed as dist
    from gpuschedule_amd.parallel.ddp import broadcast_state
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        r = JobRunner("el", "llama3-tiny", 2, "cpu", torch.float32,
                      None, 0, 1)
        if rank == 0:  # the pre-grow survivor trained solo
            r.step(3)
        group = dist.group.WORLD
        broadcast_state(r.model, r.optim, group, src_rank=0)
        if rank == 1:
            # joiner must now hold the survivor's exact state
            sums = sorted((n, float(p.double().sum()))
                          for n, p in r.model.named_parameters())
            st = [float(v.double().sum())
                  for g in r.optim.param_groups for p in g["params"]
                  for k, v in sorted(r.optim.state[p].items())
                  if isinstance(v, torch.Tensor)]
            steps = [r.optim.state[p].get("step")
                     for g in r.optim.param_groups for p in g["params"]]
            import json
            with open(out_file + ".r1", "w") as f:
                json.dump({"sums": sums, "st": st, "steps": steps}, f)
        else:
            sums = sorted((n, float(p.double().sum()))
                          for n, p in r.model.named_parameters())
            st = [float(v.double().sum())
                  for g in r.optim.param_groups for p in g["params"]
                  for k, v in sorted(r.optim.state[p].items())
                  if isinstance(v, torch.Tensor)]
            steps = [r.optim.state[p].get("step")
                     for g in r.optim.param_groups for p in g["params"]]
            import json
            with open(out_file + ".r0", "w") as f:
                json.dump({"sums": sums, "st": st, "steps": steps}, f)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_broadcast_state_syncs_joiner(tmp_path):
    import torch.multiprocessing as mp
    import json
    out = str(tmp_path / "b")
    mp.spawn(_bcast_worker, args=(2, 29320, out), nprocs=2, join=True)
    with open(out + ".r0") as f:
        a = json.load(f)
    with open(out + ".r1") as f:
        b = json.load(f)
    assert a == b, "joiner state differs from survivor after broadcast"
    assert a["steps"][0] == 3


_SCENARIO_WORLD = {"gang": 2, "mix": 2, "elastic": 2,
                   "las4": 4, "gandiva4": 4, "elastic4": 4,
                   "elastic8": 8}


@pytest.mark.parametrize("scenario", list(_SCENARIO_WORLD))
def test_multiproc_scenarios(tmp_path, scenario):
    import torch.multiprocessing as mp
    import json
    world = _SCENARIO_WORLD[scenario]
    port = 29310 + list(_SCENARIO_WORLD).index(scenario)
    out = str(tmp_path / "out.json")
    mp.spawn(_worker, args=(world, port, scenario, out), nprocs=world,
             join=True)
    with open(out) as f:
        res = json.load(f)
    s = res["summary"]
    assert s["n_complete"] == s["n_jobs"], res["events"]
    events = res["events"]
    if scenario.startswith("elastic"):
        grew = [e for e in events if e["event"] == "resize"
                and e["new_gpus"] > e["old_gpus"]]
        widest = max((len(e.get("gpus", [])) for e in events
                      if e["event"] in ("start", "resume")), default=0)
        assert grew or widest == world, events
        # grow resizes move state by RCCL broadcast from a surviving
        # rank — the /dev/shm checkpoint round-trip must NOT happen
        if grew:
            assert "el" not in res["saves"], res["saves"]
        if scenario == "elastic8":
            assert widest == 8, f"never reached width 8: {events}"
    if scenario == "las4":
        assert s["preemptions"] >= 1, events
    if scenario == "gandiva4":
        assert s["preemptions"] >= 1, events


def test_multinode_gangs_stay_node_local(tmp_path):
    """world 4 modeled as 2 nodes x 2 GPUs: a 2-wide gang must land on
    GPUs {0,1} or {2,3}, never {1,2} (xGMI/RCCL is per node)."""
    from gpuschedule_amd.sched.policies.fifo import FIFOPolicy
    sched = Scheduler(Cluster(4, gpus_per_node=2), FIFOPolicy())
    sched.submit([J("a", n=2, iters=4), J("b", n=2, iters=4),
                  J("c", n=2, iters=4)])
    plan = sched.tick(0.0)
    placed = {p.job_id: p.gpus for p in plan.places}
    for jid, gpus in placed.items():
        assert gpus in ([0, 1], [2, 3]), f"{jid} crossed nodes: {gpus}"
    assert len(placed) == 2  # third job queues: no node window free


def test_checkpoint_restores_rng_state(tmp_path):
    """The documented checkpoint format carries torch RNG state; a
    restore must resume the stochastic stream where the save left it."""
    ck = CheckpointStore(str(tmp_path / "ck"))
    r = JobRunner("j", "llama3-tiny", 2, "cpu", torch.float32, None, 0, 1)
    r.step(2)
    torch.manual_seed(1234)
    torch.rand(7)  # advance the global stream to a known point
    ck.save("j", r.model, r.optim, r.done_iters)
    want = torch.rand(4)  # what the stream yields after the save point

    torch.manual_seed(999)  # scramble
    r2 = JobRunner("j", "llama3-tiny", 2, "cpu", torch.float32, None, 0, 1,
                   state=ck.load("j"))
    got = torch.rand(4)
    assert torch.equal(got, want), "RNG stream not restored"


def test_regang_updates_pool_key(tmp_path):
    """A runner rebound to a resized gang must be pooled under the NEW
    gang's key — adoption under the old key would hand a single-GPU job
    a reducer still bound to the multi-rank group."""
    from gpuschedule_amd.sched.scheduler import Plan, Placement
    specs = [J("el", iters=100)]
    be = make_backend(specs, ckpt_root=str(tmp_path / "ck"))
    be.tick_once()
    r = be.runners["el"]
    assert r.pool_key == ("llama3-tiny", 2, (0,))
    # shrink-style overlap resize back onto [0] exercises the regang
    plan = Plan(places=[Placement(job_id="el", gpus=[0], resume=True,
                                  resize_from=2,
                                  resize_from_gpus=[0, 1])])
    meta = {"el": ("llama3-tiny", 2, [0], r.done_iters)}
    be._apply_phase_a(plan, meta)
    be._apply_phase_b(plan, meta)
    assert be.runners["el"].pool_key == ("llama3-tiny", 2, (0,))
    assert be.runners["el"].gang_world == 1


def _gang_prewarm_worker(rank, world, port, out_file):
    import json
    import torch.distributed as dist
    from gpuschedule_amd.exec.trainer import RUNNER_POOL
    torch.set_num_threads(2)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # batch unset (0) resolves to the model default, matching the
        # prewarm key — exactly how bench.py's fill specs are built
        specs = ([J("gang", n=2, iters=4, batch=0)]
                 if rank == 0 else None)
        be = make_backend(specs, "fifo", rank=rank, world=world,
                          ckpt_root=f"/tmp/gsamd_gpw_{port}")
        # what bench.py does before its warm ticks at N>=2 (the
        # collective prewarm of the 2-GPU DDP gang fill runner)
        be.prewarm_gang("llama3-tiny", [0, 1], iters=2)
        key = ("llama3-tiny", 8, (0, 1))
        pooled = RUNNER_POOL._pool.get(key)
        assert pooled, f"gang runner not pooled on rank {rank}"
        # the gang job scheduled next must ADOPT it
        ran = {}
        orig = be._apply_phase_b

        def spy(plan, meta):
            orig(plan, meta)
            for p in plan.places:
                if p.job_id in be.runners:
                    ran[p.job_id] = id(be.runners[p.job_id])
        be._apply_phase_b = spy
        pooled_id = id(pooled[0])
        log = be.run(max_seconds=120)
        assert ran.get("gang") == pooled_id, "gang job did not adopt"
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump(log.summary(), f)
    finally:
        dist.destroy_process_group()


def test_gang_prewarm_and_adopt(tmp_path):
    """bench.py's N>=2 gang prewarm: collective build+step outside the
    tick protocol, pooled symmetrically, adopted by the scheduled gang
    job on every rank (batch=0 resolves to the model default = 8,
    matching catalog.job_batch for a spec without batch_size)."""
    import torch.multiprocessing as mp
    import json
    out = str(tmp_path / "o.json")
    mp.spawn(_gang_prewarm_worker, args=(2, 29331, out), nprocs=2,
             join=True)
    with open(out) as f:
        s = json.load(f)
    assert s["n_complete"] == 1


def test_runner_pool_budget_lru_eviction():
    """RunnerPool unit semantics: byte-budgeted, LRU-evicting, get()
    returns the most recently put runner for a key and maintains the
    byte ledger; flush releases everything."""
    from gpuschedule_amd.exec.trainer import RunnerPool

    class FakeRunner:
        def __init__(self, key, mem):
            self.pool_key = key
            self.measured_mem = mem
            self.model = object()
            self.poolable = True
            self.released = False

        def release(self):
            self.released = True

    MB = 1 << 20
    pool = RunnerPool(max_bytes=5 * MB)
    a = FakeRunner(("m", 1, (0,)), 2 * MB)
    b = FakeRunner(("m", 1, (0,)), 2 * MB)
    c = FakeRunner(("n", 2, (0,)), 2 * MB)
    assert pool.put(a) and pool.put(b)
    # c does not fit next to a+b: the LRU entry (a) is evicted+released
    assert pool.put(c)
    assert a.released and not b.released and not c.released
    assert pool._bytes == 4 * MB
    # get() pops by key; most-recent-put first
    got = pool.get(("m", 1, (0,)))
    assert got is b and pool._bytes == 2 * MB
    assert pool.get(("m", 1, (0,))) is None   # a was evicted
    # oversized runner is refused outright
    huge = FakeRunner(("h", 1, (0,)), 6 * MB)
    assert not pool.put(huge) and not huge.released
    # non-poolable runner is refused
    bad = FakeRunner(("m", 1, (0,)), MB)
    bad.poolable = False
    assert not pool.put(bad)
    pool.flush()
    assert c.released and pool._bytes == 0 and pool.get(("n", 2, (0,))) is None


def test_build_runner_oom_flushes_pool_and_retries(monkeypatch, tmp_path):
    """Pooled runners hold HBM the scheduler does not account for; an
    OOM during runner construction must flush the pool and retry once
    (exec/backend.py:_build_runner)."""
    import gpuschedule_amd.exec.backend as BK
    from gpuschedule_amd.exec.trainer import RUNNER_POOL

    calls = {"n": 0, "flushed_at": None}
    real_flush = RUNNER_POOL.flush

    def spy_flush():
        calls["flushed_at"] = calls["n"]
        real_flush()

    class FlakyRunner:
        def __init__(self, *a, **kw):
            calls["n"] += 1
            if calls["n"] == 1:
                raise torch.cuda.OutOfMemoryError("fake OOM")
            self.ok = True

    monkeypatch.setattr(BK, "JobRunner", FlakyRunner)
    monkeypatch.setattr(RUNNER_POOL, "flush", spy_flush)
    be = make_backend([J("x", iters=1)])
    r = be._build_runner("x", "llama3-tiny", 2, None, [0], 0, None,
                         ("llama3-tiny", 2, (0,)))
    assert r.ok and calls["n"] == 2
    assert calls["flushed_at"] == 1  # flushed between the two attempts


def test_gang_groups_contiguity_contract():
    """Width-1 needs no group; non-contiguous gangs are a scheduler
    bug and must fail loudly (all pre-created groups are contiguous
    rank ranges)."""
    from gpuschedule_amd.exec.gangs import GangGroups
    gg = GangGroups(4)  # dist not initialized -> no groups created
    assert gg.get([2]) is None
    assert gg.get([]) is None
    with pytest.raises(AssertionError):
        gg.get([0, 2])
