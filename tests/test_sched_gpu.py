"""End-to-end scheduler scenarios on one real MI355X."""
import pytest
import torch

from gpuschedule_amd import Cluster, JobSpec, Scheduler, make_policy
from gpuschedule_amd.exec import CheckpointStore, JobRunner, TorchBackend

pytestmark = pytest.mark.gpu


def J(jid, t=0.0, iters=4, model="bert-base", batch=8, **kw):
    return JobSpec(job_id=jid, submit_time=t, n_gpus=1, total_iters=iters,
                   model=model, batch_size=batch, **kw)


def run_jobs(specs, policy, tmp_path, policy_kw=None, ipt=2):
    sched = Scheduler(Cluster(1, 1), make_policy(policy,
                                                 **(policy_kw or {})))
    sched.submit(specs)
    be = TorchBackend(sched, rank=0, world=1, device_type="cuda",
                      iters_per_tick=ipt, ckpt_root=str(tmp_path / "ck"))
    log = be.run(max_seconds=600)
    return log


def test_gandiva_timeslice_two_jobs_gpu(tmp_path):
    # Two jobs share one GPU via in-HBM suspend/resume.  Quantum much
    # smaller than total job runtime so rotation must happen.
    specs = [J("a", iters=30), J("b", iters=30, model="resnet18",
               batch=32)]
    log = run_jobs(specs, "gandiva", tmp_path,
                   {"quantum": 0.05}, ipt=1)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    assert s["preemptions"] >= 1  # actually time-sliced


def test_checkpoint_restore_gpu(tmp_path):
    ck = CheckpointStore(str(tmp_path / "ck"))
    r1 = JobRunner("cj", "llama3-tiny", 4, "cuda:0", torch.bfloat16,
                   None, 0, 1)
    r1.step(2)
    ck.save("cj", r1.model, r1.optim, r1.done_iters)
    r1.step(2)
    state = ck.load("cj")
    r2 = JobRunner("cj", "llama3-tiny", 4, "cuda:0", torch.bfloat16,
                   None, 0, 1, state=state)
    r2.step(2)
    for (n1, p1), (_, p2) in zip(r1.model.named_parameters(),
                                 r2.model.named_parameters()):
        # embedding backward uses atomics => order-nondeterministic
        # accumulation; allow bf16-scale drift, not bitwise equality
        assert torch.allclose(p1.float(), p2.float(), atol=2e-3), \
            f"GPU restore diverged at {n1}"


def test_srtf_preempt_resume_gpu(tmp_path):
    # Deterministic by construction: the short job is injected only
    # AFTER the long one has measurably progressed (a wall-clock
    # arrival raced model-build time and occasionally admitted both
    # in the same tick, where SRTF simply runs short first with no
    # preemption to observe).
    sched = Scheduler(Cluster(1, 1), make_policy("srtf"))
    sched.submit([J("long", iters=40, iter_time_hint=10.0)])
    be = TorchBackend(sched, rank=0, world=1, device_type="cuda",
                      iters_per_tick=2, ckpt_root=str(tmp_path / "ck"))
    for _ in range(50):
        be.tick_once()
        if sched.jobs["long"].done_iters >= 2:
            break
    assert sched.jobs["long"].done_iters >= 2
    sched.submit_now(J("short", iters=2, iter_time_hint=0.001,
                       model="resnet18", batch=32), be.now())
    log = be.run(max_seconds=600)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    order = [e["job"] for e in log.events if e["event"] == "complete"]
    assert order[0] == "short"
    assert any(e["event"] == "preempt" and e["job"] == "long"
               for e in log.events)


def test_eviction_restore_cycle_gpu(tmp_path):
    # Artificially small per-GPU HBM so two jobs cannot stay resident:
    # suspending one forces checkpoint-eviction, resuming restores it.
    from gpuschedule_amd.trace.job import Job
    hbm = 6 * 1024**3  # 6 GB "GPU" (resnet18 est. 4 GB: one fits)
    # quantum far below per-job runtime so rotation (and therefore
    # eviction) must happen even with warm kernels
    sched = Scheduler(Cluster(1, 1), make_policy("gandiva",
                                                 quantum=0.01))
    # catalog says resnet18 needs 4GB: two cannot co-reside on 6GB
    specs = [J("e1", iters=10, model="resnet18", batch=32),
             J("e2", iters=10, model="resnet18", batch=32)]
    for g in sched.cluster.gpus:
        g.hbm_bytes = hbm
    sched.submit(specs)
    be = TorchBackend(sched, rank=0, world=1, device_type="cuda",
                      iters_per_tick=1,
                      ckpt_root=str(tmp_path / "ck"))
    # pin the catalog estimates: the measured-footprint refinement
    # would otherwise discover both jobs fit (the point here is the
    # evict/restore machinery, not the estimator)
    sched.cluster.update_mem = lambda *a, **k: None
    log = be.run(max_seconds=600)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    assert any(e["event"] == "evict" for e in log.events), \
        "memory pressure should have evicted a suspended job"
    # restored jobs kept making progress after eviction
    states = {j: (sched.jobs[j].state.name, sched.jobs[j].done_iters)
              for j in ("e1", "e2")}
    assert sched.jobs["e1"].done_iters >= 10, (states, log.events)
    assert sched.jobs["e2"].done_iters >= 10, (states, log.events)


def test_gandiva_packing_on_gpu(tmp_path):
    # Two jobs co-running on one MI355X via per-job HIP streams.
    sched = Scheduler(Cluster(1, 1),
                      make_policy("gandiva", quantum=60.0, packing=True))
    specs = [J("p1", iters=10, model="resnet18", batch=64),
             J("p2", iters=10, model="bert-base", batch=8)]
    sched.submit(specs)
    be = TorchBackend(sched, rank=0, world=1, device_type="cuda",
                      iters_per_tick=2, ckpt_root=str(tmp_path / "ck"))
    log = be.run(max_seconds=600)
    s = log.summary()
    assert int(s["n_complete"]) == 2
    assert s["preemptions"] == 0  # co-ran, never time-sliced
    # both were running simultaneously at some point
    running, overlap = set(), False
    for e in log.events:
        if e["event"] == "start":
            running.add(e["job"])
            overlap |= len(running & {"p1", "p2"}) == 2
        elif e["event"] in ("complete", "preempt", "fail"):
            running.discard(e.get("job"))
    assert overlap


def test_sim_predicts_real_backend(tmp_path):
    """Simulator fidelity: the calibrated warm-pool simulator must
    predict the real backend's avg JCT and makespan within 2x on a
    small SRTF trace (measured agreement is ~1.1x on the 10-job trace,
    profiles/sim_fidelity.json; the loose band absorbs box variance)."""
    from gpuschedule_amd.sim.simulator import simulate

    def trace():
        return [JobSpec(job_id=f"f{i}", submit_time=0.3 * i, n_gpus=1,
                        model=("resnet18" if i % 2 == 0 else "bert-base"),
                        total_iters=(120 if i % 2 == 0 else 48))
                for i in range(6)]

    _, sim = simulate(trace(), make_policy("srtf"), n_gpus=1,
                      reeval_s=0.05, warm_pool=True)
    sched = Scheduler(Cluster(1, 1), make_policy("srtf"))
    sched.submit(trace())
    be = TorchBackend(sched, rank=0, world=1, device_type="cuda",
                      iters_per_tick=10,
                      ckpt_root=str(tmp_path / "ck"))
    be.prewarm(["resnet18", "bert-base"], count=1)
    torch.cuda.synchronize()
    real = be.run(max_seconds=300).summary()
    assert real["n_complete"] == 6
    for k in ("avg_jct", "makespan"):
        ratio = real[k] / sim[k]
        assert 0.5 < ratio < 2.0, f"{k}: real {real[k]:.2f} vs " \
                                  f"sim {sim[k]:.2f} (ratio {ratio:.2f})"
