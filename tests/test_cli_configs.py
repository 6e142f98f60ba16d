"""CLI + shipped config coverage (simulator path, CPU)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CONFIGS = [f"configs/{f}" for f in sorted(os.listdir(os.path.join(
    REPO, "configs"))) if f.endswith(".yaml")]


def run_cli(*args):
    out = subprocess.run([sys.executable, "-m", "gpuschedule_amd", *args],
                         cwd=REPO, capture_output=True, text=True,
                         timeout=300)
    assert out.returncode == 0, out.stderr
    return out.stdout


@pytest.mark.parametrize("cfg", CONFIGS)
def test_simulate_all_shipped_configs(cfg):
    res = json.loads(run_cli("simulate", "--config", cfg).strip())
    # every job either completes or (config 4: Philly Killed/Failed
    # rows, replayed via fail_at_iter) fails mid-run by design
    assert res["n_complete"] + res.get("n_failed", 0) == res["n_jobs"] > 0
    assert res["n_complete"] > 0


def test_gen_trace_and_simulate(tmp_path):
    trace = str(tmp_path / "t.csv")
    run_cli("gen-trace", "--jobs", "12", "--mix", "philly-slice",
            "-o", trace)
    out = json.loads(run_cli("simulate", "--policy", "srtf", "--trace",
                             trace, "--gpus", "4").strip())
    assert out["n_complete"] >= 10  # 4-GPU gangs may be rejected on 4


def test_simulate_writes_event_log(tmp_path):
    log = str(tmp_path / "events.jsonl")
    run_cli("simulate", "--config", "configs/c1_fifo_sim.yaml",
            "--log", log)
    lines = [json.loads(x) for x in open(log) if x.strip()]
    assert any(e["event"] == "complete" for e in lines)


def test_bench_cpu_contract(tmp_path):
    """bench.py emits the driver-contract JSON line (CPU debug mode)."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "6", "--warmup", "1",
         "--device", "cpu", "--jobs", "3"],
        cwd=REPO, capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in res
    assert res["steps"] == 6 and res["n_gpus"] == 1
    assert res["higher_is_better"] is False


def test_compare_command_runs():
    out = subprocess.run(
        [sys.executable, "-m", "gpuschedule_amd", "compare",
         "--jobs", "8", "--gpus", "2", "--policies", "fifo,srtf"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    lines = [l for l in out.stdout.splitlines() if l.strip()]
    assert lines[0].startswith("policy")
    assert {l.split()[0] for l in lines[1:]} == {"fifo", "srtf"}


def test_bench_trace_budget_fits_one_gpu():
    """Trace demand must fit ~70% of ONE GPU's tick slots so the same
    trace completes at every N (strong scaling) within K ticks."""
    sys.path.insert(0, REPO)
    from bench import build_trace
    steps, ipt = 40, 2
    arrivals, _ = build_trace(8, steps, 1, ipt)
    total = sum(s.total_iters for _, s in arrivals)
    assert total <= 0.75 * steps * ipt + 8 * ipt  # budget + rounding
    assert all(t <= steps // 4 for t, _ in arrivals)


def test_bench_trace_identical_across_n():
    """Strong scaling: the trace-job set must not depend on N (only
    the best-effort fill jobs do)."""
    sys.path.insert(0, REPO)
    from bench import build_trace
    a1, f1 = build_trace(8, 40, 1, 2)
    a8, f8 = build_trace(8, 40, 8, 2)
    assert [(t, s.job_id, s.total_iters, s.model, s.n_gpus)
            for t, s in a1] == \
           [(t, s.job_id, s.total_iters, s.model, s.n_gpus)
            for t, s in a8]
    assert len(f1) == 1 and all(s.background for s in f1)
    # at N>=2 the first fill is a 2-GPU gang (exercises RCCL all-reduce)
    assert f8[0].n_gpus == 2 and sum(s.n_gpus for s in f8) == 8


def test_simulate_warm_pool_flag():
    # --warm-pool models the backend's prewarmed RunnerPool: cheaper
    # first placements => strictly lower avg JCT on the same trace
    cold = json.loads(run_cli("simulate", "--policy", "fifo",
                              "--jobs", "6", "--gpus", "2").strip())
    warm = json.loads(run_cli("simulate", "--policy", "fifo",
                              "--jobs", "6", "--gpus", "2",
                              "--warm-pool").strip())
    assert warm["avg_jct"] < cold["avg_jct"]
    assert warm["n_complete"] == cold["n_complete"]


def test_summarize_command(tmp_path):
    log = str(tmp_path / "ev.jsonl")
    run_cli("simulate", "--policy", "srtf", "--jobs", "5",
            "--gpus", "2", "--log", log)
    s = json.loads(run_cli("summarize", "--log", log))
    assert s["n_complete"] > 0
    assert s["makespan"] > 0 and s["avg_jct"] > 0
