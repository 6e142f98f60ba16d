"""Unit tests for bench.py's trace construction (driver contract)."""
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import bench  # noqa: E402


def test_build_trace_budget_and_shape():
    for steps, n_gpus, n_jobs in [(20, 1, 8), (40, 8, 8), (80, 4, 16)]:
        arrivals, fill = bench.build_trace(n_jobs, steps, n_gpus, 2)
        assert len(arrivals) == n_jobs
        ids = [s.job_id for _, s in arrivals]
        assert len(set(ids)) == n_jobs
        # total tick-slot demand fits 70% of ONE GPU's window
        budget = int(0.70 * steps) * 2
        total = sum(s.total_iters for _, s in arrivals)
        assert total <= budget + n_jobs * 2  # rounding slack
        # arrivals land in the first 25% of the window, sorted
        ticks = [t for t, _ in arrivals]
        assert ticks == sorted(ticks)
        assert max(ticks) <= max(int(0.25 * steps), 1)
        # fill covers every GPU; 2-GPU gang present at N>=2
        covered = sum(s.n_gpus for s in fill)
        assert covered == n_gpus
        if n_gpus >= 2:
            assert any(s.n_gpus == 2 for s in fill)
        assert all(s.background for s in fill)


def test_build_trace_deterministic():
    a1, _ = bench.build_trace(8, 20, 1, 2)
    a2, _ = bench.build_trace(8, 20, 1, 2)
    assert [(t, s.job_id, s.total_iters, s.model) for t, s in a1] == \
           [(t, s.job_id, s.total_iters, s.model) for t, s in a2]


def test_build_trace_same_jobs_at_every_n():
    """Strong scaling: the job set must be identical at every N."""
    a1, _ = bench.build_trace(8, 20, 1, 2)
    a8, _ = bench.build_trace(8, 20, 8, 2)
    assert [(t, s.job_id, s.total_iters) for t, s in a1] == \
           [(t, s.job_id, s.total_iters) for t, s in a8]
