"""Llama-3-8B single-GPU training step — the 288 GB HBM3E showcase.

Full 8B model + bf16 grads + fp32 AdamW m/v + fp32 master weights
(~128 GB of state) trains on ONE MI355X with room to spare; no
sharding, no offload.  This is what the elastic policy (config 5)
schedules.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_llama8b_one_step_fits_and_trains():
    from gpuschedule_amd.exec import JobRunner
    torch.cuda.reset_peak_memory_stats()
    r = JobRunner("l8b", "llama3-8b", 1, "cuda:0", torch.bfloat16,
                  None, 0, 1)
    r.step(2)
    assert torch.isfinite(torch.tensor(r.last_loss))
    peak = torch.cuda.max_memory_allocated() / 1024**3
    print(f"llama3-8b peak HBM: {peak:.1f} GiB")
    assert peak < 280, f"over HBM budget: {peak:.1f} GiB"
    # catalog placement estimate must be an upper bound-ish (within 2x)
    from gpuschedule_amd import catalog
    est = catalog.model_meta("llama3-8b").mem_bytes / 1024**3
    assert peak < est * 2
