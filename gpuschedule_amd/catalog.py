"""Model-family catalog: metadata the scheduler needs WITHOUT torch.

Per family: default per-GPU batch, per-GPU HBM footprint estimate
(weights + optimizer state + activations at the default batch, used by
the bin-packing placement), and a simulator iteration-time model.

The real backend measures true iteration times; the footprints are
deliberately conservative upper bounds (placement safety), refined from
measured `torch.cuda.max_memory_allocated` where available.

Simulated iteration time model:
    iter_time(n) = base * (1 + allreduce_frac * (n - 1) / n) / speedup(n)
with per-GPU work constant (weak scaling in the gang: global batch =
batch * n), so iter time is roughly flat in n but pays a small
all-reduce overhead that grows toward the ring limit.
"""
from __future__ import annotations

from dataclasses import dataclass

GB = 1024**3

# throughput factor when two jobs co-run on one GPU (Gandiva packing):
# each job's iteration takes ~PACK_SLOWDOWN x longer (they share CUs
# and HBM bandwidth, but overlap hides launch gaps)
PACK_SLOWDOWN = 1.7


@dataclass(frozen=True)
class ModelMeta:
    name: str
    default_batch: int          # per-GPU
    mem_bytes: int              # per-GPU footprint estimate
    sim_iter_time: float        # seconds per iteration on 1 GPU (bench shape)
    allreduce_frac: float       # fractional overhead at large gangs


# sim_iter_time is CALIBRATED to measured MI355X hipGraph-replay
# iteration times (round 2; profiles/README.md): resnet18 b128 5.0 ms,
# resnet50 b64 17.2 ms, bert-base b32 12.5 ms, llama3-8b b1 182 ms
# (at the hand-written-GEMM default; GSAMD_GEMM_LIB=1 shaves ~10-13%
# more, profiles/gemm_mode_compare.json) — so the simulator's
# JCT/makespan and the ideal_s slowdown baseline track the real
# backend's absolutes, not just relative orderings.
CATALOG = {
    # CIFAR-shape ResNet-18 (32x32 input, 10 classes)
    "resnet18": ModelMeta("resnet18", 128, 4 * GB, 0.005, 0.10),
    # ImageNet-shape ResNet-50 (224x224, 1000 classes)
    "resnet50": ModelMeta("resnet50", 64, 16 * GB, 0.0172, 0.15),
    # BERT-base encoder, seq 128
    "bert-base": ModelMeta("bert-base", 32, 12 * GB, 0.0125, 0.12),
    # Llama-3-8B-architecture decoder (random init), seq 2048.
    # bf16 weights 16G + fp32 AdamW m/v 64G + fp32 master 32G + acts.
    "llama3-8b": ModelMeta("llama3-8b", 1, 150 * GB, 0.182, 0.20),
    # 4-layer Llama-architecture model for tests/small traces
    "llama3-tiny": ModelMeta("llama3-tiny", 8, 6 * GB, 0.010, 0.12),
}


def model_meta(name: str) -> ModelMeta:
    try:
        return CATALOG[name]
    except KeyError:
        raise ValueError(f"unknown model {name!r}; have {sorted(CATALOG)}")


def job_mem_bytes(spec) -> int:
    return spec.mem_bytes or model_meta(spec.model).mem_bytes


def job_batch(spec) -> int:
    return spec.batch_size or model_meta(spec.model).default_batch


def sim_iter_time(spec, n_gpus: int) -> float:
    m = model_meta(spec.model)
    base = spec.iter_time_hint or m.sim_iter_time
    if n_gpus <= 1:
        return base
    return base * (1.0 + m.allreduce_frac * (n_gpus - 1) / n_gpus)


# Tokens processed per sample per iteration (sequence models) or
# samples (image models count 1).  Torch-free here; models/registry
# re-exports it.
SAMPLE_TOKENS = {"resnet18": 1, "resnet50": 1, "bert-base": 128,
                 "llama3-8b": 2048, "llama3-tiny": 256}


def job_tokens_per_iter(spec, width: int = 1) -> int:
    """Tokens (sequence models) or samples (image models) processed by
    one training iteration of the whole gang: per-GPU batch x tokens
    per sample x DP width.  Drives the per-job throughput metric in
    the event log."""
    return job_batch(spec) * SAMPLE_TOKENS.get(spec.model, 1) * width
