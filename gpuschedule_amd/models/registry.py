"""Model registry: build models + synthetic batches per family key.

The scheduler's catalog (gpuschedule_amd/catalog.py) holds the
torch-free metadata; this module holds the torch constructors used by
the execution backend.  All training data is synthetic (seeded random)
— this environment has no datasets and the benchmark contract requires
random-init weights + synthetic data.
"""
from __future__ import annotations

import torch

from .. import catalog
from .bert import BertConfig, BertMLM
from .llama import LLAMA3_8B, LLAMA3_TINY, LlamaLM
from .resnet import resnet18_cifar, resnet50

# family -> (constructor, batch shape maker)
#   make_batch(batch, device, generator) -> (inputs, labels)


def _img_batch(shape, classes):
    def make(batch, device, gen):
        if torch.device(device).type == "cpu":
            x = torch.randn(batch, *shape, device=device, generator=gen,
                            dtype=torch.float32)
        else:
            # generate DIRECTLY in the compute dtype and NHWC layout:
            # the trainer's .to(bf16)/.contiguous(channels_last) then
            # no-op instead of two full-batch copy kernels per iter
            x = torch.empty(batch, *shape, device=device,
                            dtype=torch.bfloat16,
                            memory_format=torch.channels_last)
            x.normal_(generator=gen)
        y = torch.randint(0, classes, (batch,), device=device,
                          generator=gen)
        return x, y
    return make


def _tok_batch(seq, vocab):
    def make(batch, device, gen):
        t = torch.randint(0, vocab, (batch, seq), device=device,
                          generator=gen)
        y = torch.randint(0, vocab, (batch, seq), device=device,
                          generator=gen)
        return t, y
    return make


_REGISTRY = {
    "resnet18": (lambda: resnet18_cifar(10), _img_batch((3, 32, 32), 10)),
    "resnet50": (lambda: resnet50(1000), _img_batch((3, 224, 224), 1000)),
    "bert-base": (lambda: BertMLM(BertConfig()), _tok_batch(128, 30528)),
    "llama3-8b": (lambda: LlamaLM(LLAMA3_8B), _tok_batch(2048, 128256)),
    "llama3-tiny": (lambda: LlamaLM(LLAMA3_TINY), _tok_batch(256, 1024)),
}

# Tokens processed per sample per iteration (for tokens/s metrics);
# image models count images.  Defined torch-free in catalog.py.
SAMPLE_TOKENS = catalog.SAMPLE_TOKENS


def build_model(name: str, device="cpu",
                dtype: torch.dtype = torch.bfloat16, seed: int = 0):
    """Returns (model, make_batch, per_gpu_batch).

    Init happens DIRECTLY on the target device (seconds vs tens of
    seconds for big models on CPU + transfer); gang replicas stay
    identical because every rank seeds the same and MI355X GPUs
    generate identical sequences for identical seeds.
    """
    ctor, make_batch = _REGISTRY[name]
    torch.manual_seed(seed)
    if str(device) != "cpu":
        with torch.device(device):
            model = ctor()
        model = model.to(dtype=dtype)
        # BN statistics/affine params stay fp32 (standard bf16-training
        # practice; the fused BN kernels expect fp32 gamma/beta)
        from ..ops.bn import FusedBN2d
        for m in model.modules():
            if isinstance(m, FusedBN2d):
                m.float()
        if name.startswith("resnet"):
            # NHWC: MIOpen's native layout — avoids the per-conv
            # batched_transpose kernels that dominate NCHW profiles
            model = model.to(memory_format=torch.channels_last)
    else:
        model = ctor()
    return model, make_batch, catalog.model_meta(name).default_batch


def compute_loss(name: str, model, inputs, labels) -> torch.Tensor:
    if hasattr(model, "loss"):
        return model.loss(inputs, labels)
    out = model(inputs.to(next(model.parameters()).dtype))
    return torch.nn.functional.cross_entropy(out.float(), labels)
