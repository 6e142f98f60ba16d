"""Llama-3-architecture decoder (random init) on the gfx950 op set.

Per layer: FusedRMSNorm (HIP), fused-QKV MFMALinear (no bias), RoPE
from a host-precomputed cos/sin table (guide: no on-device trig in the
hot path), GQA flash attention, SwiGLU MLP on MFMALinear.  The 8B
config matches Llama-3-8B dimensions; `tiny` is the test-scale variant.
Sized for 288 GB HBM: the full 8B model + fp32 AdamW state + master
weights (~112 GB) trains on ONE MI355X without sharding.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from ..ops.attention import flash_attention
from ..ops.ce import fused_cross_entropy
from ..ops.elementwise import fused_rope, swiglu_packed
from ..ops.embedding import FusedEmbedding
from ..ops.linear import MFMALinear
from ..ops.norms import FusedRMSNorm


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden: int = 4096
    layers: int = 32
    heads: int = 32
    kv_heads: int = 8
    ffn: int = 14336
    max_seq: int = 8192
    rope_theta: float = 500000.0

    @property
    def head_dim(self):
        return self.hidden // self.heads


LLAMA3_8B = LlamaConfig()
LLAMA3_TINY = LlamaConfig(vocab_size=1024, hidden=512, layers=4, heads=8,
                          kv_heads=4, ffn=1408, max_seq=512,
                          rope_theta=10000.0)


def rope_table(cfg: LlamaConfig, device, dtype=None):
    # fp32 tables (the fused gfx950 RoPE kernel reads them directly)
    d = cfg.head_dim
    inv = 1.0 / (cfg.rope_theta **
                 (torch.arange(0, d, 2, dtype=torch.float32) / d))
    t = torch.arange(cfg.max_seq, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return freqs.cos().to(device=device), freqs.sin().to(device=device)


class LlamaLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        h, hd = cfg.hidden, cfg.head_dim
        self.qkv = MFMALinear(h, (cfg.heads + 2 * cfg.kv_heads) * hd,
                              bias=False)
        self.o = MFMALinear(h, h, bias=False)
        self.norm1 = FusedRMSNorm(h)
        self.gate_up = MFMALinear(h, 2 * cfg.ffn, bias=False)
        self.down = MFMALinear(cfg.ffn, h, bias=False)
        self.norm2 = FusedRMSNorm(h)

    def forward(self, x, cos, sin):
        cfg = self.cfg
        B, S, _ = x.shape
        hd = cfg.head_dim
        n = self.norm1(x)
        qkv = self.qkv(n)
        q, k, v = qkv.split([cfg.heads * hd, cfg.kv_heads * hd,
                             cfg.kv_heads * hd], dim=-1)
        q = q.view(B, S, cfg.heads, hd).transpose(1, 2)
        k = k.view(B, S, cfg.kv_heads, hd).transpose(1, 2)
        v = v.view(B, S, cfg.kv_heads, hd).transpose(1, 2)
        q = fused_rope(q, cos, sin)
        k = fused_rope(k, cos, sin)
        att = flash_attention(q, k, v, causal=True)
        att = att.transpose(1, 2).reshape(B, S, cfg.hidden)
        # fused residual-add + RMSNorm; summed stream comes back out
        n, x = self.norm2(self.o(att), residual=x, return_sum=True)
        # residual add fused into the down-projection epilogue
        x = self.down(swiglu_packed(self.gate_up(n)), add=x)
        return x


class LlamaLM(nn.Module):
    def __init__(self, cfg: LlamaConfig = LLAMA3_TINY):
        super().__init__()
        self.cfg = cfg
        self.tok = FusedEmbedding(cfg.vocab_size, cfg.hidden)
        nn.init.normal_(self.tok.weight, std=0.02)
        self.layers = nn.ModuleList(LlamaLayer(cfg)
                                    for _ in range(cfg.layers))
        self.norm = FusedRMSNorm(cfg.hidden)
        self.head = MFMALinear(cfg.hidden, cfg.vocab_size, bias=False)
        self.register_buffer("rope_cos", None, persistent=False)
        self.register_buffer("rope_sin", None, persistent=False)

    def _rope(self, device, dtype=None):
        if self.rope_cos is None or self.rope_cos.device != device:
            cos, sin = rope_table(self.cfg, device)
            self.rope_cos, self.rope_sin = cos, sin
        return self.rope_cos, self.rope_sin

    def forward(self, tokens):
        x = self.tok(tokens)
        cos, sin = self._rope(tokens.device, x.dtype)
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.head(self.norm(x))

    def loss(self, tokens, labels):
        logits = self(tokens)
        return fused_cross_entropy(logits, labels)
