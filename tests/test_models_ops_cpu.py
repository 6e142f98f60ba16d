"""CPU-path tests for models and op wrappers (reference math)."""
import pytest
import torch
import torch.nn.functional as F

from gpuschedule_amd.models import build_model, compute_loss
from gpuschedule_amd.ops import (FusedAdamW, FusedLayerNorm, FusedRMSNorm,
                                 FusedSGD, MFMALinear)
from gpuschedule_amd.ops.attention import _ref_attention


@pytest.mark.parametrize("name,batch", [("resnet18", 4), ("bert-base", 2),
                                        ("llama3-tiny", 2)])
def test_model_fwd_bwd(name, batch):
    m, mk, _ = build_model(name, device="cpu", dtype=torch.float32)
    g = torch.Generator().manual_seed(0)
    x, y = mk(batch, "cpu", g)
    loss = compute_loss(name, m, x, y)
    assert torch.isfinite(loss)
    loss.backward()
    for p in m.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()


def test_llama8b_config_shapes():
    from gpuschedule_amd.models import LLAMA3_8B
    c = LLAMA3_8B
    assert (c.hidden, c.layers, c.heads, c.kv_heads, c.ffn) == \
        (4096, 32, 32, 8, 14336)
    assert c.head_dim == 128


def test_layernorm_cpu_matches_torch():
    ln = FusedLayerNorm(64)
    x = torch.randn(8, 64)
    ref = F.layer_norm(x, (64,), ln.weight, ln.bias, ln.eps)
    assert torch.allclose(ln(x), ref, atol=1e-5)


def test_rmsnorm_cpu():
    rn = FusedRMSNorm(64)
    with torch.no_grad():
        rn.weight.mul_(2.0)
    x = torch.randn(8, 64)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + rn.eps) * 2.0
    assert torch.allclose(rn(x), ref, atol=1e-5)


def test_mfma_linear_cpu_matches_linear():
    lin = MFMALinear(32, 16)
    x = torch.randn(4, 32)
    ref = F.linear(x, lin.weight, lin.bias)
    assert torch.allclose(lin(x), ref, atol=1e-6)


def test_ref_attention_matches_sdpa():
    q, k, v = (torch.randn(2, 4, 16, 32) for _ in range(3))
    out = _ref_attention(q, k, v, causal=True)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    assert torch.allclose(out, ref, atol=1e-5)


@pytest.mark.parametrize("opt_cls,kw", [(FusedAdamW, dict(lr=1e-2)),
                                        (FusedSGD, dict(lr=1e-2))])
def test_fused_optimizers_cpu_reduce_loss(opt_cls, kw):
    torch.manual_seed(0)
    m = torch.nn.Linear(16, 1)
    opt = opt_cls(m.parameters(), **kw)
    x = torch.randn(64, 16)
    y = x.sum(dim=1, keepdim=True)
    first = None
    for _ in range(50):
        opt.zero_grad()
        loss = F.mse_loss(m(x), y)
        loss.backward()
        opt.step()
        first = first if first is not None else loss.item()
    assert loss.item() < 0.5 * first


def test_adamw_cpu_matches_torch_adamw():
    torch.manual_seed(1)
    a = torch.nn.Linear(8, 8)
    b = torch.nn.Linear(8, 8)
    b.load_state_dict(a.state_dict())
    oa = FusedAdamW(a.parameters(), lr=1e-2, weight_decay=0.1)
    ob = torch.optim.AdamW(b.parameters(), lr=1e-2, weight_decay=0.1)
    x = torch.randn(32, 8)
    for _ in range(5):
        for m, o in ((a, oa), (b, ob)):
            o.zero_grad()
            m(x).pow(2).mean().backward()
            o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-4), "AdamW semantics drifted"


def test_mfma_linear_fused_addend_cpu_fallback():
    import torch
    from gpuschedule_amd.ops.linear import MFMALinear
    lin = MFMALinear(16, 24, bias=True)
    x = torch.randn(4, 16)
    add = torch.randn(4, 24)
    y = lin(x, add=add)
    ref = torch.nn.functional.linear(x, lin.weight, lin.bias) + add
    assert torch.allclose(y, ref, atol=1e-5)


def test_swiglu_packed_cpu_fallback():
    import torch
    from gpuschedule_amd.ops.elementwise import swiglu_packed
    gu = torch.randn(5, 16)
    g, u = gu.chunk(2, dim=-1)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(swiglu_packed(gu), ref, atol=1e-6)
