"""GPU numerics: every HIP kernel vs a plain-torch fp32 reference.

All tolerances are scaled for bf16 I/O with fp32 accumulation.
Transpose-detecting inputs (asymmetric random) throughout.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


def ext():
    from gpuschedule_amd.ops import get_ext
    return get_ext(required=True)


def mx(err, ref):
    return (err.float() - ref.float()).abs().max().item()


# ---- norms ---------------------------------------------------------

@pytest.mark.parametrize("shape", [(2048, 768), (512, 4096), (33, 1024),
                                   (4096, 5120)])
def test_layernorm_fwd_bwd(shape):
    torch.manual_seed(0)
    N, D = shape
    x = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)

    y, _, mean, rstd = ext().norm_fwd(x, None, g, b, 1e-5, False)
    xf = x.float().requires_grad_(True)
    gf = g.float().requires_grad_(True)
    bf = b.float().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xf, (D,), gf, bf, 1e-5)
    # bf16 output quantization: 2^-8 relative of the output range
    assert mx(y, ref) / (ref.abs().max().item() + 1e-3) < 8e-3
    ref.backward(dy.float())
    dx, dgm, dbt = ext().norm_bwd(dy, x, mean, rstd, g, False)
    assert mx(dx, xf.grad) / (xf.grad.abs().max().item() + 1e-3) < 1e-2
    gs = gf.grad.abs().max().item() + 1.0
    assert mx(dgm, gf.grad) / gs < 2e-2
    bs = bf.grad.abs().max().item() + 1.0
    assert mx(dbt, bf.grad) / bs < 2e-2


@pytest.mark.parametrize("shape", [(2048, 768), (512, 4096)])
def test_norm_bwd_fused_dadd(shape):
    # norm_bwd(dy, ..., dadd=t) == norm_bwd(dy, ...)[0] + t (the
    # residual-stream grad add fused into the dx write)
    torch.manual_seed(7)
    N, D = shape
    x = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    t = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    _, _, _, rstd = ext().norm_fwd(x, None, g, None, 1e-6, True)
    dx0, _, _ = ext().norm_bwd(dy, x, None, rstd, g, True)
    dxa, _, _ = ext().norm_bwd(dy, x, None, rstd, g, True, t)
    ref = (dx0.float() + t.float())
    assert mx(dxa, ref) / (ref.abs().max().item() + 1e-3) < 8e-3


@pytest.mark.parametrize("shape", [(2048, 768), (512, 4096)])
def test_rmsnorm_fwd_bwd(shape):
    torch.manual_seed(1)
    N, D = shape
    x = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    y, _, _, rstd = ext().norm_fwd(x, None, g, None, 1e-6, True)
    xf = x.float().requires_grad_(True)
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) \
        * g.float()
    assert mx(y, ref) / (ref.abs().max().item() + 1e-3) < 8e-3
    ref.backward(dy.float())
    dx, dgm, _ = ext().norm_bwd(dy, x, None, rstd, g, True)
    assert mx(dx, xf.grad) / (xf.grad.abs().max().item() + 1e-3) < 1e-2


# ---- GEMM ----------------------------------------------------------

@pytest.mark.parametrize("mnk", [(256, 128, 64), (128, 128, 128),
                                 (512, 768, 768), (4096, 3072, 768),
                                 (100, 60, 64), (1000, 130, 192),
                                 (2048, 2048, 4096)])
def test_gemm_nt(mnk):
    torch.manual_seed(2)
    M, N, K = mnk
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    c = ext().gemm_nt(a, b, None, False)
    ref = a.float() @ b.float().t()
    err = (c.float() - ref).abs()
    tol = 0.02 * K ** 0.5  # fp32-accum bf16 roundoff grows ~sqrt(K)
    assert err.max().item() < tol, f"max err {err.max().item()} tol {tol}"


def test_gemm_identity_transpose_detecting():
    # A = I with ASYMMETRIC B: catches row/col-swapped C writes.
    K = 128
    a = torch.eye(K, device="cuda", dtype=torch.bfloat16)
    b = torch.arange(K * K, device="cuda",
                     dtype=torch.float32).reshape(K, K) % 37
    b = b.to(torch.bfloat16)
    c = ext().gemm_nt(a, b, None, False)
    assert torch.equal(c.float(), b.float().t())


def test_gemm_bias_relu():
    torch.manual_seed(3)
    a = torch.randn(256, 256, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(192, 256, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(192, device="cuda", dtype=torch.bfloat16)
    c = ext().gemm_nt(a, b, bias, True)
    ref = torch.relu(a.float() @ b.float().t() + bias.float())
    assert mx(c, ref) < 0.4
    assert (c.float() >= 0).all()


def test_transpose2d():
    torch.manual_seed(4)
    for shape in [(128, 256), (100, 60), (4096, 768), (130, 1000)]:
        x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
        assert torch.equal(ext().transpose2d(x), x.t().contiguous())


def test_mfma_linear_autograd(monkeypatch):
    # pin the hand-written MFMA path regardless of the ambient
    # GSAMD_GEMM_LIB setting — this test covers the kernel
    monkeypatch.setenv("GSAMD_GEMM_LIB", "0")
    from gpuschedule_amd.ops import MFMALinear
    torch.manual_seed(5)
    lin = MFMALinear(768, 512).cuda().bfloat16()
    x = torch.randn(300, 768, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = lin(x)
    loss = y.float().square().mean()
    loss.backward()

    xf = x.detach().float().requires_grad_(True)
    wf = lin.weight.detach().float().requires_grad_(True)
    bf = lin.bias.detach().float().requires_grad_(True)
    reff = torch.nn.functional.linear(xf, wf, bf)
    reff.float().square().mean().backward()
    assert mx(y, reff) < 0.2
    scale = xf.grad.abs().max().item() + 1e-6
    assert mx(x.grad, xf.grad) / scale < 0.05
    scale = wf.grad.abs().max().item() + 1e-6
    assert mx(lin.weight.grad, wf.grad) / scale < 0.05


# ---- optimizers ----------------------------------------------------

def test_fused_adamw_matches_reference():
    torch.manual_seed(6)
    n = 4099
    p32 = torch.randn(n, device="cuda")
    p = p32.bfloat16()
    g = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    pr, mr, vr = p32.clone(), m.clone(), v.clone()
    for t in (1, 2, 3):
        ext().adamw_step(p, p32, g, m, v, 1e-2, 0.9, 0.999, 1e-8, 0.01, t)
        gf = g.float()
        mr.mul_(0.9).add_(gf, alpha=0.1)
        vr.mul_(0.999).addcmul_(gf, gf, value=0.001)
        mh = mr / (1 - 0.9 ** t)
        vh = vr / (1 - 0.999 ** t)
        pr.add_(mh / (vh.sqrt() + 1e-8) + 0.01 * pr, alpha=-1e-2)
    assert mx(p32, pr) < 1e-5
    assert mx(p, pr.bfloat16()) == 0


def test_fused_sgd_matches_reference():
    torch.manual_seed(7)
    n = 1027
    p32 = torch.randn(n, device="cuda")
    p = p32.bfloat16()
    g = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    buf = torch.zeros(n, device="cuda")
    pr, br = p32.clone(), buf.clone()
    for _ in range(3):
        ext().sgd_step(p32, p, g, buf, 0.1, 0.9, 5e-4)
        gf = g.float() + 5e-4 * pr
        br.mul_(0.9).add_(gf)
        pr.add_(br, alpha=-0.1)
    assert mx(p32, pr) < 1e-5


# ---- end-to-end single job on GPU ---------------------------------

def test_resnet18_job_step_on_gpu():
    from gpuschedule_amd.exec import JobRunner
    r = JobRunner("gputest", "resnet18", 32, "cuda:0", torch.bfloat16,
                  None, 0, 1)
    losses = []
    for _ in range(6):
        r.step(1)
        losses.append(r.last_loss)
    assert all(torch.isfinite(torch.tensor(losses)))


# ---- fused BatchNorm ----------------------------------------------

@pytest.mark.parametrize("relu", [False, True])
@pytest.mark.parametrize("shape", [(16, 8, 8, 64), (8, 14, 14, 256),
                                   (4, 7, 7, 2048)])
def test_fused_bn_fwd_bwd(shape, relu):
    torch.manual_seed(8)
    N, H, W, C = shape
    rows = N * H * W
    x = torch.randn(rows, C, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(C, device="cuda").abs() + 0.5
    b = torch.randn(C, device="cuda")
    dy = torch.randn(rows, C, device="cuda", dtype=torch.bfloat16)

    y, mean, rstd = ext().bn_fwd(x, g, b, None, None, None, None,
                                 None, 0.1, 1e-5, relu)
    xf = x.float().requires_grad_(True)
    gf = g.clone().requires_grad_(True)
    bf = b.clone().requires_grad_(True)
    ref = torch.nn.functional.batch_norm(
        xf.t().unsqueeze(0), None, None, gf, bf, True, 0.0, 1e-5)
    ref = ref.squeeze(0).t()
    if relu:
        ref = torch.relu(ref)
    assert mx(y, ref) / (ref.abs().max().item() + 1e-3) < 1e-2
    ref.backward(dy.float())
    dx, dbeta, dgamma = ext().bn_bwd(dy, x, y if relu else None, mean,
                                     rstd, g, relu, False)
    s = xf.grad.abs().max().item() + 1e-4
    assert mx(dx, xf.grad) / s < 2e-2
    s = gf.grad.abs().max().item() + 1e-3
    assert mx(dgamma, gf.grad) / s < 2e-2
    s = bf.grad.abs().max().item() + 1e-3
    assert mx(dbeta, bf.grad) / s < 2e-2
    if relu:
        # mask-recompute path (no y, beta instead) must agree
        dx2, db2, dg2 = ext().bn_bwd(dy, x, None, mean, rstd, g, relu,
                                     False, beta=b)
        assert mx(dx2, dx.float()) / (dx.float().abs().max().item()
                                      + 1e-4) < 1e-3
        assert torch.allclose(db2, dbeta, rtol=1e-2, atol=1e-2)
        assert torch.allclose(dg2, dgamma, rtol=1e-2, atol=1e-2)


def test_fused_bn_module_matches_torch_bn():
    from gpuschedule_amd.ops.bn import FusedBN2d
    torch.manual_seed(9)
    C = 64
    fb = FusedBN2d(C, relu=False).cuda()
    tb = torch.nn.BatchNorm2d(C).cuda()
    x = torch.randn(8, C, 16, 16, device="cuda", dtype=torch.bfloat16)
    y1 = fb(x)
    y2 = tb(x.float())
    assert (y1.float() - y2).abs().max().item() < 3e-2
    assert torch.allclose(fb.running_mean, tb.running_mean, atol=1e-3)
    assert torch.allclose(fb.running_var, tb.running_var, atol=1e-2)
    fb.eval(); tb.eval()
    y1 = fb(x)
    y2 = tb(x.float())
    assert (y1.float() - y2).abs().max().item() < 3e-2


# ---- fused cross-entropy ------------------------------------------

@pytest.mark.parametrize("shape", [(512, 1024), (4096, 30528),
                                   (128, 1000)])
def test_fused_ce(shape):
    torch.manual_seed(10)
    T, V = shape
    logits = torch.randn(T, V, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, V, (T,), device="cuda")
    loss, lse = ext().ce_fwd(logits, labels)
    ref = torch.nn.functional.cross_entropy(
        logits.float().requires_grad_(False), labels)
    assert abs(loss.item() - ref.item()) / ref.item() < 2e-3

    from gpuschedule_amd.ops.ce import fused_cross_entropy
    lg = logits.clone().requires_grad_(True)
    out = fused_cross_entropy(lg, labels)
    out.backward()
    lf = logits.float().requires_grad_(True)
    torch.nn.functional.cross_entropy(lf, labels).backward()
    scale = lf.grad.abs().max().item() + 1e-9
    assert (lg.grad.float() - lf.grad).abs().max().item() / scale < 2e-2


def test_fused_bn_residual_relu():
    from gpuschedule_amd.ops.bn import FusedBN2d
    torch.manual_seed(12)
    C = 128
    fb = FusedBN2d(C, relu=True).cuda()
    x = torch.randn(4, C, 8, 8, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    res = torch.randn_like(x).requires_grad_(True)
    y = fb(x, residual=res)
    y.float().square().mean().backward()

    xf = x.detach().float().requires_grad_(True)
    rf = res.detach().float().requires_grad_(True)
    ref = torch.nn.functional.batch_norm(
        xf, None, None, fb.weight.detach().clone().requires_grad_(False),
        fb.bias.detach().clone(), True, 0.0, fb.eps)
    ref = torch.relu(ref + rf)
    ref.square().mean().backward()
    assert (y.float() - ref).abs().max().item() < 5e-2
    for got, want in ((x.grad, xf.grad), (res.grad, rf.grad)):
        s = want.abs().max().item() + 1e-6
        assert (got.float() - want).abs().max().item() / s < 3e-2


def test_multi_tensor_adamw_matches_torch():
    from gpuschedule_amd.ops import FusedAdamW
    torch.manual_seed(13)
    a = torch.nn.Sequential(torch.nn.Linear(64, 128),
                            torch.nn.Linear(128, 8)).cuda().bfloat16()
    b = torch.nn.Sequential(torch.nn.Linear(64, 128),
                            torch.nn.Linear(128, 8)).cuda()
    with torch.no_grad():
        for pa, pb in zip(a.parameters(), b.parameters()):
            pb.copy_(pa.float())
    oa = FusedAdamW(a.parameters(), lr=1e-2, weight_decay=0.05)
    ob = torch.optim.AdamW(b.parameters(), lr=1e-2, weight_decay=0.05)
    x = torch.randn(32, 64, device="cuda")
    for it in range(6):
        if not oa.fused_zero_grad:
            oa.zero_grad(set_to_none=True)
        a(x.bfloat16()).float().square().mean().backward()
        oa.step()
        ob.zero_grad()
        b(x).square().mean().backward()
        ob.step()
    assert oa._mt is not None, "multi-tensor path did not engage"
    for pa, pb in zip(a.parameters(), b.parameters()):
        err = (pa.float() - pb).abs().max().item()
        assert err < 3e-2, f"MT adamw drift {err}"
    # grads were zeroed by the kernel
    for p in a.parameters():
        assert p.grad is not None and p.grad.abs().max().item() == 0


# ---- fused RoPE / SwiGLU gate -------------------------------------

def test_fused_rope_fwd_bwd():
    from gpuschedule_amd.ops.elementwise import fused_rope
    torch.manual_seed(14)
    B, H, S, D = 2, 4, 128, 64
    inv = 1.0 / (10000 ** (torch.arange(0, D, 2).float() / D))
    t = torch.arange(S).float()
    fr = torch.outer(t, inv)
    cosb = fr.cos().cuda()
    sinb = fr.sin().cuda()
    x = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = fused_rope(x, cosb, sinb)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    c = cosb[None, None]
    s = sinb[None, None]
    x1, x2 = xf.chunk(2, dim=-1)
    ref = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    ref.backward(dy.float())
    sy = ref.abs().max().item()
    assert (y.float() - ref).abs().max().item() / sy < 1e-2
    sg = xf.grad.abs().max().item()
    assert (x.grad.float() - xf.grad).abs().max().item() / sg < 1e-2


def test_fused_silu_mul():
    from gpuschedule_amd.ops.elementwise import silu_mul
    torch.manual_seed(15)
    g = torch.randn(1024, 512, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    u = torch.randn_like(g).requires_grad_(True)
    y = silu_mul(g, u)
    dy = torch.randn_like(y)
    y.backward(dy)
    gf = g.detach().float().requires_grad_(True)
    uf = u.detach().float().requires_grad_(True)
    ref = torch.nn.functional.silu(gf) * uf
    ref.backward(dy.float())
    sy = ref.abs().max().item()
    assert (y.float() - ref).abs().max().item() / sy < 2e-2
    sc = gf.grad.abs().max().item()
    assert (g.grad.float() - gf.grad).abs().max().item() / sc < 2e-2
    su = uf.grad.abs().max().item()
    assert (u.grad.float() - uf.grad).abs().max().item() / su < 2e-2


def test_norm_residual_fusion():
    from gpuschedule_amd.ops import FusedLayerNorm, FusedRMSNorm
    torch.manual_seed(16)
    for cls, kw in ((FusedLayerNorm, {}), (FusedRMSNorm, {})):
        m = cls(256, **kw).cuda().bfloat16()
        x = torch.randn(64, 256, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        r = torch.randn_like(x).requires_grad_(True)
        y, summed = m(x, residual=r, return_sum=True)
        (y.float().square().mean() + summed.float().mean()).backward()

        m2 = cls(256, **kw).cuda().bfloat16()
        xf = x.detach().clone().requires_grad_(True)
        rf = r.detach().clone().requires_grad_(True)
        s2 = xf + rf
        y2 = m2(s2)
        (y2.float().square().mean() + s2.float().mean()).backward()
        assert (y.float() - y2.float()).abs().max().item() < 3e-2
        for got, want in ((x.grad, xf.grad), (r.grad, rf.grad)):
            sc = want.float().abs().max().item() + 1e-5
            assert (got.float() - want.float()).abs().max().item() / sc \
                < 5e-2


def test_multi_tensor_sgd_matches_torch():
    from gpuschedule_amd.ops import FusedSGD
    torch.manual_seed(17)
    # mixed dtypes like ResNet: bf16 "conv" weights + fp32 "BN" params
    a_bf = torch.nn.Linear(32, 32).cuda().bfloat16()
    a_fp = torch.nn.Linear(32, 4).cuda()
    b_bf = torch.nn.Linear(32, 32).cuda()
    b_fp = torch.nn.Linear(32, 4).cuda()
    with torch.no_grad():
        for src, dst in ((a_bf, b_bf), (a_fp, b_fp)):
            for ps, pd in zip(src.parameters(), dst.parameters()):
                pd.copy_(ps.float())
    params_a = list(a_bf.parameters()) + list(a_fp.parameters())
    oa = FusedSGD(params_a, lr=0.05, momentum=0.9, weight_decay=1e-3)
    ob = torch.optim.SGD(list(b_bf.parameters()) + list(b_fp.parameters()),
                         lr=0.05, momentum=0.9, weight_decay=1e-3)
    x = torch.randn(16, 32, device="cuda")
    for _ in range(5):
        if not oa.fused_zero_grad:
            oa.zero_grad(set_to_none=True)
        (a_fp(a_bf(x.bfloat16()).float()).square().mean()).backward()
        oa.step()
        ob.zero_grad()
        (b_fp(b_bf(x)).square().mean()).backward()
        ob.step()
    assert oa._mt, "multi-tensor SGD did not engage"
    for pa, pb in zip(params_a,
                      list(b_bf.parameters()) + list(b_fp.parameters())):
        err = (pa.float() - pb.float()).abs().max().item()
        assert err < 3e-2, f"MT sgd drift {err}"


# ---- fused embedding ------------------------------------------------

@pytest.mark.parametrize("vh", [(30528, 768), (1024, 256), (997, 64)])
def test_fused_embedding_fwd_bwd(vh):
    """Gather fwd + fp32-atomic-scatter bwd vs fp32 F.embedding,
    including repeated tokens (accumulation in the dense grad)."""
    torch.manual_seed(3)
    V, H = vh
    n = 4096
    w = torch.randn(V, H, device="cuda", dtype=torch.bfloat16)
    # force collisions: draw from a narrow band so rows repeat
    idx = torch.randint(0, min(V, 512), (n,), device="cuda")
    dy = torch.randn(n, H, device="cuda", dtype=torch.bfloat16)

    out = ext().embed_fwd(w, idx)
    ref = torch.nn.functional.embedding(idx, w.float())
    assert torch.equal(out.float(), ref.to(torch.bfloat16).float())

    dw = ext().embed_bwd(dy, idx, V)
    wf = w.float().requires_grad_(True)
    torch.nn.functional.embedding(idx, wf).backward(dy.float())
    scale = wf.grad.abs().max().item() + 1e-3
    assert mx(dw, wf.grad) / scale < 1.2e-2
    # untouched rows must be exactly zero
    mask = torch.ones(V, dtype=torch.bool, device="cuda")
    mask[idx] = False
    assert dw[mask].abs().max().item() == 0.0


def test_fused_embedding_module_autograd():
    from gpuschedule_amd.ops.embedding import FusedEmbedding
    torch.manual_seed(4)
    m = FusedEmbedding(256, 128).to("cuda", torch.bfloat16)
    idx = torch.randint(0, 256, (8, 32), device="cuda")
    y = m(idx)
    assert y.shape == (8, 32, 128) and y.dtype == torch.bfloat16
    y.float().pow(2).mean().backward()
    assert m.weight.grad is not None
    ref = torch.nn.functional.embedding(idx, m.weight.detach().float())
    ref.requires_grad_(False)
    wf = m.weight.detach().float().requires_grad_(True)
    torch.nn.functional.embedding(idx, wf).float().pow(2).mean().backward()
    scale = wf.grad.abs().max().item() + 1e-4
    assert mx(m.weight.grad, wf.grad) / scale < 2e-2


def test_gemm_nt_large_tile_path():
    """The 256-tile pipelined kernel (p8) serves grids >= 200 blocks —
    exercise it at a driver-relevant shape with bias+relu."""
    torch.manual_seed(7)
    M, N, K = 4096, 4096, 4096
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    c = ext().gemm_nt(a, b, bias, True)
    ref = (a.float() @ b.float().t() + bias.float()).relu()
    scale = ref.abs().max().item() + 1e-3
    assert mx(c, ref) / scale < 1.5e-2


def test_gemm_nn_red_major():
    """gemm_nn: C = A @ B with B stored red-major (the dX = dY @ W
    form) via hardware-transpose (ds_read_b64_tr_b16) fragment reads."""
    torch.manual_seed(9)
    M, R, N = 4096, 4096, 4096
    a = torch.randn(M, R, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(R, N, device="cuda", dtype=torch.bfloat16)
    c = ext().gemm_nn(a, b, None, False)
    ref = a.float() @ b.float()
    assert mx(c, ref) / (ref.abs().max().item() + 1e-3) < 1.5e-2


def test_fused_gelu_fwd_bwd():
    from gpuschedule_amd.ops.elementwise import fused_gelu
    torch.manual_seed(12)
    x = torch.randn(4096, 3072, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = fused_gelu(x)
    y.float().square().mean().backward()
    xf = x.detach().float().requires_grad_(True)
    ref = torch.nn.functional.gelu(xf, approximate="tanh")
    ref.square().mean().backward()
    assert mx(y, ref) / (ref.abs().max().item() + 1e-3) < 8e-3
    s = xf.grad.abs().max().item() + 1e-6
    assert (x.grad.float() - xf.grad).abs().max().item() / s < 2e-2


def test_swiglu_packed_fwd_bwd():
    from gpuschedule_amd.ops.elementwise import swiglu_packed
    torch.manual_seed(13)
    gu = torch.randn(512, 2 * 1408, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    y = swiglu_packed(gu)
    y.float().square().mean().backward()
    guf = gu.detach().float().requires_grad_(True)
    g, u = guf.chunk(2, dim=-1)
    ref = torch.nn.functional.silu(g) * u
    ref.square().mean().backward()
    assert mx(y, ref) / (ref.abs().max().item() + 1e-3) < 1e-2
    s = guf.grad.abs().max().item() + 1e-6
    assert (gu.grad.float() - guf.grad).abs().max().item() / s < 2e-2


def test_gemm_fused_addend(monkeypatch):
    """y = x @ W^T + add (residual fused into the epilogue), with the
    addend grad = dy exactly."""
    monkeypatch.setenv("GSAMD_GEMM_LIB", "0")  # cover the kernel path
    from gpuschedule_amd.ops.linear import MFMALinear
    torch.manual_seed(14)
    for M, N, K in [(4096, 4096, 4096), (2048, 512, 512)]:
        lin = MFMALinear(K, N, bias=False).to("cuda", torch.bfloat16)
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        add = torch.randn(M, N, device="cuda", dtype=torch.bfloat16,
                          requires_grad=True)
        y = lin(x, add=add)
        y.float().square().mean().backward()
        ref = x.detach().float() @ lin.weight.detach().float().t() \
            + add.detach().float()
        scale = ref.abs().max().item() + 1e-3
        assert mx(y, ref) / scale < 1.5e-2
        # d(add) must be exactly dy = 2*y/numel (identity pass-through)
        dy = (2 * y.detach().float() / y.numel()).to(torch.bfloat16)
        assert torch.equal(add.grad, dy)


def test_rope_strided_view_matches_contiguous():
    """RoPE reads the packed projection's transposed head-slice view
    directly (no .contiguous() materialization)."""
    from gpuschedule_amd.ops.elementwise import fused_rope
    from gpuschedule_amd.models.llama import LLAMA3_TINY, rope_table
    torch.manual_seed(15)
    cfg = LLAMA3_TINY
    cos, sin = rope_table(cfg, "cuda")
    B, S, H, Dh = 2, 256, cfg.heads, cfg.head_dim
    packed = torch.randn(B, S, H * Dh, device="cuda",
                         dtype=torch.bfloat16)
    view = packed.view(B, S, H, Dh).transpose(1, 2)   # strided BHSD
    assert not view.is_contiguous()
    y_str = fused_rope(view, cos, sin)
    y_cont = fused_rope(view.contiguous(), cos, sin)
    assert torch.equal(y_str, y_cont)


def test_mfma_linear_lib_and_kernel_paths_agree(monkeypatch):
    """The two MFMALinear dispatches (hand-written MFMA kernel vs
    hipBLASLt, GSAMD_GEMM_LIB) must stay numerically interchangeable:
    same bf16 inputs, outputs within bf16 rounding of each other."""
    from gpuschedule_amd.ops.linear import MFMALinear
    torch.manual_seed(21)
    lin = MFMALinear(512, 384).to("cuda", torch.bfloat16)
    x = torch.randn(640, 512, device="cuda", dtype=torch.bfloat16)
    add = torch.randn(640, 384, device="cuda", dtype=torch.bfloat16)
    outs = {}
    for mode in ("0", "1"):
        monkeypatch.setenv("GSAMD_GEMM_LIB", mode)
        outs[mode] = lin(x, add=add).float()
    ref = (x.float() @ lin.weight.float().t() + lin.bias.float()
           + add.float())
    scale = ref.abs().max().item()
    assert (outs["0"] - outs["1"]).abs().max().item() / scale < 1.5e-2
    for mode in outs:
        assert (outs[mode] - ref).abs().max().item() / scale < 1.5e-2
