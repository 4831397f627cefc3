"""Op-level GPU numerics tier: each HIP kernel vs a plain
PyTorch fp32 reference of the same op.  All tests @pytest.mark.gpu."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from scaffold.ops.backend import require_ext

    return require_ext()


def _bf(x):
    return x.to(torch.bfloat16)


# ------------------------------------------------------------------ MFMA probe
def test_mfma_layout_32x32x16(ext):
    torch.manual_seed(0)
    A = torch.randn(32, 16).cuda()
    B = torch.randn(16, 32).cuda()  # asymmetric (guide G9)
    C = ext.probe_mfma(_bf(A), _bf(B), 0)
    ref = _bf(A).float() @ _bf(B).float()
    assert torch.allclose(C, ref, rtol=1e-2, atol=1e-2), (C - ref).abs().max()


def test_mfma_layout_16x16x32(ext):
    torch.manual_seed(1)
    A = torch.randn(16, 32).cuda()
    B = torch.randn(32, 16).cuda()
    C = ext.probe_mfma(_bf(A), _bf(B), 1)
    ref = _bf(A).float() @ _bf(B).float()
    assert torch.allclose(C, ref, rtol=1e-2, atol=1e-2), (C - ref).abs().max()


# ------------------------------------------------------------------- AdamW
@pytest.mark.parametrize("grad_dtype", [torch.bfloat16, torch.float32])
def test_fused_adamw_vs_cpu(ext, grad_dtype):
    from scaffold import ops

    torch.manual_seed(2)
    shapes = [(1000,), (128, 77), (3,), (513,)]
    masters = [torch.randn(s) for s in shapes]
    grads = [torch.randn(s).to(grad_dtype) for s in shapes]
    ms = [torch.rand(s) * 0.1 for s in shapes]
    vs = [torch.rand(s) * 0.01 for s in shapes]
    outs = [m.to(torch.bfloat16) for m in masters]
    kw = dict(lr=1e-2, beta1=0.9, beta2=0.95, eps=1e-8, weight_decay=0.01,
              step=3, grad_scale=0.125)

    # CPU reference (the eager path in scaffold.ops)
    cm = [t.clone() for t in masters]
    cg = [t.clone() for t in grads]
    cms = [t.clone() for t in ms]
    cvs = [t.clone() for t in vs]
    cout = [t.clone() for t in outs]
    ops.fused_adamw_step(cm, cg, cms, cvs, cout, **kw)

    gm = [t.cuda() for t in masters]
    gg = [t.cuda() for t in grads]
    gms = [t.cuda() for t in ms]
    gvs = [t.cuda() for t in vs]
    gout = [t.cuda() for t in outs]
    ext.fused_adamw(gm, gg, gms, gvs, gout, kw["lr"], kw["beta1"], kw["beta2"],
                    kw["eps"], kw["weight_decay"], kw["step"], kw["grad_scale"])
    torch.cuda.synchronize()
    for i in range(len(shapes)):
        assert torch.allclose(gm[i].cpu(), cm[i], rtol=1e-5, atol=1e-6), f"master {i}"
        assert torch.allclose(gms[i].cpu(), cms[i], rtol=1e-5, atol=1e-6), f"m {i}"
        assert torch.allclose(gvs[i].cpu(), cvs[i], rtol=1e-5, atol=1e-6), f"v {i}"
        assert torch.allclose(gout[i].cpu().float(), cout[i].float(),
                              rtol=1e-2, atol=1e-2), f"out {i}"


# --------------------------------------------------------------- LayerNorm
@pytest.mark.parametrize("R,D", [(128, 768), (64, 1024), (7, 333)])
def test_layernorm_fwd_bwd(ext, R, D):
    torch.manual_seed(3)
    x32 = torch.randn(R, D, requires_grad=True)
    w32 = torch.randn(D, requires_grad=True)
    b32 = torch.randn(D, requires_grad=True)
    y32 = torch.nn.functional.layer_norm(x32, (D,), w32, b32, 1e-5)
    dy = torch.randn(R, D)
    y32.backward(dy)

    from scaffold import ops

    xg = _bf(x32.detach()).cuda().requires_grad_(True)
    wg = _bf(w32.detach()).cuda().requires_grad_(True)
    bg = _bf(b32.detach()).cuda().requires_grad_(True)
    yg = ops.layer_norm(xg, wg, bg, 1e-5)
    yg.backward(_bf(dy).cuda())
    torch.cuda.synchronize()

    assert torch.allclose(yg.detach().cpu().float(), y32.detach().float(),
                          rtol=5e-2, atol=5e-2)
    assert torch.allclose(xg.grad.cpu().float(), x32.grad.float(),
                          rtol=1e-1, atol=1e-1)
    # column reductions accumulate over R rows -> looser bands
    assert torch.allclose(wg.grad.cpu().float(), w32.grad.float(),
                          rtol=1e-1, atol=R * 2e-2)
    assert torch.allclose(bg.grad.cpu().float(), b32.grad.float(),
                          rtol=1e-1, atol=R * 2e-2)


# ----------------------------------------------------------------- RMSNorm
def test_rmsnorm_fwd_bwd(ext):
    torch.manual_seed(4)
    R, D = 256, 512
    x32 = torch.randn(R, D, requires_grad=True)
    w32 = torch.randn(D, requires_grad=True)
    rms = x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + 1e-5) * w32
    dy = torch.randn(R, D)
    rms.backward(dy)

    from scaffold import ops

    xg = _bf(x32.detach()).cuda().requires_grad_(True)
    wg = _bf(w32.detach()).cuda().requires_grad_(True)
    yg = ops.rms_norm(xg, wg, 1e-5)
    yg.backward(_bf(dy).cuda())
    torch.cuda.synchronize()
    assert torch.allclose(yg.detach().cpu().float(), rms.detach().float(),
                          rtol=5e-2, atol=5e-2)
    assert torch.allclose(xg.grad.cpu().float(), x32.grad.float(),
                          rtol=1e-1, atol=1e-1)
    assert torch.allclose(wg.grad.cpu().float(), w32.grad.float(),
                          rtol=1e-1, atol=R * 2e-2)


# -------------------------------------------------------------------- RoPE
def test_rope_roundtrip_and_ref(ext):
    from scaffold import ops

    torch.manual_seed(5)
    B, H, S, D = 2, 4, 64, 128
    x = torch.randn(B, H, S, D)
    cos, sin = ops.rope_tables(S, D, 500000.0, "cpu")
    ref = ops._rope_eager(x, cos, sin)

    xg = _bf(x).cuda()
    y = ext.rope_apply(xg, cos.cuda(), sin.cuda(), False)
    torch.cuda.synchronize()
    assert torch.allclose(y.cpu().float(), ref.float(), rtol=3e-2, atol=3e-2)
    # inverse rotation undoes the forward
    x2 = ext.rope_apply(y, cos.cuda(), sin.cuda(), True)
    assert torch.allclose(x2.cpu().float(), x.float(), rtol=5e-2, atol=5e-2)


# --------------------------------------------------- softmax cross-entropy
@pytest.mark.parametrize("R,V", [(64, 1000), (32, 50257)])
def test_softmax_xent(ext, R, V):
    torch.manual_seed(6)
    logits32 = (torch.randn(R, V) * 3).requires_grad_(True)
    tgt = torch.randint(0, V, (R,))
    loss32 = torch.nn.functional.cross_entropy(logits32, tgt)
    loss32.backward()

    from scaffold import ops

    lg = _bf(logits32.detach()).cuda().requires_grad_(True)
    lossg = ops.softmax_cross_entropy(lg, tgt.cuda())
    lossg.backward()
    torch.cuda.synchronize()
    assert abs(lossg.item() - loss32.item()) < 0.05 * max(1.0, abs(loss32.item()))
    assert torch.allclose(lg.grad.cpu().float(), logits32.grad.float(),
                          rtol=1e-1, atol=1e-3)


# --------------------------------------------------------------- attention
@pytest.mark.parametrize("B,H,Hkv,S,D,causal", [
    (2, 4, 4, 128, 64, True),
    (1, 8, 2, 256, 128, True),   # GQA (8-wave fwd kernel in auto mode)
    (2, 2, 2, 96, 64, False),    # non-causal + ragged S
    (1, 4, 2, 200, 128, True),   # causal + ragged S + GQA at D=128
    (1, 4, 4, 512, 64, True),    # 8-wave fwd at D=64, 2 q-blocks
    (1, 4, 2, 384, 128, False),  # 8-wave fwd, ragged S (384 = 1.5 blocks)
])
def test_attention_fwd_bwd(ext, B, H, Hkv, S, D, causal):
    torch.manual_seed(7)
    q32 = torch.randn(B, H, S, D, requires_grad=True)
    k32 = torch.randn(B, Hkv, S, D, requires_grad=True)
    v32 = torch.randn(B, Hkv, S, D, requires_grad=True)
    rep = H // Hkv
    kf = k32.repeat_interleave(rep, dim=1)
    vf = v32.repeat_interleave(rep, dim=1)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q32, kf, vf, is_causal=causal)
    do = torch.randn(B, H, S, D)
    ref.backward(do)

    from scaffold import ops

    qg = _bf(q32.detach()).cuda().requires_grad_(True)
    kg = _bf(k32.detach()).cuda().requires_grad_(True)
    vg = _bf(v32.detach()).cuda().requires_grad_(True)
    og = ops.attention(qg, kg, vg, causal=causal)
    og.backward(_bf(do).cuda())
    torch.cuda.synchronize()

    assert torch.allclose(og.detach().cpu().float(), ref.detach().float(),
                          rtol=5e-2, atol=5e-2), \
        (og.detach().cpu().float() - ref.detach().float()).abs().max()
    for name, got, want in [("dq", qg.grad, q32.grad), ("dk", kg.grad, k32.grad),
                            ("dv", vg.grad, v32.grad)]:
        g = got.cpu().float()
        w = want.float()
        err = (g - w).abs().max().item()
        tol = 0.12 * max(1.0, w.abs().max().item())
        assert err < tol, f"{name}: max err {err} vs tol {tol}"


# ------------------------------------------------------- K5 MFMA linear
@pytest.mark.parametrize("epi", ["none", "gelu", "resid"])
def test_linear_fwd256_vs_fp32(ext, epi):
    """The 256^2 counted-vmcnt kernel needs >=224 blocks to dispatch;
    M=4096 x N=3584 = 224 tiles exercises it (K=192 covers the last-tile
    wait path: 3 K-tiles)."""
    M, N, K = 4096, 3584, 192
    torch.manual_seed(12)
    x = torch.randn(M, K)
    w = torch.randn(N, K) * 0.05
    b = torch.randn(N)
    res = torch.randn(M, N)
    xb, wb, bb, rb = (_bf(t).cuda() for t in (x, w, b, res))
    ref = _bf(x).float() @ _bf(w).float().t() + _bf(b).float()
    if epi == "none":
        (y,) = ext.linear_fwd(xb, wb, bb, 0, None)
        want = ref
    elif epi == "gelu":
        y, pre = ext.linear_fwd(xb, wb, bb, 1, None)
        assert torch.allclose(pre.cpu().float(), ref, rtol=3e-2, atol=3e-2)
        want = torch.nn.functional.gelu(ref, approximate="tanh")
    else:
        (y,) = ext.linear_fwd(xb, wb, bb, 2, rb)
        want = ref + _bf(res).float()
    got = y.cpu().float()
    torch.cuda.synchronize()
    assert torch.allclose(got, want, rtol=3e-2, atol=3e-2), \
        (got - want).abs().max().item()


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (256, 384, 192), (512, 768, 320)])
@pytest.mark.parametrize("epi", ["none", "gelu", "resid"])
def test_linear_fwd_vs_fp32(ext, M, N, K, epi):
    torch.manual_seed(11)
    x = torch.randn(M, K)
    w = torch.randn(N, K) * 0.05  # asymmetric, scaled like real weights
    b = torch.randn(N)
    res = torch.randn(M, N)
    xb, wb, bb, rb = (_bf(t).cuda() for t in (x, w, b, res))

    ref = _bf(x).float() @ _bf(w).float().t() + _bf(b).float()
    if epi == "none":
        (y,) = ext.linear_fwd(xb, wb, bb, 0, None)
        want = ref
    elif epi == "gelu":
        y, pre = ext.linear_fwd(xb, wb, bb, 1, None)
        assert torch.allclose(pre.cpu().float(), ref, rtol=3e-2, atol=3e-2)
        want = torch.nn.functional.gelu(ref, approximate="tanh")
    else:
        (y,) = ext.linear_fwd(xb, wb, bb, 2, rb)
        want = ref + _bf(res).float()
    got = y.cpu().float()
    err = (got - want).abs().max().item()
    tol = 2e-2 * max(1.0, want.abs().max().item())
    assert err < tol, f"{epi} M{M} N{N} K{K}: max err {err} vs tol {tol}"


def test_linear_fwd_no_bias(ext):
    torch.manual_seed(12)
    x = _bf(torch.randn(128, 64)).cuda()
    w = _bf(torch.randn(128, 64) * 0.05).cuda()
    (y,) = ext.linear_fwd(x, w, None, 0, None)
    ref = x.float() @ w.float().t()
    assert torch.allclose(y.cpu().float(), ref.cpu(), rtol=2e-2, atol=2e-2)


def test_linear_autograd_matches_eager():
    """ops.linear/linear_gelu/linear_residual fwd+bwd vs bf16 eager."""
    from scaffold import ops

    torch.manual_seed(13)
    M, N, K = 256, 256, 128
    for fn, eager in [
        (lambda x, w, b: ops.linear(x, w, b),
         lambda x, w, b: torch.nn.functional.linear(x, w, b)),
        (lambda x, w, b: ops.linear_gelu(x, w, b),
         lambda x, w, b: torch.nn.functional.gelu(
             torch.nn.functional.linear(x, w, b), approximate="tanh")),
    ]:
        x = _bf(torch.randn(M, K)).cuda().requires_grad_(True)
        w = _bf(torch.randn(N, K) * 0.05).cuda().requires_grad_(True)
        b = _bf(torch.randn(N)).cuda().requires_grad_(True)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        dy = _bf(torch.randn(M, N)).cuda()
        y = fn(x, w, b); y.backward(dy)
        ye = eager(x2, w2, b2); ye.backward(dy)
        torch.cuda.synchronize()
        for name, g, ge in [("y", y, ye), ("dx", x.grad, x2.grad),
                            ("dw", w.grad, w2.grad), ("db", b.grad, b2.grad)]:
            a, e = g.detach().float(), ge.detach().float()
            err = (a - e).abs().max().item()
            tol = 3e-2 * max(1.0, e.abs().max().item())
            assert err < tol, f"{name}: {err} vs {tol}"

    # residual variant
    x = _bf(torch.randn(M, K)).cuda().requires_grad_(True)
    w = _bf(torch.randn(N, K) * 0.05).cuda().requires_grad_(True)
    r = _bf(torch.randn(M, N)).cuda().requires_grad_(True)
    y = ops.linear_residual(x, w, None, r)
    y.backward(_bf(torch.ones(M, N)).cuda())
    torch.cuda.synchronize()
    assert torch.allclose(r.grad.float(), torch.ones(M, N).cuda(), atol=0)


def test_linear_fwd_256_path(ext):
    """Shape that dispatches to the 256^2 8-wave kernel (grid >= 224)."""
    torch.manual_seed(14)
    M, N, K = 3584, 4096, 128
    x = _bf(torch.randn(M, K)).cuda()
    w = _bf(torch.randn(N, K) * 0.05).cuda()
    b = _bf(torch.randn(N)).cuda()
    (y,) = ext.linear_fwd(x, w, b, 0, None)
    ref = x.float() @ w.float().t() + b.float()
    err = (y.float() - ref).abs().max().item()
    tol = 2e-2 * max(1.0, ref.abs().max().item())
    assert err < tol, f"max err {err} vs tol {tol}"
    # gelu epilogue on the same path
    y2, pre = ext.linear_fwd(x, w, b, 1, None)
    want = torch.nn.functional.gelu(ref, approximate="tanh")
    err = (y2.float() - want).abs().max().item()
    assert err < tol, f"gelu: max err {err} vs tol {tol}"


def test_swiglu_fwd_bwd(ext):
    from scaffold import ops

    torch.manual_seed(15)
    g32 = torch.randn(64, 256, requires_grad=True)
    u32 = torch.randn(64, 256, requires_grad=True)
    ref = torch.nn.functional.silu(g32) * u32
    dh = torch.randn(64, 256)
    ref.backward(dh)

    g = _bf(g32.detach()).cuda().requires_grad_(True)
    u = _bf(u32.detach()).cuda().requires_grad_(True)
    h = ops.swiglu(g, u)
    h.backward(_bf(dh).cuda())
    torch.cuda.synchronize()
    for name, got, want in [("h", h.detach(), ref.detach()),
                            ("dg", g.grad, g32.grad), ("du", u.grad, u32.grad)]:
        a, e = got.cpu().float(), want.float()
        err = (a - e).abs().max().item()
        tol = 3e-2 * max(1.0, e.abs().max().item())
        assert err < tol, f"{name}: {err} vs {tol}"


def test_swiglu_packed(ext):
    from scaffold import ops

    torch.manual_seed(16)
    gu32 = torch.randn(32, 512, requires_grad=True)
    Fh = 256
    ref = torch.nn.functional.silu(gu32[..., :Fh]) * gu32[..., Fh:]
    dh = torch.randn(32, Fh)
    ref.backward(dh)

    gu = _bf(gu32.detach()).cuda().requires_grad_(True)
    h = ops.swiglu_packed(gu)
    h.backward(_bf(dh).cuda())
    torch.cuda.synchronize()
    for name, got, want in [("h", h.detach(), ref.detach()), ("dgu", gu.grad, gu32.grad)]:
        a, e = got.cpu().float(), want.float()
        err = (a - e).abs().max().item()
        tol = 3e-2 * max(1.0, e.abs().max().item())
        assert err < tol, f"{name}: {err} vs {tol}"


def test_kernel_determinism(ext):
    """SURVEY §5.2 race screen: our kernels are bitwise deterministic
    across repeated runs on identical inputs (no inter-workgroup
    hand-offs by design; kernel boundaries are the only release points)."""
    from scaffold import ops

    torch.manual_seed(17)
    q = _bf(torch.randn(2, 4, 256, 64)).cuda().requires_grad_(True)
    k = _bf(torch.randn(2, 4, 256, 64)).cuda().requires_grad_(True)
    v = _bf(torch.randn(2, 4, 256, 64)).cuda().requires_grad_(True)
    do = _bf(torch.randn(2, 4, 256, 64)).cuda()

    outs = []
    for _ in range(2):
        q.grad = k.grad = v.grad = None
        o = ops.attention(q, k, v, True)
        o.backward(do)
        torch.cuda.synchronize()
        outs.append((o.detach().clone(), q.grad.clone(), k.grad.clone(), v.grad.clone()))
    for a, b in zip(outs[0], outs[1]):
        assert torch.equal(a, b), "nondeterministic kernel output"

    x = _bf(torch.randn(512, 768)).cuda()
    w = _bf(torch.ones(768)).cuda()
    b = _bf(torch.zeros(768)).cuda()
    y1, *_ = ext.layernorm_fwd(x, w, b, 1e-5)
    y2, *_ = ext.layernorm_fwd(x, w, b, 1e-5)
    assert torch.equal(y1, y2)

    # column reduces: the slice folds must be fixed-order (an LDS
    # atomic fold would vary fp32 addition order run to run)
    xc = _bf(torch.randn(16384, 768)).cuda()
    assert torch.equal(ext.colsum(xc), ext.colsum(xc))
    xb = _bf(torch.randn(8, 64, 28, 28)).cuda().to(memory_format=torch.channels_last)
    wb = torch.rand(64, device="cuda") + 0.5
    bb = torch.randn(64, device="cuda")
    outs_bn = []
    for _ in range(2):
        y, mean, rstd = ext.bn_fwd(xb, None, wb, bb, None, None, 0.1, 1e-5, True)
        outs_bn.append((y, mean, rstd))
    for a, b2 in zip(outs_bn[0], outs_bn[1]):
        assert torch.equal(a, b2), "nondeterministic BN stats"


def test_lm_head_ce_matches_eager():
    from scaffold import ops

    torch.manual_seed(18)
    T, D, V = 512, 128, 1024
    h32 = torch.randn(T, D, requires_grad=True)
    w32 = (torch.randn(V, D) * 0.05).requires_grad_(True)
    t = torch.randint(0, V, (T,))
    ref = torch.nn.functional.cross_entropy(h32 @ w32.t(), t)
    ref.backward()

    # exercise the CHUNKED Function explicitly (the default dispatch is
    # the unfused path), with a chunk smaller than T to cover the loop
    old = ops._LmHeadCE.CHUNK
    ops._LmHeadCE.CHUNK = 128
    try:
        h = _bf(h32.detach()).cuda().requires_grad_(True)
        w = _bf(w32.detach()).cuda().requires_grad_(True)
        loss = ops._LmHeadCE.apply(h, w, t.cuda())
        loss.backward()
        torch.cuda.synchronize()
    finally:
        ops._LmHeadCE.CHUNK = old
    assert abs(loss.item() - ref.item()) < 0.03 * max(1.0, abs(ref.item()))
    for name, g, ge in [("dh", h.grad, h32.grad), ("dw", w.grad, w32.grad)]:
        a, e = g.cpu().float(), ge.float()
        err = (a - e).abs().max().item()
        tol = 5e-2 * max(0.05, e.abs().max().item())
        assert err < tol, f"{name}: {err} vs {tol}"


def test_attention_lse_dlse_backward(ext):
    """attention_lse: gradients through BOTH outputs (o AND lse) vs the
    fp32 eager reference — the context-parallel merge's building block
    (dlse folded into the backward's delta term)."""
    from scaffold import ops

    torch.manual_seed(3)
    q32 = torch.randn(1, 2, 96, 64)
    k32 = torch.randn(1, 2, 96, 64)
    v32 = torch.randn(1, 2, 96, 64)
    do = torch.randn(1, 2, 96, 64)
    dl = torch.randn(1, 2, 96) * 0.1
    qe, ke, ve = [t.clone().requires_grad_(True) for t in (q32, k32, v32)]
    oe, le = ops._eager_attn_lse(qe, ke, ve, True)
    torch.autograd.backward([oe, le], [do, dl])

    qb, kb, vb = [t.to(torch.bfloat16).cuda().requires_grad_(True)
                  for t in (q32, k32, v32)]
    ob, lb = ops.attention_lse(qb, kb, vb, True)
    torch.autograd.backward([ob, lb], [do.to(torch.bfloat16).cuda(), dl.cuda()])
    torch.cuda.synchronize()
    for n, a, b in [("o", ob, oe), ("lse", lb, le), ("dq", qb.grad, qe.grad),
                    ("dk", kb.grad, ke.grad), ("dv", vb.grad, ve.grad)]:
        err = (a.detach().cpu().float() - b.detach().float()).abs().max().item()
        tol = 0.08 * max(1.0, b.detach().float().abs().max().item())
        assert err < tol, (n, err, tol)

# ----------------------------------------------------- fused BatchNorm
@pytest.mark.parametrize("C,relu,res", [
    (64, True, False), (128, True, True), (256, False, False),
])
def test_batch_norm_act_vs_fp32(ext, C, relu, res):
    torch.manual_seed(21)
    N, H, W = 4, 14, 14
    x32 = torch.randn(N, C, H, W) * 2 + 0.5
    r32 = torch.randn(N, C, H, W) if res else None
    w32 = (torch.rand(C) + 0.5).requires_grad_(True)
    b32 = torch.randn(C).requires_grad_(True)
    rm = torch.zeros(C)
    rv = torch.ones(C)
    mom, eps = 0.1, 1e-5

    xg = _bf(x32).cuda().to(memory_format=torch.channels_last).requires_grad_(True)
    rg = (_bf(r32).cuda().to(memory_format=torch.channels_last).requires_grad_(True)
          if res else None)
    wg = w32.detach().clone().cuda().requires_grad_(True)
    bg = b32.detach().clone().cuda().requires_grad_(True)
    rmg, rvg = rm.clone().cuda(), rv.clone().cuda()

    from scaffold import ops

    y = ops.batch_norm_act(xg, rg, wg, bg, rmg, rvg, mom, eps, relu, True)
    dy = torch.randn(N, C, H, W)
    y.backward(_bf(dy).cuda().to(memory_format=torch.channels_last))
    torch.cuda.synchronize()

    # fp32 reference on the bf16-quantized inputs
    xr = _bf(x32).float().requires_grad_(True)
    rr = _bf(r32).float().requires_grad_(True) if res else None
    ref = torch.nn.functional.batch_norm(xr, rm, rv, w32, b32, True, mom, eps)
    if res:
        ref = ref + rr
    if relu:
        ref = torch.nn.functional.relu(ref)
    ref.backward(_bf(dy).float())

    assert torch.allclose(y.detach().cpu().float(), ref.detach(),
                          rtol=3e-2, atol=3e-2),         (y.detach().cpu().float() - ref.detach()).abs().max()
    assert torch.allclose(rmg.cpu(), rm, rtol=1e-3, atol=1e-4)
    assert torch.allclose(rvg.cpu(), rv, rtol=1e-3, atol=1e-3)
    for name, got, want in [("dx", xg.grad, xr.grad), ("dw", wg.grad, w32.grad),
                            ("db", bg.grad, b32.grad)] +                            ([("dres", rg.grad, rr.grad)] if res else []):
        g = got.cpu().float()
        err = (g - want).abs().max().item()
        tol = 3e-2 * max(1.0, want.abs().max().item())
        assert err < tol, f"{name}: max err {err} vs tol {tol}"

# ------------------------------------------------------------- colsum
@pytest.mark.parametrize("R,C", [(16384, 768), (1000, 333), (64, 2304)])
def test_colsum_vs_fp32(ext, R, C):
    torch.manual_seed(31)
    x = torch.randn(R, C)
    got = ext.colsum(_bf(x).cuda()).cpu().float()
    want = _bf(x).float().sum(dim=0)
    torch.cuda.synchronize()
    assert torch.allclose(got, want, rtol=1e-2, atol=R * 2e-3), \
        (got - want).abs().max().item()



# -------------------------------------------------------- decode GEMV
@pytest.mark.parametrize("B,N,K,bias", [
    (1, 512, 256, False), (4, 1000, 768, True), (8, 4096, 4096, False),
])
def test_decode_gemv_vs_fp32(ext, B, N, K, bias):
    torch.manual_seed(33)
    x = torch.randn(B, K)
    w = torch.randn(N, K) * 0.05
    b = torch.randn(N) if bias else None
    got = ext.decode_gemv(_bf(x).cuda(), _bf(w).cuda(),
                          _bf(b).cuda() if bias else None).cpu().float()
    want = _bf(x).float() @ _bf(w).float().t()
    if bias:
        want = want + _bf(b).float()
    torch.cuda.synchronize()
    assert torch.allclose(got, want, rtol=2e-2, atol=2e-2), \
        (got - want).abs().max().item()
