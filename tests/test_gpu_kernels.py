"""HIP kernel numerics vs the plain PyTorch fp32 references (same op,
same bf16 inputs) — run on the MI355X box: pytest -m gpu."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from lpp_amd import ops

    # fail loudly if the native extension is missing on a GPU box
    return ops.extension()


def _dev():
    return torch.device("cuda", 0)


def close(a, b, rtol=2e-2, atol=1e-2):
    """bf16-aware comparison: 1 bf16 ulp is ~0.4% of the value, so large
    magnitudes need rtol, not a flat atol."""
    a = a.float()
    b = b.float()
    ok = torch.allclose(a, b, rtol=rtol, atol=atol)
    if not ok:
        d = (a - b).abs()
        rel = d / b.abs().clamp(min=1e-3)
        raise AssertionError(f"max abs {d.max()} max rel {rel.max()}")
    return True


# ---------------- rmsnorm ----------------
@pytest.mark.parametrize("H", [8192, 4096, 100])
def test_rmsnorm_fwd_bwd(ext, H):
    from lpp_amd.ops.rmsnorm import rmsnorm, rmsnorm_ref

    torch.manual_seed(0)
    x = torch.randn(3, 37, H, device=_dev(), dtype=torch.bfloat16)
    w = torch.randn(H, device=_dev(), dtype=torch.bfloat16)
    dy = torch.randn_like(x)

    xk = x.clone().requires_grad_(True)
    wk = w.clone().requires_grad_(True)
    yk = rmsnorm(xk, wk, 1e-6)
    yk.backward(dy)

    xr = x.clone().float().requires_grad_(True)
    wr = w.clone().float().requires_grad_(True)
    yr = rmsnorm_ref(xr, wr, 1e-6)
    yr.backward(dy.float())

    close(yk, yr)
    close(xk.grad, xr.grad)
    close(wk.grad, wr.grad, rtol=2e-2, atol=0.5)


# ---------------- rope ----------------
def test_rope_fwd_bwd(ext):
    from lpp_amd.ops.rope import apply_rope, apply_rope_ref, build_rope_cache

    torch.manual_seed(1)
    B, S, H, D = 2, 128, 8, 128
    cos, sin = build_rope_cache(256, D, 10000.0, _dev())
    x = torch.randn(B, S, H, D, device=_dev(), dtype=torch.bfloat16)
    dy = torch.randn_like(x)

    xk = x.clone().requires_grad_(True)
    yk = apply_rope(xk, cos, sin, pos_offset=5)
    yk.backward(dy)

    xr = x.clone().float().requires_grad_(True)
    yr = apply_rope_ref(xr, cos, sin, pos_offset=5)
    yr.backward(dy.float())

    close(yk, yr)
    close(xk.grad, xr.grad)


# ---------------- swiglu ----------------
def test_swiglu_fwd_bwd(ext):
    from lpp_amd.ops.swiglu import swiglu, swiglu_ref

    torch.manual_seed(2)
    g = torch.randn(5, 64, 22016, device=_dev(), dtype=torch.bfloat16)
    u = torch.randn_like(g)
    dy = torch.randn_like(g)

    gk = g.clone().requires_grad_(True)
    uk = u.clone().requires_grad_(True)
    yk = swiglu(gk, uk)
    yk.backward(dy)

    gr = g.clone().float().requires_grad_(True)
    ur = u.clone().float().requires_grad_(True)
    yr = swiglu_ref(gr, ur)
    yr.backward(dy.float())

    close(yk, yr)
    close(gk.grad, gr.grad)
    close(uk.grad, ur.grad)


# ---------------- cross entropy ----------------
def test_cross_entropy_fwd_bwd(ext):
    from lpp_amd.ops.cross_entropy import shifted_cross_entropy, shifted_cross_entropy_ref

    torch.manual_seed(3)
    B, S, V = 2, 65, 32000
    logits = torch.randn(B, S, V, device=_dev(), dtype=torch.bfloat16) * 4
    labels = torch.randint(0, V, (B, S), device=_dev())
    labels[0, :10] = -100

    lk = logits.clone().requires_grad_(True)
    lossk = shifted_cross_entropy(lk, labels)
    lossk.backward()

    lr = logits.clone().float().requires_grad_(True)
    lossr = shifted_cross_entropy_ref(lr, labels)
    lossr.backward()

    assert abs(float(lossk.detach()) - float(lossr.detach())) < 2e-3 * float(lossr.detach())
    gk = lk.grad.float()
    gr = lr.grad
    assert torch.allclose(gk, gr, atol=1e-4), (gk - gr).abs().max()
    # ignored rows have zero grad: labels[0, 1..9] == -100 -> logits rows 0..8
    assert float(gk[0, :9].abs().sum()) == 0.0
    # the last position never receives grad (shift)
    assert float(gk[:, -1].abs().sum()) == 0.0


def test_cross_entropy_all_ignored(ext):
    from lpp_amd.ops.cross_entropy import shifted_cross_entropy

    B, S, V = 1, 8, 1024
    logits = torch.randn(B, S, V, device=_dev(), dtype=torch.bfloat16)
    labels = torch.full((B, S), -100, device=_dev())
    loss = shifted_cross_entropy(logits, labels)
    assert float(loss) == 0.0


# ---------------- fused adamw ----------------
def test_fused_adamw_matches_eager(ext):
    import os

    from lpp_amd.optim import MixedPrecisionAdamW

    torch.manual_seed(4)
    m_gpu = torch.nn.Sequential(torch.nn.Linear(256, 512), torch.nn.Linear(512, 128))
    m_gpu.to(_dev(), torch.bfloat16)
    m_cpu = torch.nn.Sequential(torch.nn.Linear(256, 512), torch.nn.Linear(512, 128))
    m_cpu.load_state_dict({k: v.cpu() for k, v in m_gpu.state_dict().items()})
    m_cpu.to(torch.bfloat16)

    o_gpu = MixedPrecisionAdamW(m_gpu.parameters(), lr=1e-2, weight_decay=0.01)
    o_cpu = MixedPrecisionAdamW(m_cpu.parameters(), lr=1e-2, weight_decay=0.01)
    for step in range(3):
        g = torch.randn(256, dtype=torch.float32)
        for (pg, pc) in zip(o_gpu.params, o_cpu.params):
            gg = torch.randn_like(pc, dtype=torch.float32)
            pc.main_grad.add_(gg)
            pg.main_grad.add_(gg.to(_dev()))
        o_gpu.step(grad_scale=0.5)
        o_cpu.step(grad_scale=0.5)
        o_gpu.zero_grad()
        o_cpu.zero_grad()
    for pg, pc, mg, mc in zip(o_gpu.params, o_cpu.params, o_gpu.masters, o_cpu.masters):
        assert torch.allclose(mg.cpu(), mc, atol=1e-5, rtol=1e-5), (mg.cpu() - mc).abs().max()
        assert torch.allclose(pg.float().cpu(), pc.float(), atol=1e-2)


# ---------------- attention (SDPA fallback until the flash kernel lands) ----
def test_attention_gpu_matches_ref(ext):
    from lpp_amd.ops.attention import causal_attention, causal_attention_ref

    torch.manual_seed(5)
    B, S, H, D = 2, 256, 8, 128
    q = torch.randn(B, S, H, D, device=_dev(), dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    out = causal_attention(q, k, v)
    ref = causal_attention_ref(q.float(), k.float(), v.float())
    assert torch.allclose(out.float(), ref, atol=3e-2), (out.float() - ref).abs().max()


def test_wgrad_f32_accum(ext):
    """dW_f32 += dY^T X via hipBLASLt vs fp32 matmul reference."""
    torch.manual_seed(11)
    T, IN, OUT = 512, 256, 384
    x = torch.randn(T, IN, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(T, OUT, device="cuda", dtype=torch.bfloat16)
    dw = torch.randn(OUT, IN, device="cuda", dtype=torch.float32)
    ref = dw + dy.float().t() @ x.float()
    ext.wgrad_f32_accum(x, dy, dw)
    err = (dw - ref).abs().max() / ref.abs().max().clamp(min=1)
    assert err < 2e-2, err


def test_lp_linear_fused_wgrad_matches_autograd():
    """The LPLinear fused path accumulates into main_grad exactly what
    stock autograd + the accumulate hook would."""
    from lpp_amd.ops.linear import lp_linear

    torch.manual_seed(12)
    B, S, IN, OUT = 2, 64, 128, 96
    w = torch.nn.Parameter(
        torch.randn(OUT, IN, device="cuda", dtype=torch.bfloat16) * 0.05
    )
    x = torch.randn(B, S, IN, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, S, OUT, device="cuda", dtype=torch.bfloat16)

    # reference: stock autograd, grads cast-accumulated to fp32
    xr = x.detach().clone().requires_grad_(True)
    out_ref = torch.nn.functional.linear(xr, w)
    out_ref.backward(dy)
    ref_main = w.grad.float()
    w.grad = None

    w.main_grad = torch.zeros(OUT, IN, device="cuda", dtype=torch.float32)
    out = lp_linear(x, w)
    assert torch.equal(out, out_ref)
    out.backward(dy)
    assert w.grad is None  # fused path bypasses autograd for the weight
    err = (w.main_grad - ref_main).abs().max() / ref_main.abs().max().clamp(min=1e-3)
    assert err < 2e-2, err
    err_x = (x.grad.float() - xr.grad.float()).abs().max()
    assert err_x < 1e-2, err_x


def test_rmsnorm_bwd_deterministic_dw(ext):
    """The partial-buffer dw reduction is order-deterministic (no atomics)."""
    torch.manual_seed(13)
    x = torch.randn(4096, 8192, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(8192, device="cuda", dtype=torch.float32)
    dy = torch.randn_like(x)
    _, inv = ext.rmsnorm_fwd(x, w, 1e-6)
    _, dw1 = ext.rmsnorm_bwd(dy, x, w, inv)
    _, dw2 = ext.rmsnorm_bwd(dy, x, w, inv)
    assert torch.equal(dw1, dw2)


def test_transpose2d(ext):
    """LDS-tiled transpose vs torch .t() for even and ragged shapes."""
    torch.manual_seed(21)
    for R, C in ((512, 256), (4096, 8192), (100, 130), (64, 72)):
        x = torch.randn(R, C, device="cuda", dtype=torch.bfloat16)
        assert torch.equal(ext.transpose2d(x), x.t().contiguous()), (R, C)
    h = torch.randn(256, 192, device="cuda", dtype=torch.float16)
    assert torch.equal(ext.transpose2d(h), h.t().contiguous())


def test_wgrad_f32_accum_pre(ext):
    """Pre-transposed wgrad formulation == natural formulation == fp32 ref."""
    torch.manual_seed(22)
    T, IN, OUT = 512, 256, 384
    x = torch.randn(T, IN, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(T, OUT, device="cuda", dtype=torch.bfloat16)
    dw = torch.randn(OUT, IN, device="cuda", dtype=torch.float32)
    ref = dw + dy.float().t() @ x.float()
    ext.wgrad_f32_accum_pre(ext.transpose2d(x), ext.transpose2d(dy), dw)
    err = (dw - ref).abs().max() / ref.abs().max().clamp(min=1)
    assert err < 2e-2, err


def test_lp_linear_pre_path_matches_natural(monkeypatch):
    """LPLinear backward with LPP_WGRAD_PRE on/off accumulates the same
    main_grad (down to GEMM reduction-order noise)."""
    from lpp_amd.ops.linear import lp_linear

    torch.manual_seed(23)
    B, S, IN, OUT = 2, 128, 256, 192

    def run(pre: str):
        monkeypatch.setenv("LPP_WGRAD_PRE", pre)
        torch.manual_seed(23)
        w = torch.nn.Parameter(
            torch.randn(OUT, IN, device="cuda", dtype=torch.bfloat16) * 0.05
        )
        x = torch.randn(B, S, IN, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        dy = torch.randn(B, S, OUT, device="cuda", dtype=torch.bfloat16)
        w.main_grad = torch.zeros(OUT, IN, device="cuda", dtype=torch.float32)
        out = lp_linear(x, w)
        out.backward(dy)
        return w.main_grad.clone(), x.grad.clone()

    g_pre, dx_pre = run("1")
    g_nat, dx_nat = run("0")
    assert torch.equal(dx_pre, dx_nat)
    err = (g_pre - g_nat).abs().max() / g_nat.abs().max().clamp(min=1e-3)
    assert err < 1e-2, err


def test_xt_cache_shared_and_invalidated(ext):
    """The transpose memo reuses xT for the same activation and refreshes
    for a new one."""
    from lpp_amd.ops import linear as L

    L._xt_cache.clear()
    x1 = torch.randn(64, 128, device="cuda", dtype=torch.bfloat16)
    t1 = L._xt_cache.get(ext, x1)
    t1b = L._xt_cache.get(ext, x1)
    assert t1 is t1b
    x2 = torch.randn(64, 128, device="cuda", dtype=torch.bfloat16)
    t2 = L._xt_cache.get(ext, x2)
    assert t2 is not t1
    assert torch.equal(t2, x2.t().contiguous())
    L._xt_cache.clear()


def test_dgrad_wt_cache_and_invalidate(ext):
    """dx via the cached W^T equals dy @ W; invalidation refreshes after an
    in-place weight update."""
    from lpp_amd.ops import linear as L

    w = torch.nn.Parameter(torch.randn(96, 128, device="cuda", dtype=torch.bfloat16))
    dy = torch.randn(32, 96, device="cuda", dtype=torch.bfloat16)
    wt = L._weight_t(ext, w)
    assert torch.equal(wt, w.data.t().contiguous())
    assert L._weight_t(ext, w) is wt  # cached
    dx_ref = dy @ w
    dx = dy @ wt.t()
    assert torch.allclose(dx.float(), dx_ref.float(), rtol=2e-2, atol=1e-2)
    with torch.no_grad():
        w.data.mul_(2.0)
    L.invalidate_weight_transposes(torch.nn.ParameterList([w]))
    wt2 = L._weight_t(ext, w)
    assert torch.equal(wt2, w.data.t().contiguous())


def test_accum_bf16_f32(ext):
    torch.manual_seed(31)
    for n in (8 * 1024, 1000, 7):
        dst = torch.randn(n, device="cuda", dtype=torch.float32)
        src = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        ref = dst + src.float()
        ext.accum_bf16_f32(dst, src)
        assert torch.equal(dst, ref), n


def test_wgrad_bf16d_matches_f32d(monkeypatch):
    """bf16-D wgrad + fp32 accumulate stays within one bf16 rounding of the
    exact fp32-D epilogue."""
    from lpp_amd.ops.linear import lp_linear

    def run(flag):
        monkeypatch.setenv("LPP_WGRAD_BF16D", flag)
        torch.manual_seed(33)
        w = torch.nn.Parameter(
            torch.randn(96, 128, device="cuda", dtype=torch.bfloat16) * 0.05)
        x = torch.randn(4, 64, 128, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        dy = torch.randn(4, 64, 96, device="cuda", dtype=torch.bfloat16)
        w.main_grad = torch.zeros(96, 128, device="cuda", dtype=torch.float32)
        lp_linear(x, w).backward(dy)
        return w.main_grad.clone()

    g16 = run("1")
    g32 = run("0")
    err = (g16 - g32).abs().max() / g32.abs().max().clamp(min=1e-3)
    assert err < 1e-2, err


def test_fused_wgrad_notifies_accumulate():
    """The fused wgrad path must fire the per-param accumulate notifier the
    DP-bucket overlap relies on (no p.grad -> no autograd hook)."""
    from lpp_amd.ops.linear import lp_linear

    w = torch.nn.Parameter(
        torch.randn(96, 128, device="cuda", dtype=torch.bfloat16) * 0.05)
    w.main_grad = torch.zeros(96, 128, device="cuda", dtype=torch.float32)
    fired = []
    w._on_accumulate = lambda p: fired.append(p is w)
    x = torch.randn(2, 64, 128, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    lp_linear(x, w).backward(torch.randn(2, 64, 96, device="cuda",
                                         dtype=torch.bfloat16))
    assert fired == [True]
