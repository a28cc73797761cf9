#!/usr/bin/env python
"""Per-op A/B microbench: HIP kernel vs eager PyTorch on MI355X.

Shapes are the 65B hot-path shapes (hidden 8192, intermediate 22016,
heads 64 x 128, vocab 32000, seq 4096, mbs 1). Reports ms and achieved
HBM GB/s (bytes moved / time) for the memory-bound ops.

Run on the GPU box:  python scripts/opbench.py [--csv out.csv]
"""

from __future__ import annotations

import argparse
import pathlib
import sys
import time

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

DEV = torch.device("cuda", 0)


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0  # ms


def bench_rmsnorm(rows=4096, H=8192):
    from lpp_amd.ops.rmsnorm import rmsnorm, rmsnorm_ref

    x = torch.randn(rows, H, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(H, device=DEV, dtype=torch.bfloat16)
    bytes_moved = x.numel() * 2 * 2  # read + write
    t_hip = timeit(lambda: rmsnorm(x, w, 1e-6))
    t_ref = timeit(lambda: rmsnorm_ref(x, w, 1e-6))
    return [("rmsnorm_fwd", t_hip, t_ref, bytes_moved)]


def bench_rmsnorm_bwd(rows=4096, H=8192):
    from lpp_amd import ops

    ext = ops.extension()
    x = torch.randn(rows, H, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(H, device=DEV, dtype=torch.float32)
    dy = torch.randn_like(x)
    _, invrms = ext.rmsnorm_fwd(x, w, 1e-6)
    bytes_moved = x.numel() * 2 * 3
    t_hip = timeit(lambda: ext.rmsnorm_bwd(dy, x, w, invrms))

    def ref():
        xr = x.detach().float().requires_grad_(True)
        from lpp_amd.ops.rmsnorm import rmsnorm_ref

        y = rmsnorm_ref(xr, w, 1e-6)
        y.backward(dy.float())

    t_ref = timeit(ref, iters=5, warmup=2)
    return [("rmsnorm_bwd", t_hip, t_ref, bytes_moved)]


def bench_rope(B=1, S=4096, H=64, D=128):
    from lpp_amd.ops.rope import apply_rope, apply_rope_ref, build_rope_cache

    cos, sin = build_rope_cache(S, D, 10000.0, DEV)
    x = torch.randn(B, S, H, D, device=DEV, dtype=torch.bfloat16)
    bytes_moved = x.numel() * 2 * 2
    t_hip = timeit(lambda: apply_rope(x, cos, sin))
    t_ref = timeit(lambda: apply_rope_ref(x, cos, sin))
    return [("rope_fwd", t_hip, t_ref, bytes_moved)]


def bench_swiglu(rows=4096, I=22016):
    from lpp_amd.ops.swiglu import swiglu, swiglu_ref

    g = torch.randn(rows, I, device=DEV, dtype=torch.bfloat16)
    u = torch.randn_like(g)
    bytes_moved = g.numel() * 2 * 3
    t_hip = timeit(lambda: swiglu(g, u))
    t_ref = timeit(lambda: swiglu_ref(g, u))
    return [("swiglu_fwd", t_hip, t_ref, bytes_moved)]


def bench_ce(rows=4096, V=32000):
    from lpp_amd import ops

    ext = ops.extension()
    logits = torch.randn(rows, V, device=DEV, dtype=torch.bfloat16) * 4
    labels = torch.randint(0, V, (rows,), device=DEV)
    bytes_moved = logits.numel() * 2

    t_hip = timeit(lambda: ext.cross_entropy_fwd(logits, labels))

    def ref():
        torch.nn.functional.cross_entropy(logits.float(), labels)

    t_ref = timeit(ref)
    rows_out = [("ce_fwd", t_hip, t_ref, bytes_moved)]

    _, lse, _ = ext.cross_entropy_fwd(logits, labels)
    t_hipb = timeit(lambda: ext.cross_entropy_bwd(logits, labels, lse, 1e-4))

    def refb():
        lf = logits.float().requires_grad_(True)
        torch.nn.functional.cross_entropy(lf, labels).backward()

    t_refb = timeit(refb, iters=5, warmup=2)
    rows_out.append(("ce_bwd", t_hipb, t_refb, bytes_moved * 2))
    return rows_out


def bench_attention(B=1, S=4096, H=64, D=128):
    """Flash fwd/bwd vs torch SDPA at the 65B shape.  NOTE: SDPA on ROCm is
    aotriton's own flash kernel — a much stronger baseline than the
    materialised-mask eager attention the reference actually runs."""
    from lpp_amd import ops
    from lpp_amd.ops.attention import causal_attention_ref

    ext = ops.extension()
    q = torch.randn(B, S, H, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    flops_fwd = 4 * B * H * S * S * D / 2
    t_hip = timeit(lambda: ext.attention_fwd(q, k, v))
    t_ref = timeit(lambda: causal_attention_ref(q, k, v))
    rows = [("attn_fwd", t_hip, t_ref, int(flops_fwd))]

    o, lse2 = ext.attention_fwd(q, k, v)
    t_hipb = timeit(lambda: ext.attention_bwd(do, q, k, v, o, lse2))

    def refb():
        qr = q.detach().requires_grad_(True)
        kr = k.detach().requires_grad_(True)
        vr = v.detach().requires_grad_(True)
        causal_attention_ref(qr, kr, vr).backward(do)

    t_refb = timeit(refb, iters=5, warmup=2)
    rows.append(("attn_bwd", t_hipb, t_refb, int(flops_fwd * 2.5)))
    return rows


def bench_transpose(R=16384, C=8192):
    from lpp_amd import ops

    ext = ops.extension()
    x = torch.randn(R, C, device=DEV, dtype=torch.bfloat16)
    bytes_moved = x.numel() * 2 * 2
    t_hip = timeit(lambda: ext.transpose2d(x))
    t_ref = timeit(lambda: x.t().contiguous())
    return [(f"transpose2d[{R}x{C}]", t_hip, t_ref, bytes_moved)]


def bench_wgrad(T=16384, IN=8192, OUT=8192):
    """wgrad formulations: eager = natural TN fp32-accum two-pass; hip =
    the production pre-transposed bf16-D GEMM + fused fp32 accumulate
    (includes its transpose costs)."""
    from lpp_amd import ops

    ext = ops.extension()
    x = torch.randn(T, IN, device=DEV, dtype=torch.bfloat16)
    dy = torch.randn(T, OUT, device=DEV, dtype=torch.bfloat16)
    dw = torch.zeros(OUT, IN, device=DEV, dtype=torch.float32)

    def hip():
        xT = ext.transpose2d(x)
        dyT = ext.transpose2d(dy)
        ext.accum_bf16_f32(dw.view(-1), torch.matmul(dyT, xT.t()).view(-1))

    def ref():
        dw.add_(torch.matmul(dy.t().float(), x.float()))

    t_hip = timeit(hip, iters=10)
    t_ref = timeit(ref, iters=3)
    return [(f"wgrad[{T}x{IN}->{OUT}]", t_hip, t_ref, 0)]


def bench_adamw(n=1_000_000_000 // 4):
    from lpp_amd import ops

    ext = ops.extension()
    n = 256 * 1024 * 1024  # 256M params
    p = torch.zeros(n, device=DEV, dtype=torch.bfloat16)
    master = torch.zeros(n, device=DEV, dtype=torch.float32)
    g = torch.randn(n, device=DEV, dtype=torch.float32)
    m = torch.zeros_like(master)
    v = torch.zeros_like(master)
    bytes_moved = n * (4 * 4 + 4 * 3 + 2)  # r: g,m,v,w; w: m,v,w,p
    t_hip = timeit(
        lambda: ext.fused_adamw([p], [master], [g], [m], [v], 1e-4, 0.9, 0.99, 1e-6,
                                0.001, 0.1, 0.1, 1.0),
        iters=10,
    )
    # eager: foreach pipeline equivalent
    def ref():
        torch._foreach_mul_([m], 0.9)
        torch._foreach_add_([m], [g], alpha=0.1)
        torch._foreach_mul_([v], 0.99)
        torch._foreach_addcmul_([v], [g], [g], value=0.01)
        denom = torch._foreach_sqrt([v])
        torch._foreach_div_(denom, 0.3)
        torch._foreach_add_(denom, 1e-6)
        torch._foreach_addcdiv_([master], [m], denom, value=-1e-4)
        p.copy_(master)

    t_ref = timeit(ref, iters=10)
    return [("fused_adamw_256M", t_hip, t_ref, bytes_moved)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--csv", type=str, default=None)
    args = ap.parse_args()
    torch.cuda.set_device(0)
    rows = []
    for fn in [bench_rmsnorm, bench_rmsnorm_bwd, bench_rope, bench_swiglu, bench_ce,
               bench_adamw, bench_attention, bench_transpose, bench_wgrad]:
        try:
            rows += fn()
        except Exception as e:
            print(f"{fn.__name__}: FAILED {type(e).__name__}: {e}")
    print(f"\n{'op':24s} {'hip ms':>9s} {'eager ms':>9s} {'speedup':>8s} {'hip GB/s':>9s}")
    lines = ["op,hip_ms,eager_ms,speedup,hip_GBps"]
    for name, t_hip, t_ref, bytes_moved in rows:
        bw = bytes_moved / (t_hip / 1000) / 1e9 if bytes_moved else 0
        print(f"{name:24s} {t_hip:9.3f} {t_ref:9.3f} {t_ref / t_hip:8.2f} {bw:9.0f}")
        lines.append(f"{name},{t_hip:.4f},{t_ref:.4f},{t_ref / t_hip:.2f},{bw:.0f}")
    if args.csv:
        with open(args.csv, "w") as f:
            f.write("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
