#!/usr/bin/env python
"""GEMM formulation ceilings at the LLaMA-65B training shapes.

For each projection shape and microbatch T, times the three GEMM roles on
hipBLASLt (via torch.matmul / our fp32-accum wgrad):
  fwd   (NT): y[T,out]  = x[T,in]  @ W[out,in]^T     bf16 D
  dgrad (NN): dx[T,in]  = dy[T,out] @ W[out,in]      bf16 D
  wgrad (TN): dW[out,in] = dy[T,out]^T @ x[T,in]     bf16 D (probe)
  wgrad-f32  same, fp32 D beta=1 (the production fused-accum path)

Output: TF/s per role — quantifies how much of the wgrad deficit is the TN
layout vs the fp32-D epilogue, and what a hand kernel must beat.
"""

from __future__ import annotations

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lpp_amd import ops

H, I, V = 8192, 22016, 32000
SHAPES = [("qkv_o", H, H), ("gate_up", H, I), ("down", I, H), ("lm_head", H, V)]


def timeit(fn, reps=10):
    fn()
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(reps):
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        fn()
        t1.record()
        torch.cuda.synchronize()
        best = min(best, t0.elapsed_time(t1))
    return best


def main():
    assert torch.cuda.is_available()
    ext = ops.extension()
    for T in (4096, 16384):
        for name, cin, cout in SHAPES:
            x = torch.randn(T, cin, device="cuda", dtype=torch.bfloat16)
            dy = torch.randn(T, cout, device="cuda", dtype=torch.bfloat16)
            w = torch.randn(cout, cin, device="cuda", dtype=torch.bfloat16)
            dw32 = torch.zeros(cout, cin, device="cuda", dtype=torch.float32)
            flops = 2.0 * T * cin * cout

            def tf(ms):
                return flops / (ms * 1e-3) / 1e12

            fwd = timeit(lambda: torch.matmul(x, w.t()))
            dgrad = timeit(lambda: torch.matmul(dy, w))
            wg_bf16 = timeit(lambda: torch.matmul(dy.t(), x))
            wg_f32 = timeit(lambda: ext.wgrad_f32_accum(x, dy, dw32))
            # pre-transposed operands turn the wgrad into the k-contiguous
            # class (same as fwd): dW[out,in] = dyT[out,T] @ xT[in,T]^T
            xT = x.t().contiguous()
            dyT = dy.t().contiguous()
            wg_pre = timeit(lambda: torch.matmul(dyT, xT.t()))
            tx = timeit(lambda: x.t().contiguous())
            tdy = timeit(lambda: dy.t().contiguous())
            xbytes = 2 * 2.0 * T * cin / 1e9
            dybytes = 2 * 2.0 * T * cout / 1e9
            print(f"[{name} T={T} in={cin} out={cout}] "
                  f"fwd(NT) {tf(fwd):5.0f}  dgrad(NN) {tf(dgrad):5.0f}  "
                  f"wgradTN(bf16D) {tf(wg_bf16):5.0f}  wgradTN(f32D,b1) {tf(wg_f32):5.0f}  "
                  f"wgradPRE(bf16D) {tf(wg_pre):5.0f} TF/s  "
                  f"| torch-transpose x {tx:6.3f} ms ({xbytes / (tx * 1e-3):4.0f} GB/s) "
                  f"dy {tdy:6.3f} ms ({dybytes / (tdy * 1e-3):4.0f} GB/s)",
                  flush=True)


if __name__ == "__main__":
    main()
