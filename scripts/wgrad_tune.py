#!/usr/bin/env python
"""Exhaustive hipBLASLt solution sweep for the fp32-accum wgrad GEMMs.

The wgrad classes (bf16 A/B, fp32 C/D, beta=1, TN layout) ran 15-25% below
the bf16 forward/dgrad GEMM classes with heuristic-48 selection
(profiles/r01_65b_1gpu_step_kernel_stats_final.csv; VERDICT.md next-round
item 3).  This sweeps EVERY library solution supported for each production
wgrad shape (hipblaslt_ext::getAllAlgos), device-times them, re-times the
top candidates carefully, and writes the winners to
lpp_amd/ops/wgrad_algos.json (pinned at import on GPU).

Run on the GPU box:
    python scripts/wgrad_tune.py [--out lpp_amd/ops/wgrad_algos.json]

Shapes: LLaMA-65B projections at the bench microbatch sizes
(T = mbs * 4096 for mbs 1, 2, 4 — pp8 runs mbs1, pp4 mbs2, pp1/2 mbs4).
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from lpp_amd import ops  # noqa: E402

H, I, V = 8192, 22016, 32000

SHAPES = [  # (name, in, out)
    ("qkv_o", H, H),
    ("gate_up", H, I),
    ("down", I, H),
    ("lm_head", H, V),
]


def time_current(ext, T, cin, cout, reps=5, kind=0) -> float:
    """Time the currently-pinned plan through the real entry point."""
    if kind == 0:
        x = torch.randn(T, cin, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(T, cout, device="cuda", dtype=torch.bfloat16)
        run = lambda: ext.wgrad_f32_accum(x, dy, dw)  # noqa: E731
    else:
        x = torch.randn(cin, T, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(cout, T, device="cuda", dtype=torch.bfloat16)
        run = lambda: ext.wgrad_f32_accum_pre(x, dy, dw)  # noqa: E731
    dw = torch.zeros(cout, cin, device="cuda", dtype=torch.float32)
    run()  # warm (also triggers heuristic pick)
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(reps):
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        run()
        t1.record()
        torch.cuda.synchronize()
        best = min(best, t0.elapsed_time(t1))
    return best


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="lpp_amd/ops/wgrad_algos.json")
    ap.add_argument("--t-values", type=int, nargs="*", default=[4096, 8192, 16384])
    ap.add_argument("--topk", type=int, default=8, help="candidates to re-time")
    ap.add_argument("--kinds", type=int, nargs="*", default=[0, 1],
                    help="0 = natural TN layout, 1 = pre-transposed k-contiguous")
    args = ap.parse_args()

    assert torch.cuda.is_available(), "run on the GPU box"
    ext = ops.extension()

    results = {"tuned_on": torch.version.hip, "device": torch.cuda.get_device_name(0),
               "shapes": []}
    for kind in args.kinds:
        for T in args.t_values:
            for name, cin, cout in SHAPES:
                flops = 2.0 * T * cin * cout
                t_start = time.time()
                # heuristic baseline through the real path
                base_ms = time_current(ext, T, cin, cout, kind=kind)
                base_idx, base_name = ext.wgrad_current_algo(T, cin, cout, kind)
                # exhaustive sweep (1 rep triage)
                table = ext.wgrad_tune(T, cin, cout, 1, kind)
                # careful re-time of the top candidates
                best = (base_idx, base_ms, base_name)
                for idx, _, kname in table[: args.topk]:
                    ext.wgrad_set_algo(T, cin, cout, idx, kind)
                    ms = time_current(ext, T, cin, cout, kind=kind)
                    if ms < best[1]:
                        best = (idx, ms, kname)
                ext.wgrad_set_algo(T, cin, cout, best[0], kind)
                tf = flops / (best[1] * 1e-3) / 1e12
                tf_base = flops / (base_ms * 1e-3) / 1e12
                print(f"[kind{kind} {name} T={T} in={cin} out={cout}] "
                      f"candidates={len(table)} "
                      f"heuristic {base_ms:.3f} ms ({tf_base:.0f} TF/s, idx {base_idx}) -> "
                      f"best {best[1]:.3f} ms ({tf:.0f} TF/s, idx {best[0]}) "
                      f"gain {100 * (base_ms / best[1] - 1):+.1f}%  "
                      f"[{time.time() - t_start:.0f}s]", flush=True)
                print(f"    kernel: {best[2][:110]}", flush=True)
                results["shapes"].append({
                    "name": name, "T": T, "in": cin, "out": cout, "kind": kind,
                    "index": int(best[0]), "ms": round(best[1], 4),
                    "tflops": round(tf, 1), "heuristic_ms": round(base_ms, 4),
                    "heuristic_index": int(base_idx), "kernel": best[2],
                })

    with open(args.out, "w") as f:
        json.dump(results, f, indent=1)
    print(f"wrote {args.out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
