#!/usr/bin/env python
"""Continuous-vs-static batching serving microbench (1 GPU).

Workload: requests with mixed output lengths arriving together.  Static
batching pads every request to the longest generation (the whole batch
waits for the slowest); continuous batching retires each request at its
own eos/max and admits queued requests into the freed slots.

Usage (GPU box):  python scripts/serve_bench.py [--model llama-7b]
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lpp_amd.config import model_config
from lpp_amd.models import LlamaForCausalLM, init_monolithic_weights
from lpp_amd.serving import ContinuousBatchingEngine, Request


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-7b")
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--requests", type=int, default=32)
    ap.add_argument("--slots", type=int, default=8)
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--max-seq", type=int, default=1024)
    args = ap.parse_args()

    assert torch.cuda.is_available(), "run on the GPU box"
    cfg = model_config(args.model, num_layers=args.layers, max_seq_len=args.max_seq)
    m = LlamaForCausalLM(cfg).to("cuda").to(torch.bfloat16)
    init_monolithic_weights(m, seed=3)
    g = torch.Generator().manual_seed(7)
    # mixed output lengths: 16..256 tokens
    lens = [16 + (i * 37) % 241 for i in range(args.requests)]
    prompts = [torch.randint(4, cfg.vocab_size, (args.prompt_len,), generator=g).cuda()
               for _ in range(args.requests)]
    total_tokens = sum(lens)

    # ---- continuous batching: record per-request completion times
    eng = ContinuousBatchingEngine(m, max_slots=args.slots, max_seq_len=args.max_seq)
    for i, (p, n) in enumerate(zip(prompts, lens)):
        eng.submit(Request(f"r{i}", p, n))
    finish_c = {}
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    while eng.pending():
        for uid in eng.step():
            torch.cuda.synchronize()
            finish_c[uid] = time.perf_counter() - t0
    torch.cuda.synchronize()
    t_cont = time.perf_counter() - t0
    lat_c = sum(finish_c.values()) / len(finish_c)

    # ---- static batching: groups of `slots`, every request padded to the
    # group's longest generation; a request completes when its GROUP does
    finish_s = {}
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(0, args.requests, args.slots):
        grp = list(range(i, min(i + args.slots, args.requests)))
        ids = torch.stack([prompts[j] for j in grp])
        m.generate(ids, max_new_tokens=max(lens[j] for j in grp))
        torch.cuda.synchronize()
        tg = time.perf_counter() - t0
        for j in grp:
            finish_s[j] = tg
    t_stat = time.perf_counter() - t0
    lat_s = sum(finish_s.values()) / len(finish_s)
    padded = sum(max(lens[j] for j in range(i, min(i + args.slots, args.requests)))
                 * min(args.slots, args.requests - i)
                 for i in range(0, args.requests, args.slots)) - total_tokens

    print(f"requests={args.requests} slots={args.slots} prompt={args.prompt_len} "
          f"generated={total_tokens} tokens (mixed 16..256)")
    print(f"continuous: {t_cont:.2f} s = {total_tokens / t_cont:.0f} tok/s, "
          f"mean completion latency {lat_c:.2f} s")
    print(f"static    : {t_stat:.2f} s = {total_tokens / t_stat:.0f} tok/s, "
          f"mean completion latency {lat_s:.2f} s "
          f"(pays {padded} padded decode positions; short requests wait for "
          f"their group)")
    print(f"throughput ratio cont/static: {t_stat / t_cont:.2f}x; "
          f"latency ratio static/cont: {lat_s / lat_c:.2f}x")
    return 0


if __name__ == "__main__":
    sys.exit(main())
