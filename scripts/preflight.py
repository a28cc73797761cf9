#!/usr/bin/env python
"""Cluster preflight: validate the distributed stack on a fresh node in
seconds before committing to a long 65B run.

Runs a tiny model through the REAL engine paths — RCCL p2p channels (with
overlap), DP bucket all-reduce, grad-norm/optimizer collectives, device
timers, checkpoint save/load — at the requested world size, and prints a
per-rank OK with the device-true section split.

Launch exactly like the real job:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 scripts/preflight.py [--stages 8]
Single process (no args) also works (CPU or one GPU).
"""

from __future__ import annotations

import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

from lpp_amd.checkpoint import load_engine_checkpoint, save_engine_checkpoint
from lpp_amd.config import TrainConfig, model_config, torch_dtype
from lpp_amd.data import CausalLMCollator, RepeatingLoader, SyntheticCausalLMDataset
from lpp_amd.engine import PipelineEngine
from lpp_amd.models import get_layers_from_config, init_pipeline_weights, loss_fn
from lpp_amd.pipeline_module import PipelineModule
from lpp_amd.topology import ProcessGrid
from lpp_amd.utils import init_distributed, set_seed


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--stages", type=int, default=0, help="0 = world size")
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--ckpt-dir", default="/tmp/lpp_preflight_ckpt")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        init_distributed()
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", rank)))
    else:
        device = torch.device("cpu")

    stages = args.stages or max(world, 1)
    mcfg = model_config("llama-tiny", num_layers=max(2 * stages, 4))
    cfg = TrainConfig(
        model=mcfg, num_stages=stages, micro_batch_size=2,
        gradient_accumulation_steps=max(2 * stages, 4), seq_len=64,
        dtype="bf16" if on_gpu else "fp32",
        watchdog_timeout_s=120.0,
    )
    cfg.optimizer.lr = 1e-3

    set_seed(7, rank)
    grid = ProcessGrid(max(world, 1), rank, stages)
    grid.build_groups()
    module = PipelineModule(
        get_layers_from_config(mcfg), grid, loss_fn=loss_fn, device=device,
        dtype=torch_dtype(cfg.dtype),
    )
    init_pipeline_weights(module, mcfg, seed=7)
    t0 = time.time()
    engine = PipelineEngine(module, cfg, grid, device=device)  # warms comms
    t_init = time.time() - t0

    ds = SyntheticCausalLMDataset(256, 64, mcfg.vocab_size, seed=1)
    loader = torch.utils.data.DataLoader(
        ds, batch_size=2, collate_fn=CausalLMCollator(64), drop_last=True)
    it = iter(RepeatingLoader(loader))

    t0 = time.time()
    losses = [float(engine.train_batch(it)) for _ in range(args.steps)]
    t_train = time.time() - t0
    assert all(l == l for l in losses), f"rank {rank}: NaN loss {losses}"

    save_engine_checkpoint(engine, args.ckpt_dir, tag="preflight")
    load_engine_checkpoint(engine, args.ckpt_dir, tag="preflight")

    t = engine.timer_summary()
    print(f"[preflight rank {rank}/{world} stage {grid.stage_id}] OK — "
          f"init {t_init:.1f}s, {args.steps} steps {t_train:.1f}s, "
          f"loss {losses[0]:.3f}->{losses[-1]:.3f}, "
          f"fwd {t['forward']:.2f}s bwd {t['backward']:.2f}s "
          f"p2p {t['p2p']:.2f}s ar {t['allreduce']:.2f}s", flush=True)
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
