"""Checkpoint save/load in the convert2ckpt on-disk layout.

The layout contract (produced by the reference's
convert2ckpt.py:19-48, consumed via the monkey-patched engine loader at
trainer_base_ds_mp.py:49-121,284):

    <dir>/latest                                   text: e.g. "global_step001"
    <dir>/global_stepNNN/
        layer_{i:02d}-model_00-model_states.pt     state_dict of flat layer i
        mp_rank_00_model_states.pt                 engine metadata dict
        engine_state_pp{stage:02d}_dp{dp:02d}.pt   (ours) optimizer/scheduler
                                                   shards — absent in a
                                                   converted-from-HF dir

Layer file numbering == flat LayerSpec index (embedding 0, decoder i -> i+1,
norm L+1, head L+2) — matching convert2ckpt.py:23-36 so converted HF
checkpoints load per-stage.  Module-only loads (a dir with no engine_state
files) work natively — the very capability the reference had to monkey-patch
into DeepSpeed (trainer_base_ds_mp.py:48, README.md:163).
"""

from __future__ import annotations

import logging
import os
import re
from pathlib import Path
from typing import Optional

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

LAYER_FMT = "layer_{idx:02d}-model_00-model_states.pt"
MP_RANK_FMT = "mp_rank_{mp:02d}_model_states.pt"
ENGINE_FMT = "engine_state_pp{stage:02d}_dp{dp:02d}.pt"
LATEST = "latest"


def layer_file(idx: int) -> str:
    return LAYER_FMT.format(idx=idx)


def normalize_ckpt_path(ckpt_dir: str, tag: Optional[str]) -> tuple[str, Optional[str]]:
    """Accept either the checkpoint root (with a ``latest`` tag file) or a
    step dir directly (``.../global_step4`` — how the reference's cfg.resume
    names it, trainer_base_ds_mp.py:452-455)."""
    p = Path(ckpt_dir)
    if tag is None and not (p / LATEST).exists() and any(p.glob("layer_*-model_*.pt")):
        return str(p.parent), p.name
    return ckpt_dir, tag


def read_latest(ckpt_dir: str) -> Optional[str]:
    p = Path(ckpt_dir) / LATEST
    if not p.exists():
        return None
    tag = p.read_text().strip()
    return tag or None


def write_latest(ckpt_dir: str, tag: str) -> None:
    (Path(ckpt_dir) / LATEST).write_text(tag)


def save_engine_checkpoint(engine, ckpt_dir: str, tag: Optional[str] = None,
                           client_state: Optional[dict] = None) -> str:
    """All ranks call this collectively.  dp_id==0 ranks write their stage's
    layer files; every rank writes its engine-state shard; global rank 0
    writes metadata + latest."""
    grid = engine.grid
    if tag is None:
        tag = f"global_step{engine.global_steps}"
    step_dir = Path(ckpt_dir) / tag
    step_dir.mkdir(parents=True, exist_ok=True)

    module = engine.module
    if grid.dp_id == 0:
        for local_idx, layer in enumerate(module.layers):
            gidx = module.global_layer_index(local_idx)
            sd = {k: v.detach().cpu() for k, v in layer.state_dict().items()}
            torch.save(sd, step_dir / layer_file(gidx))

    es = engine.state_dict_local()
    if client_state:
        es["client_state"] = client_state
    torch.save(es, step_dir / ENGINE_FMT.format(stage=grid.stage_id, dp=grid.dp_id))

    if grid.rank == 0:
        meta = {
            "dp_world_size": grid.dp_degree,
            "mp_world_size": 1,
            "num_stages": grid.num_stages,
            "module": None,
            "optimizer": None,
            "global_steps": engine.global_steps,
            "skipped_steps": engine.skipped_steps,
            "iteration": engine.global_steps,
        }
        torch.save(meta, step_dir / MP_RANK_FMT.format(mp=0))
        write_latest(ckpt_dir, tag)

    if dist.is_initialized():
        dist.barrier()
    return str(step_dir)


def load_module_weights(module, ckpt_dir: str, tag: Optional[str] = None,
                        strict: bool = True, dtype: Optional[torch.dtype] = None) -> str:
    """Module-only warm start: map this stage's layer files onto local layers.
    Works on converted-HF dirs (no optimizer state present)."""
    ckpt_dir, tag = normalize_ckpt_path(ckpt_dir, tag)
    tag = tag or read_latest(ckpt_dir)
    if tag is None:
        raise FileNotFoundError(f"no 'latest' tag in {ckpt_dir}")
    step_dir = Path(ckpt_dir) / tag
    for local_idx, layer in enumerate(module.layers):
        gidx = module.global_layer_index(local_idx)
        f = step_dir / layer_file(gidx)
        if not f.exists():
            raise FileNotFoundError(f"missing layer file {f}")
        sd = torch.load(f, map_location="cpu", weights_only=True)
        if dtype is not None:
            sd = {k: (v.to(dtype) if v.is_floating_point() else v) for k, v in sd.items()}
        missing, unexpected = layer.load_state_dict(sd, strict=strict)
        if strict and (missing or unexpected):
            raise KeyError(f"layer {gidx}: missing={missing} unexpected={unexpected}")
    # move to module device/dtype handled by caller
    return str(step_dir)


def load_engine_checkpoint(engine, ckpt_dir: str, tag: Optional[str] = None,
                           load_module_only: bool = False) -> Optional[dict]:
    """Full resume (module + optimizer + scheduler) or module-only warm start
    (load_module_only=True — the reference's load path at
    trainer_base_ds_mp.py:284 with the same flag)."""
    grid = engine.grid
    ckpt_dir, tag = normalize_ckpt_path(ckpt_dir, tag)
    tag = tag or read_latest(ckpt_dir)
    if tag is None:
        raise FileNotFoundError(f"no 'latest' tag in {ckpt_dir}")
    load_module_weights(engine.module, ckpt_dir, tag, dtype=engine.dtype)
    engine.module.to(engine.device)
    # refresh masters from (re)loaded params (mode-aware: ZeRO-1 keeps one
    # flat master shard, not per-param copies)
    engine.optimizer.refresh_masters()
    from .ops.linear import invalidate_weight_transposes

    invalidate_weight_transposes(engine.module)

    client_state = None
    if not load_module_only:
        step_dir = Path(ckpt_dir) / tag
        f = step_dir / ENGINE_FMT.format(stage=grid.stage_id, dp=grid.dp_id)
        es = None
        if f.exists():
            es = torch.load(f, map_location="cpu", weights_only=False)
            if es["optimizer"].get("shard_world", 1) != engine.optimizer.shard_world:
                es = None  # dp-degree changed: reshard below
        if es is not None:
            engine.load_state_dict_local(es)
            client_state = es.get("client_state")
        else:
            old = sorted(step_dir.glob(
                f"engine_state_pp{grid.stage_id:02d}_dp*.pt"))
            if old:
                client_state = _load_resharded(engine, old)
            else:
                logger.warning(
                    "no engine state shard at %s — module-only load (converted "
                    "checkpoint?)", f)
    if dist.is_initialized():
        dist.barrier()
    return client_state


def _load_resharded(engine, shard_files) -> Optional[dict]:
    """Resume optimizer state saved at a DIFFERENT dp_degree (ZeRO-1 shards
    or replicated plain-DP state): reconstruct the full flat fp32 state from
    every saved shard of this stage, then cut this engine's view of it —
    plain per-param copies, or this rank's 1/dp shard (with new padding).
    The reference stack cannot do this (DeepSpeed ZeRO resume is
    world-size-pinned)."""
    sds = [torch.load(f, map_location="cpu", weights_only=False) for f in shard_files]
    base = sds[0]
    opt_sds = [sd["optimizer"] for sd in sds]
    old_world = opt_sds[0].get("shard_world", 1)
    opt = engine.optimizer
    total = sum(p.numel() for p in opt.params)
    if old_world > 1 and len(sds) != old_world:
        raise FileNotFoundError(
            f"ZeRO-1 reshard needs all {old_world} saved shards of this stage, "
            f"found {len(sds)}: {[str(f) for f in shard_files]}")
    logger.info("resharding optimizer state: saved dp=%d -> engine dp shard_world=%d",
                old_world, opt.shard_world)

    def full_vec(key: str) -> torch.Tensor:
        if old_world == 1:
            parts = [t.reshape(-1).float() for t in opt_sds[0][key]]
            return torch.cat(parts)[:total]
        cat = torch.cat([osd[key][0].reshape(-1).float() for osd in opt_sds])
        return cat[:total]  # strip the old dp padding

    with torch.no_grad():
        for key, dsts in (("masters", opt.masters), ("exp_avg", opt.exp_avg),
                          ("exp_avg_sq", opt.exp_avg_sq)):
            vec = full_vec(key)
            if opt.is_sharded:
                padded = torch.zeros(opt.padded_total, dtype=torch.float32)
                padded[:total] = vec
                dsts[0].copy_(padded[opt.shard_slice].to(dsts[0].device))
            else:
                off = 0
                for p, dst in zip(opt.params, dsts):
                    n = p.numel()
                    dst.copy_(vec[off:off + n].view_as(p).to(dst.device))
                    off += n
        # re-sync model params from the (re)built masters
        if opt.is_sharded:
            opt.param_shard.copy_(opt.masters[0])
            opt._maybe_allgather_params()
        else:
            for p, m in zip(opt.params, opt.masters):
                p.data.copy_(m.to(p.dtype))
    opt.step_count = opt_sds[0]["step_count"]
    opt.lr = opt_sds[0].get("lr", opt.lr)
    engine.lr_scheduler.load_state_dict(base["lr_scheduler"])
    engine.global_steps = base.get("global_steps", 0)
    engine.skipped_steps = base.get("skipped_steps", 0)
    if engine.loss_scaler is not None and base.get("loss_scale"):
        engine.loss_scaler.scale = base["loss_scale"]
    return base.get("client_state")


def parse_checkpoint_step(path: str) -> int:
    """checkpoint-500 / global_step500 -> 500 (resume-step parsing the
    reference does on the dir name, trainer_base_ds_mp.py:452-455)."""
    m = re.search(r"(?:checkpoint-|global_step)(\d+)", os.path.basename(os.path.normpath(path)))
    return int(m.group(1)) if m else 0
