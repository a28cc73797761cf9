"""Training loop + CLI — native equivalent of trainer_base_ds_mp.py.

Capabilities mirrored from the reference (file:line cites into
/root/reference/trainer_base_ds_mp.py):
- distributed init + device pinning (:397-399)
- engine + module build from config, per-stage weight warm start (:280-299)
- per-(stage, dp) dataloaders; only first/last stage load data (:309-336)
- step-count agreement across ranks — computed once from config instead of
  per-rank len(dataloader)//gas (quirk Q3)
- resume with dataloader fast-forward (:345-351)
- periodic checkpoint save + latest tag (:203-224, 367-371)
- rank-0 logging of {loss, lr}; wandb if available (:360-374,441-447)

CLI:  python -m lpp_amd.trainer --config conf/llama_65b_pp8.yaml [k=v ...]
"""

from __future__ import annotations

import argparse
import logging
import os
import sys
import time
from typing import Optional

import torch
import torch.distributed as dist

from .checkpoint import load_engine_checkpoint, parse_checkpoint_step, save_engine_checkpoint
from .config import TrainConfig, torch_dtype
from .data import CausalLMCollator, RepeatingLoader, SyntheticCausalLMDataset, build_loader
from .engine import PipelineEngine
from .models import get_layers_from_config, init_pipeline_weights, loss_fn
from .pipeline_module import PipelineModule
from .topology import ProcessGrid
from .utils import init_distributed, set_seed

logger = logging.getLogger("lpp_amd.trainer")


def build_tokenizer(cfg: TrainConfig):
    """Tokenizer for the jsonl path: a HF tokenizer directory when given (or
    found inside model_name_or_path, as convert2ckpt passes tokenizer files
    through), else the offline SimpleTokenizer.  Either way
    expand_special_tokenizer applies (reference trainer_base_ds_mp.py:416-420)."""
    from .data import SimpleTokenizer, expand_special_tokenizer

    path = cfg.tokenizer_path
    if not path and cfg.model_name_or_path and any(
        os.path.exists(os.path.join(cfg.model_name_or_path, f))
        for f in ("tokenizer.json", "tokenizer.model", "tokenizer_config.json")
    ):
        path = cfg.model_name_or_path
    if path and path != "simple":
        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(path)
    else:
        tok = SimpleTokenizer(cfg.model.vocab_size)
    expand_special_tokenizer(tok)
    return tok


def build_dataset(cfg: TrainConfig):
    """Config-selected (dataset, collator) — the native version of the
    reference's Hydra dataset/collator instantiation
    (trainer_base_ds_mp.py:142-200,317).  Returns (dataset, collator)."""
    if cfg.data_kind == "synthetic":
        n = cfg.total_dataset_len or 4096
        ds = SyntheticCausalLMDataset(n, cfg.seq_len, cfg.model.vocab_size,
                                      seed=cfg.seed,
                                      pattern=getattr(cfg, "data_pattern", "uniform"))
        return ds, CausalLMCollator(cfg.seq_len)
    if cfg.data_kind == "jsonl":
        from .data import PromptResponseDataset, TextCollator

        if not cfg.train_file:
            raise ValueError("data_kind=jsonl requires train_file")
        tok = build_tokenizer(cfg)
        ds = PromptResponseDataset(cfg.train_file)
        return ds, TextCollator(tok, cfg.seq_len, field=cfg.data_field)
    raise ValueError(f"unknown data_kind {cfg.data_kind!r} (synthetic | jsonl)")


def train(cfg: TrainConfig, engine: PipelineEngine, resume_step: int = 0) -> dict:
    grid = engine.grid
    rank0 = grid.rank == 0

    wandb = None
    if rank0 and os.environ.get("WANDB_MODE", "") not in ("", "disabled", "offline-off"):
        try:
            import wandb as _wandb

            _wandb.init(project=os.environ.get("WANDB_PROJECT", "lpp-amd"),
                        config=cfg.to_dict())
            wandb = _wandb
        except Exception:
            wandb = None

    needs_data = grid.is_first_stage() or grid.is_last_stage()
    dataset = collator = None
    # rank-0-first build (reference barrier discipline,
    # trainer_base_ds_mp.py:163-176): rank 0 pays any tokenizer/cache cost
    # once; the other data-holding ranks hit the warm cache.
    from .utils import rank_zero_first

    with rank_zero_first(grid.rank):
        if needs_data:
            dataset, collator = build_dataset(cfg)

    # quirk Q3 fix: every rank uses the IDENTICAL example count — from
    # config when pinned, else measured on rank 0 and broadcast (middle
    # stages hold no dataset and must not guess).
    n_examples = cfg.total_dataset_len or (len(dataset) if dataset is not None else 0)
    if dist.is_initialized():
        obj = [n_examples]
        dist.broadcast_object_list(obj, src=0)
        n_examples = int(obj[0])
    n_examples = n_examples or 4096
    examples_per_step = cfg.micro_batch_size * cfg.gradient_accumulation_steps * grid.dp_degree
    steps_per_epoch = n_examples // examples_per_step
    total_steps = cfg.max_steps or steps_per_epoch * cfg.num_train_epochs
    cfg.optimizer.total_num_steps = max(cfg.optimizer.total_num_steps, total_steps)
    engine.lr_scheduler.total_num_steps = cfg.optimizer.total_num_steps

    if rank0:
        logger.info(
            "training: %d steps (%d/epoch), global batch %d, grid %s",
            total_steps, steps_per_epoch, examples_per_step, grid,
        )
        os.makedirs(cfg.output_dir, exist_ok=True)
        cfg.save(os.path.join(cfg.output_dir, "training_config.yaml"))

    step = 0
    tr_loss = 0.0
    t_start = time.time()
    done = False
    for epoch in range(cfg.num_train_epochs):
        if done:
            break
        it = None
        eval_it = None
        if needs_data:
            loader = build_loader(
                dataset, cfg.micro_batch_size, grid.dp_degree, grid.dp_id,
                seed=cfg.seed, num_workers=cfg.num_workers,
                collator=collator, epoch=epoch,
            )
            it = iter(RepeatingLoader(loader))
            if cfg.eval_steps:
                # A SEPARATE iterator for eval: drawing eval microbatches from
                # the training iterator would desynchronize the resume
                # fast-forward (which drains exactly gas draws per step).
                eval_loader = build_loader(
                    dataset, cfg.micro_batch_size, grid.dp_degree, grid.dp_id,
                    seed=cfg.seed + 7919, num_workers=0,
                    collator=collator, epoch=epoch,
                )
                eval_it = iter(RepeatingLoader(eval_loader))
        for _ in range(steps_per_epoch):
            if step >= total_steps:
                done = True
                break
            if step < resume_step:
                # fast-forward: drain the sampler without compute
                # (reference :345-351)
                if needs_data:
                    for _ in range(cfg.gradient_accumulation_steps):
                        next(it)
                step += 1
                continue
            loss = engine.train_batch(it)
            step += 1
            tr_loss += float(loss)
            if cfg.eval_steps and step % cfg.eval_steps == 0:
                # eval_batch is collective across the pipe group (p2p +
                # loss broadcast): EVERY rank must enter it.  Middle stages
                # pass eval_it=None — they never touch the iterator.
                ev = float(engine.eval_batch(eval_it, cfg.eval_micro_batches))
                if rank0:
                    logger.info("eval @ step %d: loss %.4f", step, ev)
                    if wandb:
                        wandb.log({"eval_loss": ev}, step=step)
            if cfg.logging_steps and step % cfg.logging_steps == 0:
                # fold + reset device timers on EVERY rank (bounds the event
                # backlog); only rank 0 logs its split
                t = engine.timer_summary(reset=True)
            if rank0 and cfg.logging_steps and step % cfg.logging_steps == 0:
                avg = tr_loss / cfg.logging_steps
                tr_loss = 0.0
                msg = {
                    "step": step,
                    "loss": round(avg, 4),
                    "lr": engine.get_lr(),
                    "s/step": round(engine.last_step_time, 3),
                    "fwd_s": round(t["forward"], 2),
                    "bwd_s": round(t["backward"], 2),
                    "p2p_s": round(t["p2p"], 2),
                    "allreduce_s": round(t["allreduce"], 2),
                    "optim_s": round(t["optimizer"], 2),
                }
                logger.info("%s", msg)
                if wandb:
                    wandb.log(msg, step=step)
            if cfg.save_steps and step % cfg.save_steps == 0:
                save_engine_checkpoint(
                    engine, cfg.output_dir, tag=f"global_step{step}",
                    client_state={"step": step},
                )
                if rank0 and cfg.save_hook_cmd:
                    # reference parity: post-save sync hook (s5cmd, :220)
                    import subprocess

                    cmd = cfg.save_hook_cmd.format(
                        dir=os.path.join(cfg.output_dir, f"global_step{step}"))
                    r = subprocess.run(cmd, shell=True)
                    if r.returncode != 0:
                        logger.warning("save_hook_cmd failed (%d): %s", r.returncode, cmd)
    final = {"steps": step, "elapsed": time.time() - t_start}
    if cfg.save_steps:
        save_engine_checkpoint(engine, cfg.output_dir, tag=f"global_step{step}",
                               client_state={"step": step})
    if wandb:
        wandb.finish()
    return final


def main(argv: Optional[list] = None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=str, required=True)
    ap.add_argument("overrides", nargs="*", help="key=value config overrides")
    args = ap.parse_args(argv)

    cfg = TrainConfig.load(args.config)
    for ov in args.overrides:
        k, _, v = ov.partition("=")
        obj = cfg
        parts = k.split(".")
        for p in parts[:-1]:
            obj = getattr(obj, p)
        cur = getattr(obj, parts[-1])
        setattr(obj, parts[-1], type(cur)(v) if cur is not None else v)

    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(name)s [%(levelname)s] %(message)s",
    )

    rank, world = init_distributed(backend=None if torch.cuda.is_available() else "gloo")
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    set_seed(cfg.seed, rank)
    grid = ProcessGrid(world, rank, cfg.num_stages)
    grid.build_groups()

    # selective per-layer activation checkpointing: on 288 GB MI355X most
    # stages keep full activations; checkpoint_layers_per_stage >= 0 pins
    # how many layers each stage recomputes (-1 = classic interval rules)
    ckpt_fn = None
    interval = cfg.activation_checkpoint_interval if cfg.activation_checkpointing else 0
    if cfg.checkpoint_layers_per_stage >= 0:
        lps = max(1, cfg.model.num_layers // cfg.num_stages)
        kps = cfg.checkpoint_layers_per_stage

        def ckpt_fn(i: int) -> bool:
            return (i % lps) < kps

        interval = 0
    module = PipelineModule(
        get_layers_from_config(cfg.model, checkpoint_fn=ckpt_fn),
        grid,
        loss_fn=loss_fn,
        activation_checkpoint_interval=interval,
        device=device,
        dtype=torch_dtype(cfg.dtype),
    )
    init_pipeline_weights(module, cfg.model, seed=cfg.seed)
    engine = PipelineEngine(module, cfg, grid, device=device)

    resume_step = 0
    if cfg.model_name_or_path and os.path.isdir(cfg.model_name_or_path):
        # module-only warm start from a converted HF checkpoint (:284)
        load_engine_checkpoint(engine, cfg.model_name_or_path, load_module_only=True)
    if cfg.resume:
        load_engine_checkpoint(engine, cfg.resume)
        resume_step = parse_checkpoint_step(cfg.resume) or engine.global_steps

    train(cfg, engine, resume_step=resume_step)
    if dist.is_initialized():
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
