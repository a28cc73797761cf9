"""Shared helpers to run a tiny training job single- or multi-process."""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader

from lpp_amd.config import TrainConfig, model_config
from lpp_amd.data import CausalLMCollator, RepeatingLoader, SyntheticCausalLMDataset
from lpp_amd.engine import PipelineEngine
from lpp_amd.models import get_layers_from_config, init_pipeline_weights, loss_fn
from lpp_amd.pipeline_module import PipelineModule
from lpp_amd.topology import ProcessGrid


def make_config(num_stages=1, gas=4, mbs=2, seq=32, dtype="fp32", lr=1e-3):
    # 8 decoder layers: every stage owns >=1 spec up to PP=8, and the model
    # is identical regardless of the stage count (trajectory comparisons)
    cfg = TrainConfig(
        model=model_config("llama-tiny", num_layers=8),
        num_stages=num_stages,
        micro_batch_size=mbs,
        gradient_accumulation_steps=gas,
        seq_len=seq,
        dtype=dtype,
        activation_checkpoint_interval=1,
        backend="gloo",
    )
    cfg.optimizer.lr = lr
    cfg.optimizer.warmup_steps = 2
    cfg.optimizer.total_num_steps = 100
    return cfg


def sequential_loader(cfg, dp_degree=1, dp_id=0, n_examples=64):
    ds = SyntheticCausalLMDataset(n_examples, cfg.seq_len, cfg.model.vocab_size, seed=99)
    # deterministic order; shard across dp by contiguous stride
    idx = list(range(dp_id, n_examples, dp_degree))
    sub = torch.utils.data.Subset(ds, idx)
    loader = DataLoader(sub, batch_size=cfg.micro_batch_size, shuffle=False,
                        collate_fn=CausalLMCollator(cfg.seq_len), drop_last=True)
    return iter(RepeatingLoader(loader))


def build_engine(cfg, rank, world_size, seed=11):
    grid = ProcessGrid(world_size, rank, cfg.num_stages)
    grid.build_groups()
    module = PipelineModule(
        get_layers_from_config(cfg.model),
        grid,
        loss_fn=loss_fn,
        activation_checkpoint_interval=cfg.activation_checkpoint_interval,
    )
    init_pipeline_weights(module, cfg.model, seed=seed)
    return PipelineEngine(module, cfg, grid, device=torch.device("cpu"))


def run_steps(rank, world_size, num_stages, steps=3, gas=4, dtype="fp32", seed=11,
              p2p_overlap=True, overlap_allreduce=True):
    cfg = make_config(num_stages=num_stages, gas=gas, dtype=dtype)
    cfg.p2p_overlap = p2p_overlap
    cfg.overlap_allreduce = overlap_allreduce
    engine = build_engine(cfg, rank, world_size, seed=seed)
    dp = engine.grid.dp_degree
    dp_id = engine.grid.dp_id
    it = sequential_loader(cfg, dp_degree=dp, dp_id=dp_id)
    losses = []
    for _ in range(steps):
        loss = engine.train_batch(it)
        losses.append(float(loss))
    return losses
