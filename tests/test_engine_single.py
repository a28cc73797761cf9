"""Single-process (PP=1) engine: the training loop itself, no collectives."""

import torch

from tests.engine_utils import build_engine, make_config, sequential_loader


def test_loss_decreases_overfit():
    cfg = make_config(num_stages=1, gas=2, lr=3e-3)
    engine = build_engine(cfg, rank=0, world_size=1)
    it = sequential_loader(cfg, n_examples=4)  # tiny set -> overfit fast
    losses = [float(engine.train_batch(it)) for _ in range(8)]
    assert losses[-1] < losses[0] * 0.9, losses


def test_grads_cleared_and_steps_counted():
    cfg = make_config(num_stages=1, gas=2)
    engine = build_engine(cfg, rank=0, world_size=1)
    it = sequential_loader(cfg)
    engine.train_batch(it)
    assert engine.global_steps == 1
    assert float(engine.optimizer.flat_grads.abs().sum()) == 0.0
    # no leftover bf16 .grad on params
    for p in engine.module.parameters():
        assert p.grad is None


def test_lr_warmup_applies():
    cfg = make_config(num_stages=1, gas=2)
    engine = build_engine(cfg, rank=0, world_size=1)
    it = sequential_loader(cfg)
    lr0 = engine.get_lr()
    engine.train_batch(it)
    assert engine.get_lr() > lr0  # warming up


def test_bf16_dtype_runs():
    cfg = make_config(num_stages=1, gas=2, dtype="bf16")
    engine = build_engine(cfg, rank=0, world_size=1)
    assert next(engine.module.parameters()).dtype == torch.bfloat16
    it = sequential_loader(cfg)
    loss = engine.train_batch(it)
    assert torch.isfinite(torch.tensor(float(loss)))


def test_fp16_loss_scaler_runs():
    cfg = make_config(num_stages=1, gas=2, dtype="fp16")
    engine = build_engine(cfg, rank=0, world_size=1)
    assert engine.loss_scaler is not None
    it = sequential_loader(cfg)
    loss = engine.train_batch(it)
    assert torch.isfinite(torch.tensor(float(loss)))
