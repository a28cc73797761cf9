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


def test_watchdog_and_timers(monkeypatch):
    """Watchdog arms/disarms around train_batch; timers accumulate."""
    import lpp_amd.engine as eng
    from lpp_amd.config import TrainConfig, model_config
    from lpp_amd.data import CausalLMCollator, SyntheticCausalLMDataset
    from lpp_amd.models import get_layers_from_config, loss_fn
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid
    import torch

    monkeypatch.setenv("LPP_WATCHDOG_S", "300")
    mcfg = model_config("llama-tiny", num_layers=2, hidden_size=64, num_heads=4,
                        intermediate_size=128, vocab_size=128, max_seq_len=32)
    cfg = TrainConfig(model=mcfg, num_stages=1, micro_batch_size=2,
                      gradient_accumulation_steps=2, seq_len=32, dtype="fp32")
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(mcfg), grid, loss_fn=loss_fn,
                            device=torch.device("cpu"), dtype=torch.float32)
    engine = eng.PipelineEngine(module, cfg, grid, device=torch.device("cpu"))
    assert engine.watchdog is not None
    ds = SyntheticCausalLMDataset(8, 32, mcfg.vocab_size)
    loader = torch.utils.data.DataLoader(ds, batch_size=2,
                                         collate_fn=CausalLMCollator(32))
    loss = engine.train_batch(iter(loader))
    assert torch.isfinite(loss)
    assert engine.schedule_position == "idle"
    t = engine.timer_summary()
    assert t["forward"] > 0 and t["backward"] > 0 and t["optimizer"] >= 0
    # reset happened
    assert engine.timers["forward"] == 0.0


def test_eval_batch_forward_only():
    from lpp_amd.config import TrainConfig, model_config
    from lpp_amd.data import CausalLMCollator, RepeatingLoader, SyntheticCausalLMDataset
    from lpp_amd.engine import PipelineEngine
    from lpp_amd.models import get_layers_from_config, loss_fn
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid
    import torch

    mcfg = model_config("llama-tiny", num_layers=2, max_seq_len=32)
    cfg = TrainConfig(model=mcfg, num_stages=1, micro_batch_size=2,
                      gradient_accumulation_steps=2, seq_len=32, dtype="fp32")
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(mcfg), grid, loss_fn=loss_fn,
                            device=torch.device("cpu"), dtype=torch.float32)
    engine = PipelineEngine(module, cfg, grid, device=torch.device("cpu"))
    ds = SyntheticCausalLMDataset(16, 32, mcfg.vocab_size)
    it = iter(RepeatingLoader(torch.utils.data.DataLoader(
        ds, batch_size=2, collate_fn=CausalLMCollator(32))))
    before = [p.clone() for p in module.parameters()]
    ev = engine.eval_batch(it, micro_batches=3)
    assert torch.isfinite(ev) and ev > 0
    # eval must not touch parameters or leave grads
    for p0, p1 in zip(before, module.parameters()):
        assert torch.equal(p0, p1)
    assert engine.module.training
