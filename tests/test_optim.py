"""MixedPrecisionAdamW (eager path) vs torch.optim.AdamW oracle."""

import torch

from lpp_amd.optim import MixedPrecisionAdamW, WarmupDecayLR


def _models():
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
    m2 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
    m2.load_state_dict(m1.state_dict())
    return m1, m2


def test_matches_torch_adamw_fp32():
    m1, m2 = _models()
    lr, wd = 1e-2, 0.01
    opt1 = MixedPrecisionAdamW(m1.parameters(), lr=lr, betas=(0.9, 0.99), eps=1e-6, weight_decay=wd)
    opt2 = torch.optim.AdamW(m2.parameters(), lr=lr, betas=(0.9, 0.99), eps=1e-6, weight_decay=wd)
    x = torch.randn(4, 8)
    for _ in range(5):
        loss1 = m1(x).square().mean()
        loss1.backward()  # hooks route grads into main_grad
        opt1.step()
        opt1.zero_grad()
        loss2 = m2(x).square().mean()
        loss2.backward()
        opt2.step()
        opt2.zero_grad()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_grad_accumulation_in_fp32_buffer():
    m1, _ = _models()
    opt = MixedPrecisionAdamW(m1.parameters(), lr=1e-3)
    x = torch.randn(4, 8)
    m1(x).sum().backward()
    g1 = opt.flat_grads.clone()
    m1(x).sum().backward()
    assert torch.allclose(opt.flat_grads, 2 * g1, atol=1e-6)
    for p in m1.parameters():
        assert p.grad is None  # freed by the hook


def test_grad_scale_applied():
    m1, m2 = _models()
    opt1 = MixedPrecisionAdamW(m1.parameters(), lr=1e-2, weight_decay=0.0)
    opt2 = MixedPrecisionAdamW(m2.parameters(), lr=1e-2, weight_decay=0.0)
    x = torch.randn(4, 8)
    m1(x).sum().backward()
    (m2(x).sum() * 2).backward()
    opt1.step(grad_scale=1.0)
    opt2.step(grad_scale=0.5)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_state_dict_round_trip():
    m1, _ = _models()
    opt = MixedPrecisionAdamW(m1.parameters(), lr=1e-2)
    x = torch.randn(4, 8)
    m1(x).sum().backward()
    opt.step()
    sd = opt.state_dict()
    m2 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
    opt2 = MixedPrecisionAdamW(m2.parameters(), lr=1e-2)
    opt2.load_state_dict(sd)
    for a, b in zip(opt.masters, opt2.masters):
        assert torch.equal(a, b)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2)


def test_warmup_decay_shape():
    m1, _ = _models()
    opt = MixedPrecisionAdamW(m1.parameters(), lr=1.0)
    sched = WarmupDecayLR(opt, warmup_num_steps=10, total_num_steps=100, warmup_max_lr=1.0)
    lrs = []
    for _ in range(100):
        lrs.append(opt.lr)
        sched.step()
    assert lrs[0] == 0.0
    assert abs(lrs[10] - 1.0) < 1e-9
    assert lrs[50] < 1.0
    assert lrs[99] < lrs[50]
    assert opt.lr >= 0.0


def test_bf16_params_fp32_master():
    m = torch.nn.Linear(8, 8).to(torch.bfloat16)
    opt = MixedPrecisionAdamW(m.parameters(), lr=1e-2)
    x = torch.randn(4, 8, dtype=torch.bfloat16)
    m(x).float().sum().backward()
    assert opt.flat_grads.dtype == torch.float32
    assert opt.masters[0].dtype == torch.float32
    opt.step()
    assert m.weight.dtype == torch.bfloat16


def test_dynamic_loss_scaler_schedule():
    """fp16 scaler: halve after hysteresis overflows, double after window
    good steps, floor at min_scale (conf/...yaml:137-143)."""
    from lpp_amd.engine import DynamicLossScaler

    s = DynamicLossScaler(init_scale=2.0**12, scale_window=4, hysteresis=2, min_scale=1.0)
    assert s.scale == 4096
    s.update(found_inf=True)       # hysteresis 2 -> first inf tolerated
    assert s.scale == 4096
    s.update(found_inf=True)       # second inf -> halve
    assert s.scale == 2048
    for _ in range(4):             # window good steps -> double
        s.update(found_inf=False)
    assert s.scale == 4096
    for _ in range(30):            # repeated overflow floors at min
        s.update(found_inf=True)
    assert s.scale == 1.0


def test_engine_skips_step_on_overflow():
    """Inf gradient -> skipped_steps increments, params untouched (fp16
    loss-scale skip semantics; engine._optimizer_step)."""
    import torch
    from lpp_amd.config import TrainConfig, model_config
    from lpp_amd.engine import DynamicLossScaler, PipelineEngine
    from lpp_amd.models import get_layers_from_config, loss_fn
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid

    mcfg = model_config("llama-tiny", num_layers=1, max_seq_len=32)
    cfg = TrainConfig(model=mcfg, num_stages=1, micro_batch_size=1,
                      gradient_accumulation_steps=1, seq_len=32, dtype="fp32")
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(mcfg), grid, loss_fn=loss_fn,
                            device=torch.device("cpu"), dtype=torch.float32)
    engine = PipelineEngine(module, cfg, grid, device=torch.device("cpu"))
    engine.loss_scaler = DynamicLossScaler()  # force the fp16 skip path
    engine.optimizer.flat_grads.fill_(float("inf"))
    before = [p.clone() for p in module.parameters()]
    engine._optimizer_step()
    assert engine.skipped_steps == 1
    for p0, p1 in zip(before, module.parameters()):
        assert torch.equal(p0, p1)
    # grads cleared so the next step starts clean
    assert engine.optimizer.flat_grads.abs().sum() == 0
