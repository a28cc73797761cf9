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
