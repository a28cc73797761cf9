"""Exact decisions of the MI355X schedule policy at the benchmark shapes
(these values are what the driver's 1/2/4/8-GPU runs will use)."""

from lpp_amd.config import model_config
from lpp_amd.utils.schedule import choose_schedule


def test_65b_decisions_per_depth():
    expect = {1: (4, 16, 0), 2: (4, 32, 0), 4: (2, 64, 0), 8: (1, 128, 0)}
    for stages, (mbs, gas, ckpt) in expect.items():
        m = model_config("llama-65b", num_layers=10 * stages, max_seq_len=4096)
        s = choose_schedule(m, stages, 10, 4096)
        assert (s.micro_batch_size, s.gas, s.ckpt_layers_per_stage) == (mbs, gas, ckpt), (
            stages, s)


def test_llama3_8k_checkpoints_some_layers():
    m = model_config("llama3-70b", num_layers=80, max_seq_len=8192)
    s = choose_schedule(m, 8, 10, 8192)
    # 8k sequences double activation bytes: some recompute expected at PP8
    assert s.micro_batch_size == 1
    assert 0 < s.ckpt_layers_per_stage <= 10


def test_explicit_flags_respected():
    m = model_config("llama-65b", num_layers=10, max_seq_len=4096)
    s = choose_schedule(m, 1, 10, 4096, micro_batch_size=2, gas=8)
    assert s.micro_batch_size == 2 and s.gas == 8
