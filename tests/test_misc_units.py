"""Unit coverage for small pieces: watchdog firing, rope offsets, inference
guards, layer_spec math, data patterns."""

import time

import pytest
import torch

from lpp_amd.config import model_config


def test_watchdog_fires(caplog, monkeypatch):
    """A stuck step gets a schedule-position dump (SURVEY §5.2 rebuild)."""
    import logging

    from lpp_amd.engine import _Watchdog

    class FakeGrid:
        rank = 3
        stage_id = 1

    class FakeEngine:
        grid = FakeGrid()
        schedule_position = "steady 1F1B 7/16"
        global_steps = 5
        _pending_dbg = (0, 1)

    wd = _Watchdog(FakeEngine(), timeout_s=0.2)
    with caplog.at_level(logging.ERROR, logger="lpp_amd.engine"):
        wd.arm()
        time.sleep(1.2)
    wd.disarm()
    assert any("watchdog" in r.message and "steady 1F1B 7/16" in r.message
               for r in caplog.records)


def test_rope_offset_matches_slice():
    """apply_rope(pos_offset=k) == applying the table rows k..k+S."""
    from lpp_amd.ops.rope import apply_rope_ref, build_rope_cache

    cos, sin = build_rope_cache(64, 16, 10000.0, torch.device("cpu"))
    x = torch.randn(1, 8, 2, 16)
    a = apply_rope_ref(x, cos, sin, pos_offset=5)
    b = apply_rope_ref(x, cos[5:13].contiguous(), sin[5:13].contiguous())
    assert torch.allclose(a, b, atol=1e-6)


def test_generate_rejects_overlong():
    from lpp_amd.models import LlamaForCausalLM
    from lpp_amd.inference import pipeline_generate
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid
    from lpp_amd.models import get_layers_from_config, loss_fn

    mcfg = model_config("llama-tiny", num_layers=1, max_seq_len=16)
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(mcfg), grid, loss_fn=loss_fn,
                            device=torch.device("cpu"), dtype=torch.float32)
    ids = torch.randint(0, 256, (1, 10))
    with pytest.raises(ValueError, match="max_seq_len"):
        pipeline_generate(module, grid, ids, max_new_tokens=10)


def test_kv_cache_append():
    from lpp_amd.models import KVCache

    c = KVCache(2, 10, 4, 8, torch.device("cpu"), torch.float32)
    k = torch.ones(2, 3, 4, 8)
    assert c.append(k, k) == 3
    assert c.append(k[:, :2], k[:, :2]) == 5
    assert (c.k[:, :3] == 1).all() and (c.k[:, 5:] == 0).all()


def test_arith_pattern_is_predictable():
    from lpp_amd.data import SyntheticCausalLMDataset

    ds = SyntheticCausalLMDataset(4, 16, 997, pattern="arith")
    ex = ds[2]
    ids = ex["input_ids"]
    d = (ids[1:] - ids[:-1]) % 997
    assert (d == d[0]).all()  # constant step mod V
    # deterministic per index
    assert torch.equal(ids, ds[2]["input_ids"])


def test_partition_rejects_too_many_stages():
    from lpp_amd.layer_spec import partition_balanced

    with pytest.raises(ValueError):
        partition_balanced([1, 1, 1], 4)


def test_step_timer():
    from lpp_amd.utils import StepTimer

    t = StepTimer()
    t.start()
    time.sleep(0.01)
    dt = t.stop()
    assert dt >= 0.01 and t.mean >= 0.01


def test_device_timers_cpu_path():
    from lpp_amd.utils.timers import DeviceTimers
    import time as _t

    t = DeviceTimers(None)
    with t.section("a"):
        _t.sleep(0.01)
    with t.section("a"):
        pass
    with t.section("b"):
        pass
    out = t.summary(reset=True)
    assert out["a"] >= 0.01
    assert "b" in out
    assert t.totals_nosync() == {}


def _rz_body(rank, world):
    import time as _t
    from lpp_amd.utils import rank_zero_first

    with rank_zero_first(rank):
        t_enter = _t.time()
        if rank == 0:
            _t.sleep(0.3)
        t_exit = _t.time()
    return t_enter, t_exit


def test_rank_zero_first_ordering():
    """Rank 0 must complete the body before any other rank enters it."""
    from tests.dist_utils import run_dist

    got = run_dist(2, _rz_body)
    rank0_exit = got[0][1]
    rank1_enter = got[1][0]
    assert rank1_enter >= rank0_exit - 0.05, (got,)
