import pytest

from lpp_amd.config import MODEL_PRESETS, TrainConfig, model_config


def test_param_counts_match_published_sizes():
    # LLaMA-1 sizes: 6.7B, 13.0B, 32.5B, 65.2B
    assert abs(model_config("llama-7b").num_params() - 6.74e9) / 6.74e9 < 0.02
    assert abs(model_config("llama-13b").num_params() - 13.0e9) / 13.0e9 < 0.02
    assert abs(model_config("llama-65b").num_params() - 65.2e9) / 65.2e9 < 0.02


def test_preset_overrides():
    m = model_config("llama-7b", num_layers=4)
    assert m.num_layers == 4 and m.hidden_size == 4096
    with pytest.raises(KeyError):
        model_config("llama-9000b")


def test_yaml_round_trip(tmp_path):
    cfg = TrainConfig(model=model_config("llama-tiny"), num_stages=2, seq_len=128)
    cfg.optimizer.lr = 3e-4
    p = tmp_path / "cfg.yaml"
    cfg.save(str(p))
    cfg2 = TrainConfig.load(str(p))
    assert cfg2.model.name == "llama-tiny"
    assert cfg2.num_stages == 2
    assert cfg2.seq_len == 128
    assert cfg2.optimizer.lr == 3e-4
    assert tuple(cfg2.optimizer.betas) == (0.9, 0.99)


def test_unknown_key_rejected():
    with pytest.raises(ValueError):
        TrainConfig.from_dict({"definitely_not_a_key": 1})


def test_global_batch():
    cfg = TrainConfig(micro_batch_size=8, gradient_accumulation_steps=256)
    assert cfg.global_batch_size(dp_degree=2) == 4096  # reference derived batch


def test_all_shipped_configs_load():
    """Every conf/*.yaml parses onto the dataclasses and is self-consistent."""
    import glob

    from lpp_amd.config import TrainConfig, model_config

    files = sorted(glob.glob("conf/*.yaml"))
    assert len(files) >= 6
    for f in files:
        cfg = TrainConfig.load(f)
        m = cfg.model
        assert m.hidden_size % m.num_heads == 0
        assert cfg.seq_len <= m.max_seq_len
        assert cfg.num_stages >= 1
        # named preset exists and matches the file's geometry family
        model_config(m.name)
