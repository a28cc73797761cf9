"""Trainer loop: config-driven run, save, resume (single process CPU)."""

import os

import torch
import yaml

from lpp_amd.checkpoint import read_latest
from lpp_amd.trainer import main as trainer_main


def _write_cfg(tmp_path, **kw):
    cfg = {
        "model": {"name": "llama-tiny"},
        "num_stages": 1,
        "micro_batch_size": 2,
        "gradient_accumulation_steps": 2,
        "seq_len": 32,
        "dtype": "fp32",
        "seed": 1,
        "max_steps": 4,
        "save_steps": 2,
        "logging_steps": 1,
        "total_dataset_len": 64,
        "output_dir": str(tmp_path / "out"),
        "backend": "gloo",
        "optimizer": {"lr": 1e-3, "total_num_steps": 10},
    }
    cfg.update(kw)
    p = tmp_path / "cfg.yaml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f)
    return str(p)


def test_trainer_runs_and_saves(tmp_path):
    cfg_path = _write_cfg(tmp_path)
    assert trainer_main(["--config", cfg_path]) == 0
    out = tmp_path / "out"
    assert (out / "training_config.yaml").exists()
    assert read_latest(str(out)) == "global_step4"
    assert (out / "global_step4" / "layer_00-model_00-model_states.pt").exists()


def test_trainer_resume(tmp_path):
    cfg_path = _write_cfg(tmp_path)
    trainer_main(["--config", cfg_path])
    # resume from step-4 checkpoint, extend to 6 steps
    cfg_path2 = _write_cfg(
        tmp_path, max_steps=6, resume=str(tmp_path / "out" / "global_step4")
    )
    assert trainer_main(["--config", cfg_path2]) == 0
    assert read_latest(str(tmp_path / "out")) == "global_step6"


def test_trainer_cli_overrides(tmp_path):
    cfg_path = _write_cfg(tmp_path, max_steps=1, save_steps=0)
    assert trainer_main(["--config", cfg_path, "optimizer.lr=5e-4", "max_steps=2"]) == 0


def test_save_hook_cmd(tmp_path):
    """Post-save shell hook runs with {dir} substituted (reference s5cmd
    sync slot, trainer_base_ds_mp.py:220)."""
    from lpp_amd.trainer import main

    marker = tmp_path / "hook_ran"
    out = tmp_path / "out"
    rc = main([
        "--config", "conf/llama_7b_pp2_cpu.yaml",
        "num_stages=1", "model.name=llama-tiny", "model.num_layers=2",
        "micro_batch_size=2", "gradient_accumulation_steps=2", "seq_len=32",
        "max_steps=2", "save_steps=2", "logging_steps=1", "total_dataset_len=16",
        f"output_dir={out}",
        f"save_hook_cmd=echo {{dir}} > {marker}",
    ])
    assert rc == 0
    assert marker.exists()
    assert "global_step2" in marker.read_text()
