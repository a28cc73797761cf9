"""Checkpoint layout + round trips (SURVEY.md §2.6 contract)."""

import os

import torch

from lpp_amd import checkpoint as ckpt
from tests.engine_utils import build_engine, make_config, sequential_loader


def test_layer_file_names():
    assert ckpt.layer_file(0) == "layer_00-model_00-model_states.pt"
    assert ckpt.layer_file(81) == "layer_81-model_00-model_states.pt"


def test_parse_checkpoint_step():
    assert ckpt.parse_checkpoint_step("/x/checkpoint-500") == 500
    assert ckpt.parse_checkpoint_step("/x/global_step120/") == 120
    assert ckpt.parse_checkpoint_step("/x/nope") == 0


def test_save_load_round_trip(tmp_path):
    cfg = make_config(num_stages=1, gas=2)
    engine = build_engine(cfg, 0, 1)
    it = sequential_loader(cfg)
    engine.train_batch(it)
    d = str(tmp_path)
    ckpt.save_engine_checkpoint(engine, d, tag="global_step1")
    assert ckpt.read_latest(d) == "global_step1"
    # layout: embedding + 4 layers + norm + head = 7 files
    files = sorted(os.listdir(tmp_path / "global_step1"))
    layer_files = [f for f in files if f.startswith("layer_")]
    assert len(layer_files) == cfg.model.num_layers + 3
    assert "mp_rank_00_model_states.pt" in files

    # perturb then reload -> exact restore (weights + optimizer)
    engine2 = build_engine(cfg, 0, 1, seed=999)
    ckpt.load_engine_checkpoint(engine2, d)
    for p1, p2 in zip(engine.module.parameters(), engine2.module.parameters()):
        assert torch.equal(p1, p2)
    assert engine2.optimizer.step_count == engine.optimizer.step_count
    for a, b in zip(engine.optimizer.exp_avg, engine2.optimizer.exp_avg):
        assert torch.equal(a, b)
    assert engine2.global_steps == engine.global_steps


def test_module_only_warm_start(tmp_path):
    """A dir with ONLY layer files (convert2ckpt output shape) must load —
    the capability the reference monkey-patched in (trainer_base_ds_mp.py:48)."""
    cfg = make_config(num_stages=1, gas=2)
    engine = build_engine(cfg, 0, 1)
    step_dir = tmp_path / "global_step001"
    step_dir.mkdir()
    for i, layer in enumerate(engine.module.layers):
        torch.save(layer.state_dict(), step_dir / ckpt.layer_file(i))
    ckpt.write_latest(str(tmp_path), "global_step001")

    engine2 = build_engine(cfg, 0, 1, seed=777)
    ckpt.load_engine_checkpoint(engine2, str(tmp_path), load_module_only=True)
    for p1, p2 in zip(engine.module.parameters(), engine2.module.parameters()):
        assert torch.equal(p1, p2)
    # training continues after a warm start
    it = sequential_loader(cfg)
    loss = engine2.train_batch(it)
    assert torch.isfinite(torch.tensor(float(loss)))


def test_resume_continues_identically(tmp_path):
    """save at step2 -> two more steps  ==  fresh engine load -> two steps."""
    cfg = make_config(num_stages=1, gas=2)
    e1 = build_engine(cfg, 0, 1)
    it1 = sequential_loader(cfg)
    for _ in range(2):
        e1.train_batch(it1)
    ckpt.save_engine_checkpoint(e1, str(tmp_path), tag="global_step2")
    ref = [float(e1.train_batch(it1)) for _ in range(2)]

    e2 = build_engine(cfg, 0, 1, seed=555)
    ckpt.load_engine_checkpoint(e2, str(tmp_path))
    it2 = sequential_loader(cfg)
    # fast-forward the loader by the 2 consumed steps (2 steps x gas=2)
    for _ in range(2 * cfg.gradient_accumulation_steps):
        next(it2)
    got = [float(e2.train_batch(it2)) for _ in range(2)]
    assert got == ref
