"""End-to-end real-corpus training: convert -> train on jsonl -> resume
(VERDICT round-1 item 4; reference trainer_base_ds_mp.py:142-200,317)."""

import json
import os

import pytest
import torch
import yaml

from tests.dist_utils import run_dist


def _write_corpus(path, n=64):
    rows = [{"inputs": f"question number {i} about topic {i % 7}",
             "targets": f"the answer is {i * 3}"} for i in range(n)]
    with open(path, "w") as f:
        for r in rows:
            f.write(json.dumps(r) + "\n")
    return str(path)


def _write_cfg(tmp_path, corpus, **kw):
    cfg = {
        "model": {"name": "llama-tiny"},
        "num_stages": 1,
        "micro_batch_size": 2,
        "gradient_accumulation_steps": 2,
        "seq_len": 32,
        "dtype": "fp32",
        "seed": 3,
        "max_steps": 2,
        "save_steps": 2,
        "logging_steps": 1,
        "output_dir": str(tmp_path / "out"),
        "backend": "gloo",
        "data_kind": "jsonl",
        "train_file": corpus,
        "tokenizer_path": "simple",
        "optimizer": {"lr": 1e-3, "total_num_steps": 10},
    }
    cfg.update(kw)
    p = tmp_path / "cfg.yaml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f)
    return str(p)


def test_jsonl_convert_train_resume(tmp_path):
    """Full loop: random-init converted checkpoint -> warm start -> train on
    a jsonl corpus -> checkpoint -> resume and extend."""
    import convert2ckpt as c2c
    from lpp_amd.checkpoint import read_latest
    from lpp_amd.trainer import main as trainer_main

    ckpt = tmp_path / "converted"
    c2c.convert_random("llama-tiny", ckpt, None, seed=0)
    corpus = _write_corpus(tmp_path / "corpus.jsonl")

    cfg_path = _write_cfg(tmp_path, corpus, model_name_or_path=str(ckpt))
    assert trainer_main(["--config", cfg_path]) == 0
    out = tmp_path / "out"
    assert read_latest(str(out)) == "global_step2"

    cfg2 = _write_cfg(tmp_path, corpus, model_name_or_path=str(ckpt),
                      max_steps=4, resume=str(out / "global_step2"))
    assert trainer_main(["--config", cfg2]) == 0
    assert read_latest(str(out)) == "global_step4"


def test_jsonl_loss_decreases(tmp_path):
    """The tiny model actually learns the toy corpus through the full stack."""
    from lpp_amd.config import TrainConfig
    from lpp_amd.trainer import build_dataset, train
    from tests.engine_utils import build_engine

    corpus = _write_corpus(tmp_path / "c.jsonl", n=32)
    cfg = TrainConfig.load(_write_cfg(tmp_path, corpus, max_steps=30,
                                      num_train_epochs=5,
                                      save_steps=0, logging_steps=0))
    cfg.optimizer.lr = 5e-3
    engine = build_engine(cfg, 0, 1)
    losses = []
    orig = engine.train_batch

    def spy(it):
        loss = orig(it)
        losses.append(float(loss))
        return loss

    engine.train_batch = spy
    train(cfg, engine)
    assert len(losses) == 30
    assert sum(losses[-5:]) / 5 < sum(losses[:5]) / 5 - 0.5, losses


def _pp2_jsonl(rank, world, corpus, outdir):
    from lpp_amd.config import TrainConfig
    from lpp_amd.trainer import train
    from tests.engine_utils import build_engine

    cfg = TrainConfig.from_dict({
        "model": {"name": "llama-tiny"},
        "num_stages": world,
        "micro_batch_size": 2,
        "gradient_accumulation_steps": 2,
        "seq_len": 32,
        "dtype": "fp32",
        "seed": 3,
        "max_steps": 2,
        "save_steps": 0,
        "logging_steps": 0,
        "backend": "gloo",
        "data_kind": "jsonl",
        "train_file": corpus,
        "tokenizer_path": "simple",
        "output_dir": os.path.join(outdir, f"r{rank}"),
        "optimizer": {"lr": 1e-3, "total_num_steps": 10},
    })
    engine = build_engine(cfg, rank, world)
    out = train(cfg, engine)
    return out["steps"]


def test_jsonl_pp2_step_count_broadcast(tmp_path):
    """Middle/last stages learn the corpus length via the rank-0 broadcast
    (quirk Q3) and the rank-0-first build keeps barrier counts matched."""
    corpus = _write_corpus(tmp_path / "c.jsonl", n=48)
    got = run_dist(2, _pp2_jsonl, corpus, str(tmp_path), timeout=120.0)
    assert got == [2, 2]


def test_rank_zero_first_single_process():
    from lpp_amd.utils import rank_zero_first

    with rank_zero_first(0):
        x = 1
    assert x == 1
