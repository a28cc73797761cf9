"""Regression tests for the round-2 ADVICE/VERDICT fixes:

- eval_batch is collective: middle stages must enter it (ADVICE high,
  lpp_amd/trainer.py) — a PP=3 gloo run with eval_steps=1 must terminate.
- completion_labels masks padding by POSITION so the terminating EOS keeps
  loss when pad_token == eos_token (ADVICE medium).
- prompt_len is the longest common prefix of standalone-prompt and
  concatenated encodings (ADVICE low: BPE boundary merges).
- tie_word_embeddings=True fails loudly (VERDICT weak #2).
- convert2ckpt --mp_world_size N writes one mp_rank_XX file per rank
  (VERDICT missing #6; reference convert2ckpt.py:16,38-48).
- watchdog: bad LPP_WATCHDOG_S env does not crash engine construction and
  stop() terminates the thread (ADVICE low).
"""

import os

import pytest
import torch

from tests.dist_utils import run_dist


# ---------------------------------------------------------------- eval
def _train_with_eval(rank, world, outdir):
    from tests.engine_utils import make_config, build_engine
    from lpp_amd.trainer import train

    cfg = make_config(num_stages=world, gas=2)
    cfg.max_steps = 2
    cfg.eval_steps = 1
    cfg.eval_micro_batches = 2
    cfg.total_dataset_len = 32
    cfg.save_steps = 0
    cfg.logging_steps = 0
    cfg.output_dir = os.path.join(outdir, f"r{rank}")
    engine = build_engine(cfg, rank, world)
    out = train(cfg, engine)
    return out["steps"]


def test_eval_steps_pp3_terminates(tmp_path):
    """With eval enabled and >=3 stages, every rank (including the middle
    stage, which holds no data iterator) must participate in eval_batch or
    the pipeline deadlocks at the first eval step."""
    got = run_dist(3, _train_with_eval, str(tmp_path), timeout=120.0)
    assert got == [2, 2, 2]


def _train_eval_resume(rank, world, outdir, resume):
    """Training twice with eval enabled: the training-data stream must be
    identical whether or not a resume fast-forward happened (ADVICE low:
    eval draws must not come from the training iterator)."""
    from tests.engine_utils import make_config, build_engine
    from lpp_amd.trainer import train

    cfg = make_config(num_stages=world, gas=2)
    cfg.optimizer.lr = 0.0  # freeze weights: loss identifies the data drawn
    cfg.max_steps = 4
    cfg.eval_steps = 2
    cfg.eval_micro_batches = 2
    cfg.total_dataset_len = 64
    cfg.save_steps = 0
    cfg.logging_steps = 0
    cfg.output_dir = os.path.join(outdir, f"rr{rank}")
    engine = build_engine(cfg, rank, world)
    losses = []
    orig = engine.train_batch

    def spy(it):
        loss = orig(it)
        losses.append(float(loss))
        return loss

    engine.train_batch = spy
    train(cfg, engine, resume_step=2 if resume else 0)
    return losses


def test_eval_does_not_shift_training_stream(tmp_path):
    full = run_dist(2, _train_eval_resume, str(tmp_path / "a"), False, timeout=120.0)
    resumed = run_dist(2, _train_eval_resume, str(tmp_path / "b"), True, timeout=120.0)
    # resumed run skips steps 1-2 then must see the same batches for steps 3-4
    assert full[0][2:] == pytest.approx(resumed[0], abs=1e-6)


# ------------------------------------------------------- label masking
def test_completion_labels_eos_kept_when_pad_is_eos():
    from lpp_amd.data.text import completion_labels

    eos = 2
    # row: [prompt prompt completion eos pad pad] with pad id == eos id
    ids = torch.tensor([[5, 6, 7, eos, eos, eos]])
    labels = completion_labels(ids, prompt_lens=torch.tensor([2]),
                               pad_token_id=eos, lengths=torch.tensor([4]))
    assert labels[0].tolist() == [-100, -100, 7, eos, -100, -100]


def test_completion_labels_id_mask_without_lengths():
    from lpp_amd.data.text import completion_labels

    ids = torch.tensor([[5, 6, 7, 2, 0, 0]])
    labels = completion_labels(ids, prompt_lens=torch.tensor([2]), pad_token_id=0)
    assert labels[0].tolist() == [-100, -100, 7, 2, -100, -100]


class _MergingTokenizer:
    """Tokenizer stub where the last prompt token merges with the first
    completion token in the concatenated encoding (BPE boundary merge)."""

    name_or_path = "stub"
    pad_token = "<pad>"
    pad_token_id = 0
    eos_token = "</s>"
    eos_token_id = 2
    bos_token = "<s>"
    unk_token = "<unk>"

    def add_special_tokens(self, special_tokens_dict):
        return 0

    def __call__(self, texts, max_length=None, padding="longest", truncation=True,
                 return_tensors="pt", add_special_tokens=True):
        rows = []
        for t in texts:
            if t == "ab c":                # standalone prompt
                rows.append([10, 11])      # [ab, c]
            else:                          # concatenated "ab c d</s>": c+d merge
                rows.append([10, 99, 2])   # [ab, cd-merged, eos]
        L = max(len(r) for r in rows)
        ids = torch.full((len(rows), L), self.pad_token_id, dtype=torch.long)
        mask = torch.zeros((len(rows), L), dtype=torch.long)
        for i, r in enumerate(rows):
            ids[i, : len(r)] = torch.tensor(r)
            mask[i, : len(r)] = 1
        return {"input_ids": ids, "attention_mask": mask}


def test_prompt_len_stops_at_boundary_merge():
    from lpp_amd.data.text import Seq2SeqToCausalLM

    conv = Seq2SeqToCausalLM(_MergingTokenizer(), max_seq_length=8)
    out = conv([{"inputs": "ab c", "targets": "d"}])
    # common prefix is just [10]; the merged token counts as completion
    assert int(out["prompt_lens"][0]) == 1
    assert int(out["lengths"][0]) == 3


# ------------------------------------------------------------ dead knobs
def test_tie_word_embeddings_fails_loudly():
    from lpp_amd.config import model_config
    from lpp_amd.models import get_layers_from_config

    cfg = model_config("llama-tiny", tie_word_embeddings=True)
    with pytest.raises(ValueError, match="tie_word_embeddings"):
        get_layers_from_config(cfg)


# ------------------------------------------------------------ converter
def test_convert2ckpt_mp_world_size(tmp_path):
    import convert2ckpt as c2c

    out = tmp_path / "ckpt"
    c2c.convert_random("llama-tiny", out, None, seed=0, mp_world_size=4)
    step = out / "global_step001"
    for r in range(4):
        f = step / f"mp_rank_{r:02d}_model_states.pt"
        assert f.exists(), f
        meta = torch.load(f, weights_only=True)
        assert meta["mp_world_size"] == 4
    assert not (step / "mp_rank_04_model_states.pt").exists()


# ------------------------------------------------------------- watchdog
def test_watchdog_env_defensive_and_stop(monkeypatch):
    from tests.engine_utils import make_config, build_engine

    monkeypatch.setenv("LPP_WATCHDOG_S", "not-a-number")
    cfg = make_config(num_stages=1, gas=1)
    cfg.watchdog_timeout_s = 5.0
    engine = build_engine(cfg, 0, 1)  # must not raise
    assert engine.watchdog is not None
    engine.watchdog.stop()
    assert not engine.watchdog._thread.is_alive()
