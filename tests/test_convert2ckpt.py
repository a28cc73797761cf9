"""Checkpoint conversion + HF numerics parity.

The HF-parity test is the strongest correctness oracle in the suite: a tiny
HF LlamaForCausalLM's weights are exported in the HF on-disk format,
converted with convert2ckpt.py, loaded per-stage into our PipelineModule,
and the logits must match transformers' own forward pass.
"""

import sys
from pathlib import Path

import pytest
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import convert2ckpt  # noqa: E402
from lpp_amd.checkpoint import load_module_weights, read_latest  # noqa: E402
from lpp_amd.config import model_config  # noqa: E402
from lpp_amd.models import (  # noqa: E402
    LlamaForCausalLM,
    get_layers_from_config,
    init_monolithic_weights,
    loss_fn,
)
from lpp_amd.pipeline_module import PipelineModule  # noqa: E402
from lpp_amd.topology import ProcessGrid  # noqa: E402


def test_random_init_convert_round_trip(tmp_path):
    convert2ckpt.convert_random("llama-tiny", tmp_path, None, seed=5)
    assert read_latest(str(tmp_path)) == "global_step001"
    cfg = model_config("llama-tiny")
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(cfg), grid, loss_fn=loss_fn)
    load_module_weights(module, str(tmp_path))
    mono = LlamaForCausalLM(cfg)
    init_monolithic_weights(mono, seed=5)
    for p1, p2 in zip(module.parameters(), mono.parameters()):
        assert torch.equal(p1, p2)


def test_sharded_stage_load(tmp_path):
    """BASELINE config #5 shape: per-stage sharded load — each stage maps
    only its own layer files."""
    convert2ckpt.convert_random("llama-tiny", tmp_path, None, seed=6)
    cfg = model_config("llama-tiny")
    mono = LlamaForCausalLM(cfg)
    init_monolithic_weights(mono, seed=6)
    x = torch.randint(0, cfg.vocab_size, (1, 16))
    ref = mono(x)
    # run the two stages sequentially in-process
    h = x
    for stage in range(2):
        grid = ProcessGrid(2, stage, 2)
        module = PipelineModule(get_layers_from_config(cfg), grid, loss_fn=loss_fn)
        load_module_weights(module, str(tmp_path))
        with torch.no_grad():
            h = module(h)
    assert torch.allclose(h, ref, atol=1e-5)


@pytest.fixture(scope="module")
def hf_tiny(tmp_path_factory):
    transformers = pytest.importorskip("transformers")
    from transformers import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaForCausalLM as HFLlama

    hf_cfg = LlamaConfig(
        vocab_size=256,
        hidden_size=64,
        intermediate_size=176,
        num_hidden_layers=4,
        num_attention_heads=4,
        num_key_value_heads=4,
        max_position_embeddings=256,
        rms_norm_eps=1e-6,
        tie_word_embeddings=False,
    )
    torch.manual_seed(123)
    model = HFLlama(hf_cfg).eval()
    d = tmp_path_factory.mktemp("hf_tiny")
    torch.save(model.state_dict(), d / "pytorch_model.bin")
    return model, d


def test_hf_parity_logits(hf_tiny, tmp_path):
    """Converted HF weights produce HF-identical logits through our stack."""
    hf_model, hf_dir = hf_tiny
    convert2ckpt.convert_hf(hf_dir, tmp_path, pad_vocab_to=0, dtype=None)

    cfg = model_config("llama-tiny")  # same geometry as the HF config above
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(cfg), grid, loss_fn=loss_fn)
    load_module_weights(module, str(tmp_path))
    module.eval()

    x = torch.randint(0, cfg.vocab_size, (2, 32))
    with torch.no_grad():
        ours = module(x)
        theirs = hf_model(x).logits
    assert torch.allclose(ours, theirs, atol=2e-4), (ours - theirs).abs().max()


def test_hf_parity_loss(hf_tiny, tmp_path):
    hf_model, hf_dir = hf_tiny
    convert2ckpt.convert_hf(hf_dir, tmp_path, pad_vocab_to=0, dtype=None)
    cfg = model_config("llama-tiny")
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(get_layers_from_config(cfg), grid, loss_fn=loss_fn)
    load_module_weights(module, str(tmp_path))
    module.eval()
    x = torch.randint(0, cfg.vocab_size, (2, 32))
    with torch.no_grad():
        ours = loss_fn(module(x), x.clone())
        theirs = hf_model(x, labels=x.clone()).loss
    assert abs(float(ours) - float(theirs)) < 1e-4


def test_pad_vocab(tmp_path):
    convert2ckpt.convert_random("llama-tiny", tmp_path / "a", None, seed=1)
    import torch as t

    sd = t.load(tmp_path / "a" / "global_step001" / "layer_00-model_00-model_states.pt",
                weights_only=True)
    assert sd["weight"].shape[0] == 256
    padded = convert2ckpt._pad_rows(sd["weight"], 300)
    assert padded.shape[0] == 300
    assert float(padded[256:].abs().sum()) == 0.0


def test_hf_sharded_bins_and_safetensors(hf_tiny, tmp_path):
    """A 65B convert streams sharded checkpoints; sharded .bin files and
    .safetensors must produce the identical layer files as a single bin."""
    hf_model, hf_dir = hf_tiny
    sd = hf_model.state_dict()

    # sharded bins
    keys = sorted(sd)
    shards = [keys[i::3] for i in range(3)]
    bdir = tmp_path / "sharded"
    bdir.mkdir()
    for i, ks in enumerate(shards):
        torch.save({k: sd[k] for k in ks},
                   bdir / f"pytorch_model-{i + 1:05d}-of-00003.bin")
    out_b = tmp_path / "out_b"
    convert2ckpt.convert_hf(bdir, out_b, pad_vocab_to=0, dtype=None)

    # safetensors
    st = pytest.importorskip("safetensors.torch")
    sdir = tmp_path / "safet"
    sdir.mkdir()
    st.save_file({k: v.contiguous() for k, v in sd.items()},
                 str(sdir / "model.safetensors"))
    out_s = tmp_path / "out_s"
    convert2ckpt.convert_hf(sdir, out_s, pad_vocab_to=0, dtype=None)

    # reference: single-bin convert
    out_ref = tmp_path / "out_ref"
    convert2ckpt.convert_hf(hf_dir, out_ref, pad_vocab_to=0, dtype=None)

    for out in (out_b, out_s):
        ref_files = sorted((out_ref / "global_step001").glob("layer_*.pt"))
        got_files = sorted((out / "global_step001").glob("layer_*.pt"))
        assert [f.name for f in got_files] == [f.name for f in ref_files]
        for rf, gf in zip(ref_files, got_files):
            a = torch.load(rf, weights_only=True)
            b = torch.load(gf, weights_only=True)
            assert sorted(a) == sorted(b), rf.name
            for k in a:
                assert torch.equal(a[k], b[k]), (rf.name, k)
