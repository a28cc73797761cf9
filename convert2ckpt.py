#!/usr/bin/env python
"""HF LLaMA checkpoint -> per-layer pipeline checkpoint layout.

Native equivalent of the reference's convert2ckpt.py (:19-48 write_ckpt,
:51-80 main): writes

    <out>/latest                       ("global_step001")
    <out>/global_step001/layer_{i:02d}-model_00-model_states.pt
    <out>/global_step001/mp_rank_00_model_states.pt

with layer numbering embedding=0, decoder i -> i+1, final norm=L+1,
lm_head=L+2 and the "model.layers.{i}." key prefix stripped — exactly the
files lpp_amd.checkpoint.load_module_weights maps onto each pipeline stage.

Sources:
  --hf-dir DIR     a local HF LLaMA checkpoint (reads *.safetensors or
                   pytorch_model*.bin shards directly; no model class is
                   instantiated, so a 65B convert needs only one layer in
                   RAM at a time when sharded)
  --random-init M  fabricate a random-init checkpoint of preset M (e.g.
                   llama-7b) — used by tests and the no-network benchmark.

Vocab expansion: --pad-vocab-to N zero-pads embed/lm_head rows (the
reference resizes after expand_special_tokenizer, convert2ckpt.py:59-63).
"""

from __future__ import annotations

import argparse
import re
from pathlib import Path

import torch


def iter_hf_state_dict(hf_dir: Path):
    """Yield (key, tensor) from a local HF checkpoint dir, shard by shard."""
    safes = sorted(hf_dir.glob("*.safetensors"))
    if safes:
        from safetensors import safe_open

        for f in safes:
            with safe_open(str(f), framework="pt", device="cpu") as sf:
                for k in sf.keys():
                    yield k, sf.get_tensor(k)
        return
    bins = sorted(hf_dir.glob("pytorch_model*.bin"))
    if not bins:
        raise FileNotFoundError(f"no *.safetensors or pytorch_model*.bin in {hf_dir}")
    for f in bins:
        sd = torch.load(str(f), map_location="cpu", weights_only=True)
        yield from sd.items()


def write_ckpt(outdir: Path, num_layers: int, collect, pad_vocab_to: int = 0,
               dtype: torch.dtype | None = None, tag: str = "global_step001",
               mp_world_size: int = 1) -> None:
    """collect: dict mapping flat-layer-index -> state dict (built by caller).

    ``mp_world_size`` writes one ``mp_rank_{r:02d}_model_states.pt`` metadata
    file per (vestigial) model-parallel rank, as the reference converter does
    (convert2ckpt.py:16,38-48) — no tensor-parallel compute exists in either
    framework; the extra files are loader-compat metadata only."""
    step_dir = outdir / tag
    step_dir.mkdir(parents=True, exist_ok=True)
    for idx, sd in collect.items():
        if dtype is not None:
            sd = {k: (v.to(dtype) if v.is_floating_point() else v) for k, v in sd.items()}
        if pad_vocab_to and idx in (0, num_layers + 2):
            sd = {
                k: _pad_rows(v, pad_vocab_to) if v.dim() == 2 else v for k, v in sd.items()
            }
        torch.save(sd, step_dir / f"layer_{idx:02d}-model_00-model_states.pt")
    meta = {
        "dp_world_size": 1,
        "mp_world_size": mp_world_size,
        "module": None,
        "optimizer": None,
        "global_steps": 1,
        "skipped_steps": 1,
        "iteration": 1,
    }
    for r in range(max(mp_world_size, 1)):
        torch.save(meta, step_dir / f"mp_rank_{r:02d}_model_states.pt")
    (outdir / "latest").write_text(tag)


def _pad_rows(t: torch.Tensor, rows: int) -> torch.Tensor:
    if t.size(0) >= rows:
        return t
    pad = torch.zeros(rows - t.size(0), t.size(1), dtype=t.dtype)
    return torch.cat([t, pad], dim=0)


def copy_tokenizer_and_config(hf_dir: Path, outdir: Path) -> None:
    """Pass tokenizer/config files through to the checkpoint dir, as the
    reference converter does (convert2ckpt.py:79-80) so a training run can
    point model_name_or_path at one directory."""
    import shutil

    for name in ("config.json", "generation_config.json", "tokenizer.json",
                 "tokenizer.model", "tokenizer_config.json",
                 "special_tokens_map.json"):
        f = hf_dir / name
        if f.exists():
            shutil.copy2(f, outdir / name)


def convert_hf(hf_dir: Path, outdir: Path, pad_vocab_to: int, dtype,
               mp_world_size: int = 1) -> None:
    layer_re = re.compile(r"^model\.layers\.(\d+)\.(.+)$")
    collect: dict[int, dict] = {}
    num_layers = 0
    for k, v in iter_hf_state_dict(hf_dir):
        m = layer_re.match(k)
        if m:
            i = int(m.group(1))
            num_layers = max(num_layers, i + 1)
            sub = m.group(2)
            if "rotary_emb" in sub:  # non-parameter buffer in old HF checkpoints
                continue
            collect.setdefault(i + 1, {})[sub] = v
        elif k == "model.embed_tokens.weight":
            collect.setdefault(0, {})["weight"] = v
        elif k == "model.norm.weight":
            collect.setdefault("norm", {})["weight"] = v
        elif k == "lm_head.weight":
            collect.setdefault("head", {})["weight"] = v
    # renumber norm/head now that L is known
    collect[num_layers + 1] = collect.pop("norm")
    collect[num_layers + 2] = collect.pop("head")
    write_ckpt(outdir, num_layers, collect, pad_vocab_to, dtype,
               mp_world_size=mp_world_size)
    copy_tokenizer_and_config(hf_dir, outdir)
    print(f"wrote {num_layers + 3} layer files to {outdir}")


def convert_random(model_name: str, outdir: Path, dtype, seed: int = 0,
                   mp_world_size: int = 1) -> None:
    from lpp_amd.config import model_config
    from lpp_amd.layer_spec import LayerSpec
    from lpp_amd.models import deterministic_layer_init, get_layers_from_config

    cfg = model_config(model_name)
    specs = get_layers_from_config(cfg)
    collect = {}
    for idx, spec in enumerate(specs):
        layer = spec.build()
        deterministic_layer_init(layer, cfg, seed, idx)
        collect[idx] = {k: v.clone() for k, v in layer.state_dict().items()}
    write_ckpt(outdir, cfg.num_layers, collect, 0, dtype,
               mp_world_size=mp_world_size)
    print(f"wrote random-init {model_name} ({cfg.num_layers + 3} layer files) to {outdir}")


def main() -> int:
    ap = argparse.ArgumentParser()
    src = ap.add_mutually_exclusive_group(required=True)
    src.add_argument("--hf-dir", type=str)
    src.add_argument("--random-init", type=str, metavar="MODEL_PRESET")
    ap.add_argument("--output-dir", type=str, required=True)
    ap.add_argument("--pad-vocab-to", type=int, default=0)
    ap.add_argument("--dtype", type=str, default=None, choices=[None, "fp16", "bf16", "fp32"])
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--mp_world_size", type=int, default=1,
                    help="emit one mp_rank_XX metadata file per rank "
                         "(reference convert2ckpt.py:16; metadata only)")
    args = ap.parse_args()

    dt = {None: None, "fp16": torch.float16, "bf16": torch.bfloat16, "fp32": torch.float32}[
        args.dtype
    ]
    out = Path(args.output_dir)
    if args.hf_dir:
        convert_hf(Path(args.hf_dir), out, args.pad_vocab_to, dt,
                   mp_world_size=args.mp_world_size)
    else:
        convert_random(args.random_init, out, dt, args.seed,
                       mp_world_size=args.mp_world_size)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
