"""Pipeline-parallel KV-cached generation.

Serving path across pipeline stages: the prompt prefills every stage's
KV caches in one forward sweep, then decoding ping-pongs one token per
step — stage 0 embeds the last token, hidden states hop stage-to-stage
over RCCL/xGMI ([B, 1, H] bf16, ~16 KB at 65B: latency-bound, not
bandwidth-bound), the last stage samples, and the sampled token is
broadcast so stage 0 can continue.  The reference has no inference
capability at all (training-only template); the single-device analog is
``LlamaForCausalLM.generate``.

All ranks call ``pipeline_generate`` collectively with the same arguments;
``input_ids`` must be identical on the first and last stages (middle
stages may pass the same tensor or one of matching shape).  Returns the
full [B, S0 + new] sequence on every rank.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from .models import DecoderLayerPipe, KVCache
from .pipeline_module import PipelineModule
from .topology import ProcessGrid


def _send(t: torch.Tensor, dst: int) -> None:
    dist.batch_isend_irecv([dist.P2POp(dist.isend, t.contiguous(), dst)])[0].wait()


def _recv(shape, dtype, device, src: int) -> torch.Tensor:
    buf = torch.empty(shape, dtype=dtype, device=device)
    dist.batch_isend_irecv([dist.P2POp(dist.irecv, buf, src)])[0].wait()
    return buf


@torch.no_grad()
def pipeline_generate(
    module: PipelineModule,
    grid: ProcessGrid,
    input_ids: torch.Tensor,
    max_new_tokens: int,
    device: Optional[torch.device] = None,
    dtype: Optional[torch.dtype] = None,
    eos_token_id: Optional[int] = None,
    temperature: float = 0.0,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    was_training = module.training
    module.eval()
    dev = device or input_ids.device
    p = next(module.parameters())
    act_dtype = dtype or p.dtype

    B, S0 = input_ids.shape
    decoders = [l for l in module.layers if isinstance(l, DecoderLayerPipe)]
    attn0 = decoders[0].self_attn if decoders else None
    max_len = S0 + max_new_tokens
    if attn0 is not None and max_len > attn0.max_seq_len:
        raise ValueError(f"S0+new={max_len} exceeds max_seq_len={attn0.max_seq_len}")
    caches = [
        KVCache(B, max_len, d.self_attn.num_kv_heads, d.self_attn.head_dim, dev, act_dtype)
        for d in decoders
    ]

    def run_local(x: torch.Tensor) -> torch.Tensor:
        ci = 0
        for layer in module.layers:
            if isinstance(layer, DecoderLayerPipe):
                x = layer(x, cache=caches[ci])
                ci += 1
            else:
                x = layer(x)
        return x

    first, last = grid.is_first_stage(), grid.is_last_stage()
    prev_rank, next_rank = grid.prev_rank, grid.next_rank
    last_rank = grid.stage_to_rank(grid.num_stages - 1)
    # incoming-hop width for non-first stages
    if attn0 is not None:
        H = attn0.hidden_size
    else:
        l0 = module.layers[0]
        H = getattr(l0, "in_features", None) or l0.weight.numel()

    def stage_step(tokens: torch.Tensor, seq: int) -> Optional[torch.Tensor]:
        """Run one pipeline sweep of ``seq`` positions; returns last-position
        logits on the last stage, None elsewhere."""
        if first:
            x = run_local(tokens.to(dev))
            if next_rank is not None:
                _send(x, next_rank)
                return None
            return x[:, -1]
        x = _recv((B, seq, H), act_dtype, dev, prev_rank)
        x = run_local(x)
        if last:
            return x[:, -1]
        _send(x, next_rank)
        return None

    out = input_ids.to(dev) if input_ids is not None else None
    finished = torch.zeros(B, dtype=torch.bool, device=dev)

    logits = stage_step(input_ids, S0)
    for _ in range(max_new_tokens):
        # last stage samples; token + stop flag broadcast to the pipe group
        msg = torch.zeros(B + 1, dtype=torch.long, device=dev)
        if last:
            if temperature > 0:
                probs = torch.softmax(logits.float() / temperature, dim=-1)
                nxt = torch.multinomial(probs, 1, generator=generator).squeeze(-1)
            else:
                nxt = logits.argmax(dim=-1)
            if eos_token_id is not None:
                nxt = torch.where(finished, torch.full_like(nxt, eos_token_id), nxt)
                finished |= nxt == eos_token_id
                msg[B] = int(finished.all())
            msg[:B] = nxt
        if dist.is_initialized() and grid.num_stages > 1:
            dist.broadcast(msg, src=last_rank, group=grid.pipe_group)
        nxt = msg[:B]
        out = torch.cat([out, nxt[:, None]], dim=1)
        if eos_token_id is not None and bool(msg[B]):
            break
        logits = stage_step(nxt[:, None], 1)

    if was_training:
        module.train()
    return out
