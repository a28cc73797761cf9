"""LLaMA as a flat list of pipeline-stage layer modules.

Native re-design of the reference's models/llama_ds_mp_wrap.py ("family B":
``get_layers_from_config`` at :209-224 building LayerSpecs for
EmbeddingPipe :128, ParallelTransformerLayerPipe :135, LayerNormPipe :184,
LMLayerPipe :191, with ``loss_fn`` :105-116), with the reference's contract
quirks fixed:

- The inter-stage tensor is ONE bf16 hidden-state tensor [B, S, H].
  No [B,1,S,S] additive mask is ever built or shipped (the reference sends
  the O(S^2) mask through every p2p hop — data/flan.py:194-243,
  models/llama_ds_mp_wrap.py:37,76,148); causality is implicit in the
  attention kernel.  No position_ids travel either: each stage regenerates
  RoPE phases from its static cos/sin table.
- Labels never smuggle an index column (Q2); the collator emits
  ``(input_ids, labels)`` and only the last stage consumes labels.

State-dict keys match HF ``LlamaForCausalLM`` per-layer keys with the
``model.layers.{i}.`` prefix stripped, so convert2ckpt's on-disk layout
(convert2ckpt.py:19-48) round-trips against real HF weights.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from ..config import ModelConfig
from ..layer_spec import LayerSpec
from .. import ops
from ..ops.linear import LPLinear, lp_linear


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rmsnorm(x, self.weight, self.variance_epsilon)


class KVCache:
    """Per-layer key/value cache for autoregressive decoding
    ([B, max_len, Hkv, D] bf16/fp32; ``length`` = filled prefix).  The
    reference is a training template with no inference path; this serves
    the framework's deployment story (single-device generation)."""

    def __init__(self, batch: int, max_len: int, kv_heads: int, head_dim: int,
                 device, dtype):
        self.k = torch.zeros(batch, max_len, kv_heads, head_dim, device=device, dtype=dtype)
        self.v = torch.zeros_like(self.k)
        self.length = 0

    def append(self, k: torch.Tensor, v: torch.Tensor) -> int:
        S = k.shape[1]
        self.k[:, self.length : self.length + S] = k
        self.v[:, self.length : self.length + S] = v
        self.length += S
        return self.length


class SlotKVCache:
    """Row-view of ONE slot of a RaggedKVCache, exposing the KVCache API so
    a single request's prefill runs through the unchanged flash path."""

    def __init__(self, parent: "RaggedKVCache", slot: int):
        self.parent = parent
        self.slot = slot

    @property
    def k(self):
        return self.parent.k[self.slot : self.slot + 1]

    @property
    def v(self):
        return self.parent.v[self.slot : self.slot + 1]

    @property
    def length(self) -> int:
        return int(self.parent.lengths[self.slot])

    def append(self, k: torch.Tensor, v: torch.Tensor) -> int:
        S = k.shape[1]
        ln = self.length
        self.parent.k[self.slot, ln : ln + S] = k[0]
        self.parent.v[self.slot, ln : ln + S] = v[0]
        self.parent.lengths[self.slot] = ln + S
        return ln + S


class RaggedKVCache:
    """Slot-based KV cache with PER-ROW lengths — the substrate for
    continuous batching (lpp_amd/serving.py): each slot holds an
    independent request at its own decode position.  ``per_row`` marks the
    ragged decode contract for LlamaAttention (per-row RoPE offsets +
    key-validity mask).  Beyond the reference (training-only template)."""

    per_row = True

    def __init__(self, slots: int, max_len: int, kv_heads: int, head_dim: int,
                 device, dtype):
        self.k = torch.zeros(slots, max_len, kv_heads, head_dim, device=device, dtype=dtype)
        self.v = torch.zeros_like(self.k)
        self.lengths = torch.zeros(slots, dtype=torch.long, device=device)

    def slot_view(self, slot: int) -> SlotKVCache:
        return SlotKVCache(self, slot)

    def free(self, slot: int) -> None:
        self.lengths[slot] = 0


class MultiSlotKVCache:
    """Batched-prefill view over several EMPTY slots of a RaggedKVCache:
    the G admitted prompts (equal length) prefill as one [G, S] forward
    through the unchanged flash path, K/V scattered into their rows."""

    def __init__(self, parent: RaggedKVCache, slots: torch.Tensor):
        self.parent = parent
        self.slots = slots
        self.length = 0  # uniform across the (empty) slots

    def append(self, k: torch.Tensor, v: torch.Tensor) -> int:
        S = k.shape[1]
        self.parent.k[self.slots, self.length : self.length + S] = k
        self.parent.v[self.slots, self.length : self.length + S] = v
        self.length += S
        self.parent.lengths[self.slots] = self.length
        return self.length


class GatherKVCache:
    """Decode-time view over the ACTIVE slots of a RaggedKVCache (ragged
    per-row lengths).  ``append`` scatters each row's new K/V at that
    row's own position.  When the active slots form a contiguous range
    (the common steady state — admission fills the lowest holes), pass
    ``contiguous_range`` so ``kv_to`` is a zero-copy slice instead of an
    advanced-indexing gather (the gather dominated decode tick time)."""

    per_row = True

    def __init__(self, parent: RaggedKVCache, slots: torch.Tensor,
                 contiguous_range: Optional[tuple] = None,
                 total_hint: Optional[int] = None):
        self.parent = parent
        self.slots = slots  # long tensor of active slot indices
        self.range = contiguous_range  # (lo, hi) half-open, or None
        # host-known max row length AFTER this tick's append — avoids a
        # device sync (lengths.max().item()) per layer per decode tick
        self.total_hint = total_hint

    @property
    def k(self):
        return self.parent.k[self.slots]

    @property
    def v(self):
        return self.parent.v[self.slots]

    def kv_to(self, T: int):
        """Active rows narrowed to the first T positions: a zero-copy slice
        for a contiguous slot range, else one advanced-indexing gather."""
        if self.range is not None:
            lo, hi = self.range
            return self.parent.k[lo:hi, :T], self.parent.v[lo:hi, :T]
        return self.parent.k[self.slots, :T], self.parent.v[self.slots, :T]

    @property
    def lengths(self) -> torch.Tensor:
        if self.range is not None:
            lo, hi = self.range
            return self.parent.lengths[lo:hi]
        return self.parent.lengths[self.slots]

    def append_one(self, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
        """Scatter single-token K/V ([N,1,Hkv,D]) at each active row's own
        length; returns the new per-row lengths."""
        lens = self.lengths
        if self.range is not None:
            lo, hi = self.range
            rows = torch.arange(lo, hi, device=lens.device)
        else:
            rows = self.slots
        self.parent.k[rows, lens] = k[:, 0]
        self.parent.v[rows, lens] = v[:, 0]
        new = lens + 1
        self.parent.lengths[rows] = new
        return new


class LlamaAttention(nn.Module):
    """Self-attention with RoPE; q/k/v/o projections named for HF key parity."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.num_heads = cfg.num_heads
        self.num_kv_heads = cfg.kv_heads
        self.head_dim = cfg.head_dim
        self.hidden_size = cfg.hidden_size
        self.rope_theta = cfg.rope_theta
        self.max_seq_len = cfg.max_seq_len
        self.q_proj = LPLinear(cfg.hidden_size, self.num_heads * self.head_dim)
        self.k_proj = LPLinear(cfg.hidden_size, self.num_kv_heads * self.head_dim)
        self.v_proj = LPLinear(cfg.hidden_size, self.num_kv_heads * self.head_dim)
        self.o_proj = LPLinear(self.num_heads * self.head_dim, cfg.hidden_size)

    def forward(self, x: torch.Tensor, cache: Optional["KVCache"] = None) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.num_heads, self.head_dim)
        k = self.k_proj(x).view(B, S, self.num_kv_heads, self.head_dim)
        v = self.v_proj(x).view(B, S, self.num_kv_heads, self.head_dim)
        cos, sin = ops.build_rope_cache(
            self.max_seq_len, self.head_dim, self.rope_theta, x.device
        )
        if cache is not None and getattr(cache, "per_row", False):
            # ragged single-token decode (continuous batching): every row
            # sits at its OWN position; RoPE phases gathered per row and
            # attention masked by per-row key validity.
            assert S == 1, "ragged cache is a single-token decode contract"
            tick = getattr(cache, "tick", None)  # per-tick hoisted tensors
            if tick is not None:
                q = ops.apply_rope_cs(q, tick.cos, tick.sin)
                k = ops.apply_rope_cs(k, tick.cos, tick.sin)
            else:
                lens = cache.lengths  # positions BEFORE append
                q = ops.apply_rope_positions(q, cos, sin, lens)
                k = ops.apply_rope_positions(k, cos, sin, lens)
            new_lens = cache.append_one(k, v)
            hint = getattr(cache, "total_hint", None)
            T = hint if hint is not None else int(new_lens.max())
            if hasattr(cache, "kv_to"):
                kc, vc = cache.kv_to(T)
            else:
                kc, vc = cache.k[:, :T], cache.v[:, :T]
            rep = self.num_heads // self.num_kv_heads
            qt = q.transpose(1, 2)
            kt = kc.transpose(1, 2)
            vt = vc.transpose(1, 2)
            if rep > 1:
                kt = kt.repeat_interleave(rep, dim=1)
                vt = vt.repeat_interleave(rep, dim=1)
            if tick is not None:
                mask = tick.mask
            else:
                mask = (torch.arange(T, device=x.device)[None, :]
                        < new_lens[:, None])[:, None, None, :]
            o = torch.nn.functional.scaled_dot_product_attention(
                qt, kt, vt, attn_mask=mask
            ).transpose(1, 2).contiguous()
            return self.o_proj(o.reshape(B, S, self.num_heads * self.head_dim))
        pos = cache.length if cache is not None else 0
        q = ops.apply_rope(q, cos, sin, pos_offset=pos)
        k = ops.apply_rope(k, cos, sin, pos_offset=pos)
        if cache is None:
            o = ops.causal_attention(q, k, v)  # [B,S,H,D]
        else:
            total = cache.append(k, v)
            if S == total:
                # prefill: plain causal over the whole prefix (flash path;
                # no cache read-back — MultiSlotKVCache is write-only)
                o = ops.causal_attention(q, k, v)
            else:
                kc = cache.k[:, :total]
                vc = cache.v[:, :total]
                # decode: q attends the full cached prefix.  rows see
                # positions <= their own: causal offset mask for S > 1,
                # no mask needed for single-token decode.
                rep = self.num_heads // self.num_kv_heads
                qt = q.transpose(1, 2)
                kt = kc.transpose(1, 2)
                vt = vc.transpose(1, 2)
                if rep > 1:
                    kt = kt.repeat_interleave(rep, dim=1)
                    vt = vt.repeat_interleave(rep, dim=1)
                mask = None
                if S > 1:
                    qpos = torch.arange(pos, total, device=x.device)
                    kpos = torch.arange(total, device=x.device)
                    mask = kpos[None, :] <= qpos[:, None]
                o = torch.nn.functional.scaled_dot_product_attention(
                    qt, kt, vt, attn_mask=mask
                ).transpose(1, 2).contiguous()
        return self.o_proj(o.reshape(B, S, self.num_heads * self.head_dim))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.gate_proj = LPLinear(cfg.hidden_size, cfg.intermediate_size)
        self.up_proj = LPLinear(cfg.hidden_size, cfg.intermediate_size)
        self.down_proj = LPLinear(cfg.intermediate_size, cfg.hidden_size)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))


class DecoderLayerPipe(nn.Module):
    """One LLaMA decoder layer; hidden in -> hidden out.

    Mirrors ParallelTransformerLayerPipe (models/llama_ds_mp_wrap.py:135-181)
    minus the tuple/mask plumbing.  Activation checkpointing recomputes the
    whole layer (interval semantics handled by the engine).
    """

    def __init__(self, cfg: ModelConfig, activation_checkpointing: bool = False):
        super().__init__()
        self.self_attn = LlamaAttention(cfg)
        self.mlp = LlamaMLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.activation_checkpointing = activation_checkpointing

    def _forward_impl(self, hidden: torch.Tensor,
                      cache: Optional[KVCache] = None) -> torch.Tensor:
        hidden = hidden + self.self_attn(self.input_layernorm(hidden), cache=cache)
        hidden = hidden + self.mlp(self.post_attention_layernorm(hidden))
        return hidden

    def forward(self, hidden: torch.Tensor,
                cache: Optional[KVCache] = None) -> torch.Tensor:
        if self.activation_checkpointing and self.training and hidden.requires_grad:
            return torch.utils.checkpoint.checkpoint(
                self._forward_impl, hidden, use_reentrant=False, preserve_rng_state=False
            )
        return self._forward_impl(hidden, cache=cache)

    @staticmethod
    def spec_param_count(cfg: ModelConfig, activation_checkpointing: bool = False) -> int:
        h, i = cfg.hidden_size, cfg.intermediate_size
        kv = cfg.kv_heads * cfg.head_dim
        return h * h * 2 + 2 * h * kv + 3 * h * i + 2 * h


class EmbeddingPipe(nn.Embedding):
    """Stage-0 entry: input_ids [B,S] int64 -> hidden [B,S,H].
    (reference: EmbeddingPipe, models/llama_ds_mp_wrap.py:128-132)"""

    def __init__(self, vocab_size: int, hidden_size: int):
        super().__init__(vocab_size, hidden_size)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        return super().forward(input_ids)

    @staticmethod
    def spec_param_count(vocab_size: int, hidden_size: int) -> int:
        return vocab_size * hidden_size


class NormPipe(RMSNorm):
    """Final RMSNorm (reference: LayerNormPipe, models/llama_ds_mp_wrap.py:184-188)."""

    @staticmethod
    def spec_param_count(hidden_size: int, eps: float = 1e-6) -> int:
        return hidden_size


class LMHeadPipe(nn.Linear):
    """LM head (reference: LMLayerPipe, models/llama_ds_mp_wrap.py:191-195).
    Untied from the embedding by design (README.md:44-46)."""

    def __init__(self, hidden_size: int, vocab_size: int):
        super().__init__(hidden_size, vocab_size, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return lp_linear(x, self.weight)

    @staticmethod
    def spec_param_count(hidden_size: int, vocab_size: int) -> int:
        return hidden_size * vocab_size


def loss_fn(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Shift-by-one causal CE (reference loss_fn, models/llama_ds_mp_wrap.py:105-116).
    Runs only on the last stage."""
    return ops.shifted_cross_entropy(logits, labels)


def get_layers_from_config(
    cfg: ModelConfig, activation_checkpointing: bool = False,
    checkpoint_fn=None,
) -> List[LayerSpec]:
    """Flat LayerSpec list: embedding, L decoder layers, final norm, LM head.
    Index i of this list == layer file number in the checkpoint layout
    (convert2ckpt.py:23-36).

    ``checkpoint_fn(layer_idx) -> bool`` enables SELECTIVE activation
    checkpointing: on 288 GB MI355X most stages can keep full activations
    (no recompute, ~+25% throughput); checkpoint only the layers the memory
    budget demands (the reference's all-or-nothing flag was an 80 GB-HBM
    coping mechanism, conf/...yaml:19)."""
    if cfg.tie_word_embeddings:
        # Tying embedding and LM head across pipeline stages would need a
        # tied-weight all-reduce the reference deliberately avoids for LLaMA
        # (README.md:44-46: never tie for LLaMA) — fail loudly instead of
        # silently training untied.
        raise ValueError(
            "tie_word_embeddings=True is not supported: LLaMA checkpoints are "
            "untied and the reference (README.md:44-46) warns against tying "
            "under pipeline parallelism. Set it to False.")
    specs: List[LayerSpec] = [LayerSpec(EmbeddingPipe, cfg.vocab_size, cfg.hidden_size)]
    for i in range(cfg.num_layers):
        ck = checkpoint_fn(i) if checkpoint_fn is not None else activation_checkpointing
        specs.append(LayerSpec(DecoderLayerPipe, cfg, ck))
    specs.append(LayerSpec(NormPipe, cfg.hidden_size, cfg.rms_norm_eps))
    specs.append(LayerSpec(LMHeadPipe, cfg.hidden_size, cfg.vocab_size))
    return specs


@torch.no_grad()
def init_weights(module: nn.Module, cfg: ModelConfig, generator: Optional[torch.Generator] = None):
    """HF-style init: normal(0, initializer_range) for weight matrices,
    ones for norms.  Deterministic given the generator."""
    std = cfg.initializer_range
    for m in module.modules():
        if isinstance(m, nn.Linear):
            m.weight.normal_(0.0, std, generator=generator)
            if m.bias is not None:
                m.bias.zero_()
        elif isinstance(m, nn.Embedding):
            m.weight.normal_(0.0, std, generator=generator)
        elif isinstance(m, RMSNorm):
            m.weight.fill_(1.0)


@torch.no_grad()
def deterministic_layer_init(layer: nn.Module, cfg: ModelConfig, seed: int, global_idx: int):
    """Seed per GLOBAL layer index so a pipeline stage initialises exactly the
    weights the monolithic model has for that layer — regardless of how the
    stage boundaries fall.  This is the oracle-alignment mechanism for the
    PP-vs-monolithic loss-equivalence tests."""
    g = torch.Generator().manual_seed(seed * 100003 + global_idx)
    std = cfg.initializer_range
    for name, p in sorted(layer.named_parameters()):
        if p.dim() >= 2:
            p.copy_(torch.empty(p.shape, dtype=torch.float32).normal_(0.0, std, generator=g).to(p.dtype))
        else:
            p.fill_(1.0) if "norm" in name or isinstance(layer, (RMSNorm,)) else p.zero_()
    # norm weights inside decoder layers
    for m in layer.modules():
        if isinstance(m, RMSNorm):
            m.weight.fill_(1.0)


def init_pipeline_weights(pipeline_module, cfg: ModelConfig, seed: int) -> None:
    for local_idx, layer in enumerate(pipeline_module.layers):
        deterministic_layer_init(layer, cfg, seed, pipeline_module.global_layer_index(local_idx))


def init_monolithic_weights(model: "LlamaForCausalLM", seed: int) -> None:
    for gidx, layer in enumerate(model.layers):
        deterministic_layer_init(layer, model.cfg, seed, gidx)


def layers_from_model(model: "LlamaForCausalLM") -> List[nn.Module]:
    """Family-A construction (reference ``get_model``,
    models/llama_ds_mp_wrap.py:119-125): take an ALREADY-LOADED monolithic
    model and return its flat layer list for a PipelineModule — each rank
    keeps references (no copy), so like the reference this materialises the
    full model per worker first (the drawback README.md:21 calls out; the
    recommended path is ``get_layers_from_config`` + per-stage checkpoint
    load).  No weight tying, by design (README.md:44-46)."""
    return list(model.layers)


class LlamaForCausalLM(nn.Module):
    """Monolithic (non-pipeline) model built from the same layer specs —
    the numerics oracle for PP-vs-single-process loss-equivalence tests
    (SURVEY.md §4)."""

    def __init__(self, cfg: ModelConfig, activation_checkpointing: bool = False):
        super().__init__()
        self.cfg = cfg
        self.layers = nn.ModuleList(
            [s.build() for s in get_layers_from_config(cfg, activation_checkpointing)]
        )

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        x = input_ids
        for layer in self.layers:
            x = layer(x)
        return x

    def compute_loss(self, input_ids: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        return loss_fn(self.forward(input_ids), labels)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int,
                 eos_token_id: Optional[int] = None,
                 temperature: float = 0.0,
                 generator: Optional[torch.Generator] = None) -> torch.Tensor:
        """KV-cached autoregressive generation (greedy, or sampling with
        ``temperature`` > 0).  Single-device serving path; the reference has
        no inference capability at all (it is a training template)."""
        was_training = self.training
        self.eval()
        B, S0 = input_ids.shape
        dev = input_ids.device
        p = next(self.parameters())
        max_len = S0 + max_new_tokens
        caches = [
            KVCache(B, max_len, self.cfg.kv_heads, self.cfg.head_dim, dev, p.dtype)
            for _ in range(self.cfg.num_layers)
        ]

        def step(ids: torch.Tensor) -> torch.Tensor:
            x = ids
            li = 0
            for layer in self.layers:
                if isinstance(layer, DecoderLayerPipe):
                    x = layer(x, cache=caches[li])
                    li += 1
                else:
                    x = layer(x)
            return x[:, -1]  # last-position logits [B, V]

        out = input_ids
        logits = step(input_ids)
        finished = torch.zeros(B, dtype=torch.bool, device=dev)
        for _ in range(max_new_tokens):
            if temperature > 0:
                probs = torch.softmax(logits.float() / temperature, dim=-1)
                nxt = torch.multinomial(probs, 1, generator=generator).squeeze(-1)
            else:
                nxt = logits.argmax(dim=-1)
            if eos_token_id is not None:
                nxt = torch.where(finished, torch.full_like(nxt, eos_token_id), nxt)
                finished |= nxt == eos_token_id
            out = torch.cat([out, nxt[:, None]], dim=1)
            if eos_token_id is not None and bool(finished.all()):
                break
            logits = step(nxt[:, None])
        if was_training:
            self.train()
        return out
