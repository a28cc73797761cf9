"""Continuous batching for single-device serving.

Requests arrive at any time; each is prefetched into a free KV-cache slot
(its prefill runs through the flash-attention path via a slot row-view)
and then decodes in ONE batched single-token step per engine tick together
with every other active request — finished requests retire and free their
slot immediately, so short requests never wait for long ones (the
"continuous"/in-flight batching of production serving stacks).  Decode
attention is the ragged per-row-length branch of ``LlamaAttention``
(eager SDPA: single-token decode is latency/memory-bound; a dedicated
decode GEMV kernel is BACKLOG work).

The reference has no inference capability at all (training-only template,
SURVEY.md §0); this extends the framework's own ``generate``/
``pipeline_generate`` serving story.

Usage:
    eng = ContinuousBatchingEngine(model, max_slots=8, max_seq_len=512)
    eng.submit(Request("a", prompt_ids_a, max_new_tokens=32))
    eng.submit(Request("b", prompt_ids_b, max_new_tokens=8))
    while eng.pending():
        eng.step()
    eng.results["a"]  # full sequence tensor (prompt + generated)
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from .models.llama import (DecoderLayerPipe, GatherKVCache, MultiSlotKVCache,
                           RaggedKVCache)


@dataclass
class Request:
    uid: str
    prompt: torch.Tensor  # [S0] long
    max_new_tokens: int
    temperature: float = 0.0
    # filled by the engine
    slot: int = -1
    generated: List[int] = field(default_factory=list)
    finished: bool = False


class ContinuousBatchingEngine:
    def __init__(self, model, max_slots: int, max_seq_len: int,
                 eos_token_id: Optional[int] = None,
                 generator: Optional[torch.Generator] = None):
        self.model = model
        self.model.eval()
        self.max_slots = max_slots
        self.max_seq_len = max_seq_len
        self.eos_token_id = eos_token_id
        self.generator = generator
        p = next(model.parameters())
        self.device = p.device
        cfg = model.cfg
        self.decoders = [l for l in model.layers if isinstance(l, DecoderLayerPipe)]
        self.caches = [
            RaggedKVCache(max_slots, max_seq_len, cfg.kv_heads, cfg.head_dim,
                          self.device, p.dtype)
            for _ in self.decoders
        ]
        self.free_slots = deque(range(max_slots))
        self.active: Dict[int, Request] = {}  # slot -> request
        self.waiting: deque[Request] = deque()
        self.results: Dict[str, torch.Tensor] = {}
        self._last_logits: Dict[int, torch.Tensor] = {}  # slot -> [V]

    # ------------------------------------------------------------------
    def submit(self, req: Request) -> None:
        if req.prompt.numel() + req.max_new_tokens > self.max_seq_len:
            raise ValueError(
                f"request {req.uid}: prompt+new "
                f"{req.prompt.numel() + req.max_new_tokens} > max_seq_len "
                f"{self.max_seq_len}")
        self.waiting.append(req)

    def pending(self) -> int:
        return len(self.waiting) + len(self.active)

    # ------------------------------------------------------------------
    def _run_layers(self, x: torch.Tensor, caches) -> torch.Tensor:
        ci = 0
        for layer in self.model.layers:
            if isinstance(layer, DecoderLayerPipe):
                x = layer(x, cache=caches[ci])
                ci += 1
            else:
                x = layer(x)
        return x

    @torch.no_grad()
    def _admit(self) -> None:
        """Move waiting requests into free slots; requests with EQUAL
        prompt lengths prefill together as one batched flash forward."""
        while self.waiting and self.free_slots:
            group = [self.waiting.popleft()]
            S0 = group[0].prompt.numel()
            while (self.waiting and len(group) < len(self.free_slots)
                   and self.waiting[0].prompt.numel() == S0):
                group.append(self.waiting.popleft())
            slots = [self.free_slots.popleft() for _ in group]
            for c in self.caches:
                for s in slots:
                    c.free(s)
            slot_t = torch.tensor(slots, dtype=torch.long, device=self.device)
            views = [MultiSlotKVCache(c, slot_t) for c in self.caches]
            ids = torch.stack([r.prompt for r in group]).to(self.device)
            logits = self._run_layers(ids, views)[:, -1]
            for i, (req, slot) in enumerate(zip(group, slots)):
                req.slot = slot
                self._last_logits[slot] = logits[i]
                self.active[slot] = req

    def _sample_batched(self, slots: List[int]) -> List[int]:
        """One device->host sync per tick (per-slot .item() calls dominated
        the first version's decode time)."""
        logits = torch.stack([self._last_logits[s] for s in slots])  # [N, V]
        if all(self.active[s].temperature == 0 for s in slots):
            return logits.argmax(dim=-1).tolist()
        temps = torch.tensor([self.active[s].temperature for s in slots],
                             device=logits.device)
        toks = logits.argmax(dim=-1)
        if bool((temps > 0).any()):
            probs = torch.softmax(
                logits.float() / temps.clamp(min=1e-6)[:, None], dim=-1)
            gen = self.generator if (self.generator is not None and
                                     str(self.generator.device) ==
                                     str(logits.device)) else None
            sampled = torch.multinomial(probs, 1, generator=gen).squeeze(-1)
            toks = torch.where(temps > 0, sampled, toks)
        return toks.tolist()

    @torch.no_grad()
    def step(self) -> List[str]:
        """One engine tick: admit -> sample each active slot's pending
        logits (batched, one sync) -> retire finished -> ONE batched
        ragged decode for the rest.  Returns the uids finished this tick."""
        self._admit()
        done: List[str] = []
        slots = sorted(self.active)
        if not slots:
            return done
        # sample from the logits produced by the previous forward (the
        # prefill for newly admitted requests)
        for slot, tok in zip(slots, self._sample_batched(slots)):
            req = self.active[slot]
            req.generated.append(tok)
            if ((self.eos_token_id is not None and tok == self.eos_token_id)
                    or len(req.generated) >= req.max_new_tokens):
                req.finished = True
                self.results[req.uid] = torch.cat(
                    [req.prompt.view(-1).cpu(),
                     torch.tensor(req.generated, dtype=torch.long)])
                done.append(req.uid)
                del self.active[slot]
                del self._last_logits[slot]
                for c in self.caches:
                    c.free(slot)
                self.free_slots.append(slot)
        if not self.active:
            return done
        # one batched ragged decode over every active slot (views memoized
        # per active-set; contiguous slot ranges decode zero-copy)
        slots = sorted(self.active)
        key = tuple(slots)
        if getattr(self, "_views_key", None) != key:
            slot_t = torch.tensor(slots, dtype=torch.long, device=self.device)
            rng = ((slots[0], slots[-1] + 1)
                   if slots == list(range(slots[0], slots[-1] + 1)) else None)
            self._views = [GatherKVCache(c, slot_t, contiguous_range=rng)
                           for c in self.caches]
            self._views_key = key
        # host-side per-tick tensors, shared by EVERY layer: max length,
        # gathered RoPE phases at each row's position, key-validity mask —
        # rebuilding these per layer made the decode tick launch-bound
        # the token being fed is generated[-1]: its position = tokens
        # already in the cache = prompt + len(generated) - 1
        lens_h = [self.active[s].prompt.numel() + len(self.active[s].generated) - 1
                  for s in slots]
        total = max(lens_h) + 1
        from lpp_amd import ops

        cfg = self.model.cfg
        cos, sin = ops.build_rope_cache(cfg.max_seq_len, cfg.head_dim,
                                        cfg.rope_theta, self.device)
        pos = torch.tensor(lens_h, dtype=torch.long, device=self.device)

        class _Tick:
            pass

        tick = _Tick()
        tick.cos = cos[pos].view(len(slots), 1, 1, -1)
        tick.sin = sin[pos].view(len(slots), 1, 1, -1)
        tick.mask = (torch.arange(total, device=self.device)[None, :]
                     <= pos[:, None])[:, None, None, :]
        for v in self._views:
            v.total_hint = total
            v.tick = tick
        toks = torch.tensor([self.active[s].generated[-1] for s in slots],
                            dtype=torch.long, device=self.device).view(-1, 1)
        logits = self._run_layers(toks, self._views)[:, -1]
        for i, s in enumerate(slots):
            self._last_logits[s] = logits[i]
        return done
