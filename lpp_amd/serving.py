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

from .models.llama import DecoderLayerPipe, RaggedKVCache, GatherKVCache


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
        """Move waiting requests into free slots; prefill each one."""
        while self.waiting and self.free_slots:
            req = self.waiting.popleft()
            slot = self.free_slots.popleft()
            req.slot = slot
            for c in self.caches:
                c.free(slot)
            views = [c.slot_view(slot) for c in self.caches]
            ids = req.prompt.view(1, -1).to(self.device)
            logits = self._run_layers(ids, views)[0, -1]
            self._last_logits[slot] = logits
            self.active[slot] = req

    def _sample(self, req: Request, logits: torch.Tensor) -> int:
        if req.temperature > 0:
            probs = torch.softmax(logits.float() / req.temperature, dim=-1)
            return int(torch.multinomial(probs, 1, generator=self.generator))
        return int(logits.argmax())

    @torch.no_grad()
    def step(self) -> List[str]:
        """One engine tick: admit -> sample each active slot's pending
        logits -> retire finished -> ONE batched ragged decode for the
        rest.  Returns the uids finished this tick."""
        self._admit()
        done: List[str] = []
        # sample from the logits produced by the previous forward (the
        # prefill for newly admitted requests)
        for slot, req in list(self.active.items()):
            tok = self._sample(req, self._last_logits[slot])
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
        # one batched ragged decode over every active slot
        slots = sorted(self.active)
        slot_t = torch.tensor(slots, dtype=torch.long, device=self.device)
        toks = torch.tensor([self.active[s].generated[-1] for s in slots],
                            dtype=torch.long, device=self.device).view(-1, 1)
        views = [GatherKVCache(c, slot_t) for c in self.caches]
        logits = self._run_layers(toks, views)[:, -1]
        for i, s in enumerate(slots):
            self._last_logits[s] = logits[i]
        return done
