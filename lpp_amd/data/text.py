"""Text data pipeline: prompt/completion corpora -> the engine's tensor
contract.

Native re-design of the reference's data layer (file:line cites into
/root/reference/):

- ``Seq2SeqToCausalLM`` replaces ``vanilla_seq2seq_convertor``
  (data/flan.py:149-170): decoder-only conversion concatenates
  ``inputs + " " + targets + eos``, and derives the prompt length so the
  loss covers COMPLETION tokens only.  The reference derives prompt lengths
  with a re-tokenisation + halving heuristic (data/flan.py:163-166); here
  prompt lengths come from tokenising the prompt prefix directly — exact,
  no heuristic.
- ``completion_labels`` replaces ``get_lm_labels`` (data/flan.py:181-190):
  labels = input_ids with pad positions and positions < prompt_len masked
  to -100.
- ``TextCollator`` replaces ``FlanCollatorOverCollator``
  (data/flan.py:263-309) in its ``return_standard_inputs`` role — but emits
  the clean contract ``{"input_ids", "labels"}`` (quirks Q1/Q2 fixed,
  SURVEY.md §2.7): no 4-D attention mask is ever built (the reference
  materialises a [B,1,S,S] fp16 mask host-side, data/flan.py:194-243, and
  ships it through every pipeline stage; our attention kernels make
  causality implicit), no position_ids travel (each stage regenerates RoPE
  phases), and no index column is smuggled into the labels.
- ``RoundRobinMixDataset`` replaces WikiPathDatasetV5WFlan /
  WikiPathDatasetV5WithDataset (data/flan.py:65-121): zips datasets by
  ``index % len``, ``__len__`` = max.
- ``FlattenedGroupDataset`` replaces FlanCollectionGroupDataset
  (data/flan.py:124-146): flattens a dataset of example-groups.
- ``PromptResponseDataset`` replaces PromptDataset/FLANDataset
  (data/flan.py:36-63): jsonl or torch-saved list of
  {"inputs"/"prompt": ..., "targets"/"response": ...} dicts, with the same
  empty-example filter (data/flan.py:15-29).
- ``PlaceholderDataset`` replaces TestDataset (data/test.py:4-22,
  README.md:64-129): a constant example with a configurable pseudo length.
  The reference NEEDS it (middle ranks must construct a dataloader whose
  length agrees with the real one or the step counts diverge — quirk Q3);
  in this engine middle stages read no data at all, so this exists only for
  API parity and for driving the trainer with a synthetic corpus.
- ``expand_special_tokenizer`` replaces
  general_util/tokenization_utils.py:15-56: LLaMA default special tokens
  with EOS_TOKEN/BOS_TOKEN/UNK_TOKEN/PAD_TOKEN env overrides and the
  pad -> eos fallback.
"""

from __future__ import annotations

import json
import logging
import os
from typing import Dict, List, Optional, Sequence

import torch
from torch.utils.data import Dataset

logger = logging.getLogger(__name__)

IGNORE_INDEX = -100

DEFAULT_PAD_TOKEN = "[PAD]"
DEFAULT_EOS_TOKEN = "</s>"
DEFAULT_BOS_TOKEN = "<s>"
DEFAULT_UNK_TOKEN = "<unk>"


def expand_special_tokenizer(tokenizer) -> int:
    """Add LLaMA default special tokens (env-var overridable) and fall back
    pad -> eos (general_util/tokenization_utils.py:15-56).  Returns the
    number of tokens added (callers resize embeddings when > 0,
    convert2ckpt.py:60-63)."""
    name = getattr(tokenizer, "name_or_path", "") or type(tokenizer).__name__
    name = name.lower()
    added = 0
    if "llama" in name:
        mapping = {}
        eos = os.environ.get("EOS_TOKEN")
        if eos or not tokenizer.eos_token:
            mapping["eos_token"] = eos or DEFAULT_EOS_TOKEN
        bos = os.environ.get("BOS_TOKEN")
        if bos or not tokenizer.bos_token:
            mapping["bos_token"] = bos or DEFAULT_BOS_TOKEN
        if not tokenizer.unk_token:
            mapping["unk_token"] = os.environ.get("UNK_TOKEN") or DEFAULT_UNK_TOKEN
        if not tokenizer.pad_token:
            mapping["pad_token"] = os.environ.get("PAD_TOKEN") or DEFAULT_PAD_TOKEN
        if mapping:
            added = tokenizer.add_special_tokens(special_tokens_dict=mapping)
    if not tokenizer.pad_token:
        tokenizer.pad_token = tokenizer.eos_token
        tokenizer.pad_token_id = tokenizer.eos_token_id
    return added


def load_prompt_response_data(file_path: str) -> List[Dict[str, str]]:
    """Load a list of {"inputs": ..., "targets": ...} dicts from .jsonl/.json
    or a torch-saved list, filtering empty examples (data/flan.py:15-29).
    Accepts "prompt"/"response" keys as aliases (data/flan.py:45-49)."""
    if file_path.endswith((".jsonl", ".json")):
        with open(file_path) as f:
            if file_path.endswith(".jsonl"):
                data = [json.loads(line) for line in f if line.strip()]
            else:
                data = json.load(f)
    else:
        data = torch.load(file_path, map_location="cpu")
    out, dropped = [], 0
    for item in data:
        inputs = item.get("inputs", item.get("prompt", ""))
        targets = item.get("targets", item.get("response", ""))
        if not str(inputs).strip() or not str(targets).strip():
            dropped += 1
            continue
        out.append({"inputs": str(inputs), "targets": str(targets)})
    if dropped:
        logger.info("dropped %d empty examples from %s", dropped, file_path)
    return out


class PromptResponseDataset(Dataset):
    """File-backed prompt/completion dataset (reference PromptDataset /
    FLANDataset, data/flan.py:36-63)."""

    def __init__(self, file_path: str):
        self.data = load_prompt_response_data(file_path)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx) -> Dict[str, str]:
        return self.data[idx]


class RoundRobinMixDataset(Dataset):
    """Zip datasets by index % len; length = max over members
    (reference WikiPathDatasetV5WFlan, data/flan.py:65-90).  Each item is a
    dict merging the members' items under their given names."""

    def __init__(self, **datasets: Dataset):
        assert datasets, "need at least one member dataset"
        self.names = list(datasets)
        self.datasets = datasets

    def __len__(self):
        return max(len(d) for d in self.datasets.values())

    def __getitem__(self, idx):
        return {n: d[idx % len(d)] for n, d in self.datasets.items()}


class FlattenedGroupDataset(Dataset):
    """Flatten a dataset whose items are LISTS of examples
    (reference FlanCollectionGroupDataset, data/flan.py:124-146)."""

    def __init__(self, grouped: Sequence[Sequence]):
        self.index = []
        self.grouped = grouped
        for gi, group in enumerate(grouped):
            for ei in range(len(group)):
                self.index.append((gi, ei))

    def __len__(self):
        return len(self.index)

    def __getitem__(self, idx):
        gi, ei = self.index[idx]
        return self.grouped[gi][ei]


class PlaceholderDataset(Dataset):
    """Constant-example dataset with a configurable pseudo length
    (reference TestDataset, data/test.py:4-22).  Kept for parity with the
    reference's middle-rank memory trick; this engine's middle stages do
    not construct dataloaders at all (engine.py)."""

    def __init__(self, pseudo_dataset_len: int = -1,
                 example: Optional[Dict[str, str]] = None):
        self.pseudo_dataset_len = pseudo_dataset_len
        self.example = example or {"inputs": "placeholder prompt",
                                   "targets": "placeholder completion"}

    def __len__(self):
        return self.pseudo_dataset_len if self.pseudo_dataset_len > 0 else 100000000

    def __getitem__(self, idx):
        return dict(self.example)


def completion_labels(input_ids: torch.Tensor, prompt_lens: torch.Tensor,
                      pad_token_id: int, ignore_index: int = IGNORE_INDEX,
                      lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
    """labels = input_ids with padding AND positions < prompt_len masked
    (loss on completion tokens only; reference get_lm_labels,
    data/flan.py:181-190).

    Padding is masked BY POSITION when ``lengths`` (real tokens per row) is
    given: position >= length is padding.  Masking by token id alone breaks
    when pad_token falls back to eos_token (expand_special_tokenizer) — the
    EOS terminating each example would be masked and the model could never
    learn to stop.  Without ``lengths`` the id-based mask is kept for
    callers whose pad id is distinct."""
    labels = input_ids.clone()
    pos = torch.arange(labels.size(1), device=labels.device)[None, :]
    if lengths is not None:
        keep = pos < lengths.to(labels.device)[:, None]
    else:
        keep = labels.ne(pad_token_id)
    keep &= pos >= prompt_lens.to(labels.device)[:, None]
    return labels.masked_fill(~keep, ignore_index).contiguous()


class Seq2SeqToCausalLM:
    """Tokenise {"inputs","targets"} pairs into decoder-only training rows:
    ids = tok(inputs + " " + targets + eos), prompt_len = len(tok(inputs))
    — exact, replacing the reference's re-tokenise-and-halve heuristic
    (vanilla_seq2seq_convertor, data/flan.py:149-170)."""

    def __init__(self, tokenizer, max_seq_length: int):
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        expand_special_tokenizer(tokenizer)

    def __call__(self, examples: List[Dict[str, str]]) -> Dict[str, torch.Tensor]:
        texts = [e["inputs"] + " " + e["targets"] + self.tokenizer.eos_token
                 for e in examples]
        enc = self.tokenizer(texts, max_length=self.max_seq_length, padding="longest",
                             truncation=True, return_tensors="pt", add_special_tokens=True)
        prompts = self.tokenizer([e["inputs"] for e in examples],
                                 max_length=self.max_seq_length, padding="longest",
                                 truncation=True, return_tensors="pt",
                                 add_special_tokens=True)
        ids = enc["input_ids"]
        # Real row lengths by position, not by pad-id (pad may alias eos).
        if "attention_mask" in enc:
            lengths = enc["attention_mask"].sum(dim=1)
        else:
            lengths = ids.ne(self.tokenizer.pad_token_id).sum(dim=1)
        if "attention_mask" in prompts:
            standalone_lens = prompts["attention_mask"].sum(dim=1)
        else:
            standalone_lens = prompts["input_ids"].ne(
                self.tokenizer.pad_token_id).sum(dim=1)
        # BPE/SentencePiece can merge tokens across the prompt/completion
        # boundary, so the standalone prompt encoding need not be a prefix of
        # the concatenated encoding.  Use the longest common prefix between
        # the two encodings as the loss boundary: a boundary-merged token
        # counts as completion (conservative — loss starts at the merge).
        prompt_lens = torch.empty_like(standalone_lens)
        pids = prompts["input_ids"]
        for i in range(ids.size(0)):
            n = int(min(standalone_lens[i], lengths[i]))
            eq = ids[i, :n].eq(pids[i, :n])
            prompt_lens[i] = n if bool(eq.all()) else int(eq.logical_not().byte().argmax())
        prompt_lens = torch.minimum(
            prompt_lens, torch.full_like(prompt_lens, self.max_seq_length)
        )
        return {"input_ids": ids, "prompt_lens": prompt_lens, "lengths": lengths}


class TextCollator:
    """Batch of {"inputs","targets"} dicts (optionally nested under a key, as
    the mixing datasets produce) -> {"input_ids", "labels"} — the engine's
    contract.  Replaces FlanCollatorOverCollator(return_standard_inputs=True)
    (data/flan.py:263-309) minus the mask/position plumbing (implicit here)
    and the label-index smuggling (quirk Q2)."""

    def __init__(self, tokenizer, max_seq_length: int, field: Optional[str] = None,
                 pad_to_max: bool = True):
        self.convert = Seq2SeqToCausalLM(tokenizer, max_seq_length)
        self.field = field
        self.pad_to_max = pad_to_max
        self.max_seq_length = max_seq_length

    def __call__(self, batch: List[Dict]) -> Dict[str, torch.Tensor]:
        examples = [b[self.field] if self.field else b for b in batch]
        enc = self.convert(examples)
        ids = enc["input_ids"]
        tok = self.convert.tokenizer
        if self.pad_to_max and ids.size(1) < self.max_seq_length:
            pad = torch.full((ids.size(0), self.max_seq_length - ids.size(1)),
                             tok.pad_token_id, dtype=ids.dtype)
            ids = torch.cat([ids, pad], dim=1)
        labels = completion_labels(ids, enc["prompt_lens"], tok.pad_token_id,
                                   lengths=enc["lengths"])
        return {"input_ids": ids, "labels": labels}


def combine_on_length(a: torch.Tensor, b: torch.Tensor,
                      pad_value: int = 0) -> torch.Tensor:
    """Stack two batches of id rows whose sequence lengths may differ,
    padding the shorter to the longer (reference combine_tensor_on_length,
    data/flan.py:173-178)."""
    L = max(a.size(-1), b.size(-1))

    def _pad(t: torch.Tensor) -> torch.Tensor:
        if t.size(-1) == L:
            return t
        pad = torch.full((*t.shape[:-1], L - t.size(-1)), pad_value, dtype=t.dtype)
        return torch.cat([t, pad], dim=-1)

    return torch.cat([_pad(a), _pad(b)], dim=0)


class WrappingCollator:
    """Multi-task batching: wrap ANOTHER collator and merge this collator's
    converted examples into its batch under ``<prefix>_*`` keys (reference
    FlanCollatorOverCollator's wrapping mode, data/flan.py:263-309 — the
    items are dicts carrying both the inner collator's example and a
    ``field`` entry with this side's {"inputs","targets"} example, as
    RoundRobinMixDataset produces).  Tensor keys present on both sides are
    ALSO merged into one batch via length-padding (combine_on_length) so a
    single forward can span both tasks."""

    def __init__(self, inner_collator, tokenizer, max_seq_length: int,
                 field: str = "flan", prefix: Optional[str] = None,
                 merge_keys: Sequence[str] = ()):
        self.inner = inner_collator
        self.own = TextCollator(tokenizer, max_seq_length, field=None)
        self.field = field
        self.prefix = prefix if prefix is not None else field
        self.merge_keys = tuple(merge_keys)

    def __call__(self, batch: List[Dict]) -> Dict[str, torch.Tensor]:
        inner_items = [{k: v for k, v in b.items() if k != self.field} for b in batch]
        own_items = [b[self.field] for b in batch]
        out = dict(self.inner(inner_items))
        own = self.own(own_items)
        for k, v in own.items():
            out[f"{self.prefix}_{k}"] = v
        for k in self.merge_keys:
            pad = IGNORE_INDEX if k == "labels" else self.own.convert.tokenizer.pad_token_id
            out[k] = combine_on_length(out[k], own[k], pad_value=pad)
        return out


class SimpleTokenizer:
    """Minimal offline whitespace tokenizer implementing the subset of the
    HF tokenizer protocol the data layer uses.  Exists because this
    environment has no network for real tokenizer files; tests and synthetic
    corpora run on it, and a real ``transformers`` tokenizer drops in
    unchanged (both go through expand_special_tokenizer)."""

    name_or_path = "simple-llama"

    def __init__(self, vocab_size: int = 32000):
        self.vocab_size = vocab_size
        self.bos_token = "<s>"
        self.eos_token = "</s>"
        self.unk_token = "<unk>"
        self.pad_token = None
        self._special = {self.bos_token: 1, self.eos_token: 2, self.unk_token: 0}
        self.pad_token_id = None
        self.bos_token_id = 1
        self.eos_token_id = 2

    def add_special_tokens(self, special_tokens_dict):
        added = 0
        if "pad_token" in special_tokens_dict and self.pad_token is None:
            self.pad_token = special_tokens_dict["pad_token"]
            self._special[self.pad_token] = 3
            self.pad_token_id = 3
            added += 1
        return added

    def _tok(self, word: str) -> int:
        if word in self._special:
            return self._special[word]
        return 4 + (hash(word) % (self.vocab_size - 4))

    def __call__(self, texts, max_length=None, padding="longest", truncation=True,
                 return_tensors="pt", add_special_tokens=True):
        if isinstance(texts, str):
            texts = [texts]
        rows = []
        for t in texts:
            # split eos off if appended with no space
            t = t.replace(self.eos_token, " " + self.eos_token + " ")
            ids = [self._tok(w) for w in t.split()]
            if add_special_tokens:
                ids = [self.bos_token_id] + ids
            if truncation and max_length:
                ids = ids[:max_length]
            rows.append(ids)
        L = max(len(r) for r in rows)
        pad_id = self.pad_token_id if self.pad_token_id is not None else 0
        out = torch.full((len(rows), L), pad_id, dtype=torch.long)
        mask = torch.zeros((len(rows), L), dtype=torch.long)
        for i, r in enumerate(rows):
            out[i, : len(r)] = torch.tensor(r, dtype=torch.long)
            mask[i, : len(r)] = 1
        return {"input_ids": out, "attention_mask": mask}
