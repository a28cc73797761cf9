"""Datasets, collator and loaders for the pipeline engine.

The tensor contract (fixing the reference's quirks Q1/Q2):
a batch is a dict ``{"input_ids": int64 [B, S], "labels": int64 [B, S]}``.
Only the first stage consumes ``input_ids``; only the last consumes
``labels``; middle stages consume nothing (so they need no dataset at all —
the property the reference engineered with its TestDataset placeholder,
data/test.py:4-22, README.md:64-129, falls out of the design here).

Labels are input_ids with ignored positions set to -100 (pad/prompt masking
as in data/flan.py:181-190); the engine's loss does the shift-by-one.

``SyntheticCausalLMDataset`` generates deterministic random token streams
(seeded per index) — the BASELINE benchmark path (no network, random-init
weights, synthetic data of the reference's shape).

``RepeatingLoader`` mirrors deepspeed.utils.RepeatingLoader
(trainer_base_ds_mp.py:339): infinite cycling iterator.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler, RandomSampler


class SyntheticCausalLMDataset(Dataset):
    """``pattern="uniform"`` (default): i.i.d. random tokens — the
    throughput-benchmark shape (nothing to learn beyond the unigram).
    ``pattern="arith"``: each sequence is an arithmetic progression
    ``(a + b*t) mod V`` with per-example (a, b) — next-token is fully
    determined by the context, so the training loss visibly descends
    (used for learning-dynamics demos and convergence tests)."""

    def __init__(self, length: int, seq_len: int, vocab_size: int, seed: int = 1234,
                 ignore_fraction: float = 0.0, pattern: str = "uniform"):
        self.length = length
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed
        self.ignore_fraction = ignore_fraction
        self.pattern = pattern

    def __len__(self) -> int:
        return self.length

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed + idx)
        if self.pattern == "arith":
            a = int(torch.randint(0, self.vocab_size, (1,), generator=g))
            b = int(torch.randint(1, 64, (1,), generator=g))
            t = torch.arange(self.seq_len, dtype=torch.long)
            ids = (a + b * t) % self.vocab_size
        else:
            ids = torch.randint(0, self.vocab_size, (self.seq_len,), generator=g)
        labels = ids.clone()
        if self.ignore_fraction > 0:
            n = int(self.seq_len * self.ignore_fraction)
            labels[:n] = -100
        return {"input_ids": ids, "labels": labels}


class CausalLMCollator:
    """Stacks example dicts; pads to seq_len if ragged (pad id 0, label -100)."""

    def __init__(self, seq_len: Optional[int] = None, pad_id: int = 0):
        self.seq_len = seq_len
        self.pad_id = pad_id

    def __call__(self, examples: List[Dict[str, torch.Tensor]]) -> Dict[str, torch.Tensor]:
        S = self.seq_len or max(e["input_ids"].numel() for e in examples)
        B = len(examples)
        input_ids = torch.full((B, S), self.pad_id, dtype=torch.long)
        labels = torch.full((B, S), -100, dtype=torch.long)
        for i, e in enumerate(examples):
            ids = e["input_ids"][:S]
            lab = e["labels"][:S]
            input_ids[i, : ids.numel()] = ids
            labels[i, : lab.numel()] = lab
        return {"input_ids": input_ids, "labels": labels}


class RepeatingLoader:
    def __init__(self, loader):
        self.loader = loader
        self._it = iter(loader)

    def __iter__(self):
        return self

    def __next__(self):
        try:
            return next(self._it)
        except StopIteration:
            self._it = iter(self.loader)
            return next(self._it)


def build_loader(
    dataset: Dataset,
    micro_batch_size: int,
    dp_degree: int,
    dp_id: int,
    seed: int,
    num_workers: int = 0,
    collator: Optional[CausalLMCollator] = None,
    epoch: int = 0,
) -> DataLoader:
    """Per-(stage, dp_id) loader.  First and last stage of the SAME dp column
    must draw identical sample sequences — guaranteed by the shared
    (dp_degree, dp_id, seed) sampler arguments (the reference does this with
    DistributedSampler(num_replicas=dp_degree, rank=dp_id),
    trainer_base_ds_mp.py:310-327, README.md:48-62)."""
    if dp_degree > 1:
        sampler = DistributedSampler(
            dataset, num_replicas=dp_degree, rank=dp_id, shuffle=True, seed=seed
        )
        sampler.set_epoch(epoch)
    else:
        g = torch.Generator().manual_seed(seed + epoch)
        sampler = RandomSampler(dataset, generator=g)
    return DataLoader(
        dataset,
        batch_size=micro_batch_size,
        sampler=sampler,
        num_workers=num_workers,
        collate_fn=collator or CausalLMCollator(),
        drop_last=True,
        pin_memory=torch.cuda.is_available(),
    )
