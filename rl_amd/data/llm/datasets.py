"""LLM prompt / preference datasets over local files.

Reference: pytorch/rl torchrl/data/llm/ (prompt/reward/topk datasets) —
no-egress build: loaders read local jsonl/txt; `PromptTensorDictLoader`
yields batched TensorDicts with History entries for ChatEnv/GRPO loops.
"""
from __future__ import annotations

import json
from typing import Iterator, List, Optional, Sequence

import torch

from ...tensordict import TensorDict, TensorDictBase
from .history import History

__all__ = ["PromptDataset", "PairwisePreferenceDataset", "PromptTensorDictLoader"]


class PromptDataset:
    """Prompts from a .txt (one per line) or .jsonl ({"prompt": ...})."""

    def __init__(self, path_or_list, repeat: bool = True):
        if isinstance(path_or_list, (list, tuple)):
            self.prompts = [str(p) for p in path_or_list]
        elif str(path_or_list).endswith(".jsonl"):
            self.prompts = [
                json.loads(line).get("prompt", "")
                for line in open(path_or_list)
                if line.strip()
            ]
        else:
            self.prompts = [l.rstrip("\n") for l in open(path_or_list) if l.strip()]
        self.repeat = repeat

    def __len__(self):
        return len(self.prompts)

    def __getitem__(self, i):
        return self.prompts[i % len(self.prompts)]

    def __iter__(self) -> Iterator[str]:
        i = 0
        while True:
            if i >= len(self.prompts):
                if not self.repeat:
                    return
                i = 0
            yield self.prompts[i]
            i += 1


class PairwisePreferenceDataset:
    """(prompt, chosen, rejected) triples from jsonl — reward-model /
    DPO-style data."""

    def __init__(self, path):
        self.rows = [json.loads(l) for l in open(path) if l.strip()]

    def __len__(self):
        return len(self.rows)

    def __getitem__(self, i):
        r = self.rows[i]
        return r["prompt"], r["chosen"], r["rejected"]

    def as_tensordict(self, tokenizer, max_length: int = 512) -> TensorDictBase:
        prompts = [r["prompt"] for r in self.rows]
        chosen = [r["prompt"] + r["chosen"] for r in self.rows]
        rejected = [r["prompt"] + r["rejected"] for r in self.rows]
        enc_c = tokenizer(chosen, return_tensors="pt", padding=True)
        enc_r = tokenizer(rejected, return_tensors="pt", padding=True)
        td = TensorDict({}, batch_size=[len(self.rows)])
        td.set("chosen_ids", enc_c["input_ids"][:, :max_length])
        td.set("rejected_ids", enc_r["input_ids"][:, :max_length])
        td.set_non_tensor("prompts", prompts)
        return td


class PromptTensorDictLoader:
    """Yield [batch] TensorDicts carrying History prompts — feeds ChatEnv
    or direct GRPO generation loops."""

    def __init__(self, dataset: PromptDataset, batch_size: int, group_repeats: int = 1):
        self.dataset = dataset
        self.batch_size = batch_size
        self.group_repeats = group_repeats
        self._it = iter(dataset)

    def __iter__(self):
        return self

    def __next__(self) -> TensorDictBase:
        prompts: List[str] = []
        while len(prompts) < self.batch_size:
            p = next(self._it)
            prompts.extend([p] * self.group_repeats)
        prompts = prompts[: self.batch_size]
        h = History.from_text(prompts)
        td = TensorDict({}, batch_size=[len(prompts)])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        return td
