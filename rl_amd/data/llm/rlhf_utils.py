"""RLHF data utilities: tokenizer wrappers, structured containers,
model rollouts with KL-penalized rewards, top-k selection.

Reference: pytorch/rl torchrl/data/llm/utils.py:130 (RolloutFromModel),
dataset.py (TokenizedDatasetLoader, TensorDictTokenizer), prompt.py
(PromptData, PromptTensorDictTokenizer), reward.py (RewardData),
postprocs (TopKRewardSelector).
"""
from __future__ import annotations

from typing import Any, Callable, List, Optional, Sequence

import torch

from ...tensordict import TensorDict, TensorDictBase

__all__ = [
    "TensorDictTokenizer",
    "PromptTensorDictTokenizer",
    "PromptData",
    "RewardData",
    "TokenizedDatasetLoader",
    "RolloutFromModel",
    "TopKRewardSelector",
]


class TensorDictTokenizer:
    """Tokenize a batch of strings into ``input_ids``/``attention_mask``
    TensorDicts (reference dataset.py TensorDictTokenizer)."""

    def __init__(self, tokenizer, max_length: int = 512, key: str = "text",
                 padding: str = "max_length", truncation: bool = True):
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.key = key
        self.padding = padding
        self.truncation = truncation

    def __call__(self, sample) -> TensorDictBase:
        texts = sample[self.key] if isinstance(sample, dict) else sample
        if isinstance(texts, str):
            texts = [texts]
        enc = self.tokenizer(
            list(texts),
            max_length=self.max_length,
            padding=self.padding,
            truncation=self.truncation,
            return_tensors="pt",
        )
        return TensorDict(
            {"input_ids": enc["input_ids"], "attention_mask": enc["attention_mask"]},
            batch_size=[enc["input_ids"].shape[0]],
        )


class PromptTensorDictTokenizer(TensorDictTokenizer):
    """Prompt variant (reference prompt.py): also records the prompt
    length under ``prompt_rindex`` so generation knows where the label
    starts."""

    def __call__(self, sample) -> TensorDictBase:
        td = super().__call__(sample)
        rindex = td.get("attention_mask").sum(-1)
        td.set("prompt_rindex", rindex)
        return td


class PromptData:
    """Structured prompt container (reference prompt.py PromptData)."""

    def __init__(self, input_ids, attention_mask, prompt_rindex, labels=None):
        self.input_ids = input_ids
        self.attention_mask = attention_mask
        self.prompt_rindex = prompt_rindex
        self.labels = labels

    @classmethod
    def from_tensordict(cls, td: TensorDictBase) -> "PromptData":
        return cls(
            td.get("input_ids"),
            td.get("attention_mask"),
            td.get("prompt_rindex"),
            td.get("labels", None),
        )

    def to_tensordict(self) -> TensorDictBase:
        data = {
            "input_ids": self.input_ids,
            "attention_mask": self.attention_mask,
            "prompt_rindex": self.prompt_rindex,
        }
        if self.labels is not None:
            data["labels"] = self.labels
        return TensorDict(data, batch_size=[self.input_ids.shape[0]])


class RewardData:
    """Chosen/rejected pair container (reference reward.py RewardData)."""

    def __init__(self, input_ids, attention_mask, rewards=None, end_scores=None):
        self.input_ids = input_ids
        self.attention_mask = attention_mask
        self.rewards = rewards
        self.end_scores = end_scores


class TokenizedDatasetLoader:
    """Tokenize-and-memmap a text dataset once, reload thereafter
    (reference dataset.py TokenizedDatasetLoader).  Offline image: works
    with any in-memory list of strings or a `datasets` split."""

    def __init__(self, split_or_texts, tokenizer, max_length: int = 512,
                 dataset_name: Optional[str] = None, key: str = "text"):
        self.texts = split_or_texts
        self.tok = TensorDictTokenizer(tokenizer, max_length=max_length, key=key)

    def load(self) -> TensorDictBase:
        if isinstance(self.texts, (list, tuple)):
            return self.tok(list(self.texts))
        return self.tok([row for row in self.texts])


class RolloutFromModel:
    """Causal-LM rollouts with a KL-penalized reward (reference
    utils.py:130): generate from ``model``, score with ``reward_model``,
    subtract ``kl_coef * (log pi - log pi_ref)`` per generated token.

    Returns a TED-style TensorDict ready for PPO value/advantage
    estimation over the generated-token time axis.
    """

    def __init__(self, model, ref_model, reward_model, *, max_new_tokens: int = 50,
                 kl_coef: float = 0.1):
        self.model = model
        self.ref_model = ref_model
        self.reward_model = reward_model
        self.max_new_tokens = max_new_tokens
        self.kl_coef = kl_coef

    @staticmethod
    def _log_probs(model, ids, mask, prompt_len):
        out = model(input_ids=ids, attention_mask=mask)
        logits = out.logits if hasattr(out, "logits") else out[0]
        lp = logits[:, prompt_len - 1 : -1].log_softmax(-1)
        resp = ids[:, prompt_len:]
        return lp.gather(-1, resp.unsqueeze(-1)).squeeze(-1)

    @torch.no_grad()
    def rollout_from_data(self, batch: TensorDictBase) -> TensorDictBase:
        ids = batch.get("input_ids")
        mask = batch.get("attention_mask")
        P = ids.shape[1]
        gen = self.model.generate(
            input_ids=ids,
            attention_mask=mask,
            max_new_tokens=self.max_new_tokens,
            min_new_tokens=self.max_new_tokens,
            do_sample=True,
            pad_token_id=getattr(self.model.config, "pad_token_id", 0) or 0,
        )
        full_mask = torch.ones_like(gen)
        lp = self._log_probs(self.model, gen, full_mask, P)
        ref_lp = self._log_probs(self.ref_model, gen, full_mask, P)
        kl = lp - ref_lp
        _, end_scores = self.reward_model(gen, full_mask)
        G = gen.shape[1] - P
        reward = -self.kl_coef * kl
        reward[:, -1] += end_scores.squeeze(-1)
        B = gen.shape[0]
        done = torch.zeros(B, G, 1, dtype=torch.bool, device=gen.device)
        done[:, -1] = True
        # per-step view of the full sequence (expanded stride, no copy)
        ids_steps = gen.unsqueeze(1).expand(B, G, gen.shape[1])
        return TensorDict(
            {
                "input_ids": ids_steps,
                "sample_log_prob": lp,
                "ref_log_prob": ref_lp,
                "next": {
                    "reward": reward.unsqueeze(-1),
                    "done": done,
                    "terminated": done.clone(),
                },
            },
            batch_size=[B, G],
        )


class TopKRewardSelector:
    """Keep the top-k rewarded responses per prompt group (reference
    postprocs TopKRewardSelector) — best-of-N filtering before SFT."""

    def __init__(self, k: int, group_size: int, reward_key=("next", "reward")):
        self.k = k
        self.group_size = group_size
        self.reward_key = reward_key

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        n = td.batch_size[0]
        g = self.group_size
        if n % g:
            raise ValueError("batch not divisible by group_size")
        r = td.get(self.reward_key)
        # sequence reward = last-step reward when a time axis exists
        while r.dim() > 1:
            r = r[..., -1] if r.shape[-1] == 1 else r.sum(-1)
        scores = r.reshape(n // g, g)
        top = scores.topk(self.k, dim=-1).indices
        base = torch.arange(n // g, device=top.device).unsqueeze(-1) * g
        idx = (base + top).reshape(-1)
        return td[idx]
