"""Offline LLM test fixtures (reference: pytorch/rl
torchrl/testing/llm_mocks.py) — a byte-level tokenizer and a tiny
random-init causal LM so the LLM vertical tests run with no downloads."""
from __future__ import annotations

from typing import List, Optional, Union

import torch

__all__ = ["ByteTokenizer", "make_tiny_lm"]


class _Encoding(dict):
    def __getattr__(self, k):
        return self[k]


class ByteTokenizer:
    """UTF-8 byte tokenizer with pad/eos; mimics the HF tokenizer API
    surface used by TransformersWrapper."""

    vocab_size = 258
    pad_token_id = 256
    eos_token_id = 257

    def __call__(self, texts: Union[str, List[str]], return_tensors=None,
                 padding=False, max_length=None, truncation=False):
        if isinstance(texts, str):
            texts = [texts]
        cutoff = max_length if (truncation and max_length) else 512
        seqs = [list(t.encode("utf-8"))[:cutoff] for t in texts]
        max_len = max(len(s) for s in seqs)
        if padding == "max_length" and max_length:
            max_len = max_length
        ids, mask = [], []
        for s in seqs:
            pad = max_len - len(s)
            ids.append([self.pad_token_id] * pad + s)
            mask.append([0] * pad + [1] * len(s))
        return _Encoding(
            input_ids=torch.tensor(ids, dtype=torch.long),
            attention_mask=torch.tensor(mask, dtype=torch.long),
        )

    def decode(self, ids, skip_special_tokens=True):
        out = bytes(
            int(i) for i in ids if int(i) < 256
        )
        return out.decode("utf-8", errors="replace")

    def batch_decode(self, batch, skip_special_tokens=True):
        return [self.decode(row, skip_special_tokens) for row in batch]


def make_tiny_lm(vocab_size: int = 258, n_layer: int = 2, n_head: int = 2, n_embd: int = 32):
    """Random-init tiny GPT-2 (config-only — no network needed)."""
    from transformers import GPT2Config, GPT2LMHeadModel

    cfg = GPT2Config(
        vocab_size=vocab_size,
        n_layer=n_layer,
        n_head=n_head,
        n_embd=n_embd,
        n_positions=1024,
        pad_token_id=256,
        eos_token_id=257,
    )
    return GPT2LMHeadModel(cfg)
