"""LLMHashingEnv — token-chain identification via hashing.

Reference: pytorch/rl torchrl/envs/llm/envs.py:602 (LLMHashingEnv):
each step appends the sampled token to the sequence and re-hashes the
chain, so MCTS-style search structures (MCTSForest) can key nodes on a
single int64 hash instead of the full token tensor.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch

from ...data.map import SipHash
from ...data.tensor_specs import Categorical, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["LLMHashingEnv"]


class LLMHashingEnv(EnvBase):
    """Text-generation env whose observation is a hash of the token
    chain so far.  ``action`` is the next token; the env appends it,
    hashes the new chain and reports it under ``hash``.  With a
    ``tokenizer``, decoded text is kept under ``text``."""

    def __init__(
        self,
        vocab_size: Optional[int] = None,
        *,
        hashing_module: Optional[Callable[[torch.Tensor], torch.Tensor]] = None,
        observation_key: str = "observation",
        text_output: bool = True,
        tokenizer=None,
        device=None,
    ):
        super().__init__(device=device, batch_size=())
        if vocab_size is None:
            if tokenizer is None:
                raise TypeError("pass vocab_size or a tokenizer")
            vocab_size = len(tokenizer)
        self.vocab_size = vocab_size
        self._hash = hashing_module if hashing_module is not None else SipHash()
        self.observation_key = observation_key
        self.tokenizer = tokenizer
        self.text_output = text_output and tokenizer is not None
        self.observation_spec = Composite(
            {
                observation_key: Unbounded(shape=(-1,), dtype=torch.int64, device=self.device),
                "hash": Unbounded(shape=(1,), dtype=torch.int64, device=self.device),
            },
            shape=(),
            device=self.device,
        )
        self.action_spec = Categorical(vocab_size, shape=(), device=self.device)
        self.reward_spec = Unbounded(shape=(1,), device=self.device)

    def _hash_chain(self, tokens: torch.Tensor) -> torch.Tensor:
        h = self._hash(tokens.reshape(1, -1).float())
        return torch.as_tensor(h, device=self.device).reshape(1).long()

    def _make_obs(self, tokens: torch.Tensor) -> TensorDictBase:
        out = TensorDict(
            {
                self.observation_key: tokens,
                "hash": self._hash_chain(tokens),
                "done": torch.zeros(1, dtype=torch.bool, device=self.device),
                "terminated": torch.zeros(1, dtype=torch.bool, device=self.device),
            },
            batch_size=(),
            device=self.device,
        )
        if self.text_output:
            out.set_non_tensor("text", self.tokenizer.decode(tokens.tolist()))
        return out

    def _reset(self, tensordict=None, **kwargs) -> TensorDictBase:
        if tensordict is not None and self.observation_key in tensordict:
            tokens = tensordict.get(self.observation_key).reshape(-1).long()
        else:
            tokens = torch.zeros(1, dtype=torch.int64, device=self.device)
        self._tokens = tokens
        return self._make_obs(tokens)

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        action = tensordict.get("action").reshape(-1)[-1:].long()
        prev = tensordict.get(self.observation_key, self._tokens).reshape(-1).long()
        tokens = torch.cat([prev, action])
        self._tokens = tokens
        out = self._make_obs(tokens)
        out.set("reward", torch.zeros(1, device=self.device))
        return out

    def _set_seed(self, seed):
        if seed is not None:
            torch.manual_seed(seed)
