"""VLA (vision-language-action) data containers: action tokenizers.

Reference: pytorch/rl torchrl/data/vla (ActionTokenizerBase and bin
tokenizers used by ActionTokenizerTransform, _action.py:2105).  The
rl_amd form keeps the codec self-contained (bins live in the tokenizer;
no env needed) so it can sit on a replay buffer, an env inverse path, or
after a token-head policy interchangeably.
"""
from __future__ import annotations

from typing import Optional, Sequence, Union

import torch

__all__ = ["ActionTokenizerBase", "UniformActionTokenizer"]


class ActionTokenizerBase:
    """Bidirectional continuous-action <-> token-id codec."""

    vocab_size: int

    def encode(self, action: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def decode(self, tokens: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


class UniformActionTokenizer(ActionTokenizerBase):
    """Uniform binning over ``[low, high]`` per action dim.

    ``encode`` maps each action coordinate to a bin id in
    ``[0, num_bins)``; ``decode`` maps a bin id back to the bin
    midpoint.  Round-trip error is bounded by half a bin width.
    """

    def __init__(
        self,
        num_bins: int = 256,
        low: Union[float, torch.Tensor] = -1.0,
        high: Union[float, torch.Tensor] = 1.0,
    ):
        self.num_bins = int(num_bins)
        self.vocab_size = self.num_bins
        self.low = torch.as_tensor(low, dtype=torch.float32)
        self.high = torch.as_tensor(high, dtype=torch.float32)

    def encode(self, action: torch.Tensor) -> torch.Tensor:
        low = self.low.to(action.device)
        high = self.high.to(action.device)
        frac = (action.float() - low) / (high - low)
        return (frac * self.num_bins).long().clamp_(0, self.num_bins - 1)

    def decode(self, tokens: torch.Tensor) -> torch.Tensor:
        low = self.low.to(tokens.device)
        high = self.high.to(tokens.device)
        frac = (tokens.float() + 0.5) / self.num_bins
        return low + frac * (high - low)
