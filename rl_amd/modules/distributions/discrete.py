"""Discrete policy-head distributions: OneHotCategorical, MaskedCategorical,
Ordinal.

Reference: pytorch/rl torchrl/modules/distributions/discrete.py
(OneHotCategorical:65, MaskedCategorical:175, Ordinal:629).
"""
from __future__ import annotations

from enum import Enum
from typing import Optional

import torch
from torch import distributions as D

__all__ = ["OneHotCategorical", "MaskedCategorical", "Ordinal", "MaskedOneHotCategorical", "LLMMaskedCategorical"]


class OneHotCategorical(D.Categorical):
    """Categorical that samples one-hot vectors (reference discrete.py:65)."""

    def __init__(self, logits: Optional[torch.Tensor] = None, probs: Optional[torch.Tensor] = None, **kwargs):
        kwargs.setdefault("validate_args", False)
        super().__init__(probs=probs, logits=logits, **kwargs)
        self.num_samples = self._num_events

    def sample(self, sample_shape=torch.Size()):
        idx = super().sample(sample_shape)
        return torch.nn.functional.one_hot(idx, self._num_events).to(torch.int64)

    def rsample(self, sample_shape=torch.Size()):
        # straight-through gumbel for differentiable sampling
        logits = self.logits
        shape = torch.Size([*sample_shape, *logits.shape])
        g = -torch.log(-torch.log(torch.rand(shape, device=logits.device).clamp_min(1e-10)).clamp_min(1e-10))
        idx = (logits + g).argmax(-1)
        hard = torch.nn.functional.one_hot(idx, self._num_events).to(logits.dtype)
        soft = torch.softmax(logits + g, -1)
        return hard + soft - soft.detach()

    def log_prob(self, value):
        return super().log_prob(value.argmax(-1))

    def entropy(self):
        return super().entropy()

    @property
    def mode(self):
        idx = self.logits.argmax(-1)
        return torch.nn.functional.one_hot(idx, self._num_events).to(torch.int64)

    @property
    def deterministic_sample(self):
        return self.mode


class MaskedCategorical(D.Categorical):
    """Categorical with invalid actions masked out
    (reference discrete.py:175)."""

    def __init__(
        self,
        logits: Optional[torch.Tensor] = None,
        probs: Optional[torch.Tensor] = None,
        mask: Optional[torch.Tensor] = None,
        indices: Optional[torch.Tensor] = None,
        neg_inf: float = float("-inf"),
        **kwargs,
    ):
        if mask is None and indices is not None:
            mask = torch.zeros_like(
                logits if logits is not None else probs, dtype=torch.bool
            ).scatter_(-1, indices, True)
        self._mask = mask
        if logits is not None and mask is not None:
            logits = logits.masked_fill(~mask, neg_inf if neg_inf != float("-inf") else -3.4e38)
        if probs is not None and mask is not None:
            probs = probs * mask
            probs = probs / probs.sum(-1, keepdim=True).clamp_min(1e-10)
        kwargs.setdefault("validate_args", False)
        super().__init__(probs=probs, logits=logits, **kwargs)

    @property
    def mode(self):
        return self.logits.argmax(-1)

    @property
    def deterministic_sample(self):
        return self.mode


class MaskedOneHotCategorical(OneHotCategorical):
    def __init__(self, logits=None, probs=None, mask=None, **kwargs):
        if logits is not None and mask is not None:
            logits = logits.masked_fill(~mask, -3.4e38)
        if probs is not None and mask is not None:
            probs = probs * mask
            probs = probs / probs.sum(-1, keepdim=True).clamp_min(1e-10)
        super().__init__(logits=logits, probs=probs, **kwargs)


class Ordinal(D.Categorical):
    """Ordinal-structured categorical for discretized continuous actions
    (reference discrete.py:629): p(k) built from cumulative sigmoids so
    nearby categories have correlated mass."""

    def __init__(self, scores: torch.Tensor):
        logits = _ordinal_logits(scores)
        super().__init__(logits=logits)

    @property
    def mode(self):
        return self.logits.argmax(-1)

    @property
    def deterministic_sample(self):
        return self.mode


def _ordinal_logits(scores: torch.Tensor) -> torch.Tensor:
    log_sig = torch.nn.functional.logsigmoid(scores)
    log_one_minus = torch.nn.functional.logsigmoid(-scores)
    cum = torch.cumsum(log_sig, -1)
    rev = torch.flip(torch.cumsum(torch.flip(log_one_minus, (-1,)), -1), (-1,)) - log_one_minus
    return cum + rev


class LLMMaskedCategorical(MaskedCategorical):
    """Token-level masked categorical for LLM losses
    (reference discrete.py:708): mask shape [*, T, V] over vocab per
    position; log_prob reduces over the sequence with the mask."""

    def __init__(self, logits: torch.Tensor, mask: torch.Tensor, **kwargs):
        super().__init__(logits=logits, mask=mask, **kwargs)
        self._token_mask = mask

    def log_prob_token(self, tokens: torch.Tensor) -> torch.Tensor:
        return super().log_prob(tokens)

    def log_prob(self, tokens: torch.Tensor) -> torch.Tensor:
        lp = super().log_prob(tokens)
        valid = self._token_mask.gather(-1, tokens.unsqueeze(-1)).squeeze(-1)
        return (lp * valid.to(lp.dtype)).sum(-1)


class ReparamGradientStrategy(Enum):
    """How discrete samples get surrogate gradients (reference
    discrete.py:60): straight-through pass-through, or relaxed one-hot
    (Gumbel-softmax style)."""

    PassThrough = 1
    RelaxedOneHot = 2


class OneHotOrdinal(OneHotCategorical):
    """One-hot version of :class:`Ordinal` (reference discrete.py:677):
    ordinal-structured logits, one-hot samples/log-probs."""

    def __init__(self, scores: torch.Tensor):
        super().__init__(logits=_ordinal_logits(scores))


__all__ += ["ReparamGradientStrategy", "OneHotOrdinal"]
