"""Functional helpers shared by DreamerV3-style models and losses.

Reference: pytorch/rl torchrl/modules/functional.py (symlog/symexp) and
torchrl/modules/models/dreamer_v3.py (two-hot codecs, unimix).
"""
from __future__ import annotations

from typing import Optional

import torch

__all__ = [
    "symlog",
    "symexp",
    "default_bins",
    "two_hot_encode",
    "two_hot_decode",
    "two_hot_cross_entropy",
    "unimix_probs",
]

DEFAULT_NUM_BINS = 255


def symlog(x: torch.Tensor) -> torch.Tensor:
    """sign(x) * log(1 + |x|) — the DreamerV3 squashing."""
    return torch.sign(x) * torch.log1p(x.abs())


def symexp(x: torch.Tensor) -> torch.Tensor:
    """Inverse of :func:`symlog`: sign(x) * (exp(|x|) - 1)."""
    return torch.sign(x) * torch.expm1(x.abs())


def default_bins(
    num_bins: int = DEFAULT_NUM_BINS,
    low: float = -20.0,
    high: float = 20.0,
    device=None,
) -> torch.Tensor:
    """Uniform symlog-space bin centers."""
    return torch.linspace(low, high, num_bins, device=device)


def two_hot_encode(x: torch.Tensor, bins: torch.Tensor) -> torch.Tensor:
    """Encode scalars as a two-hot distribution over ``bins``
    (weight split between the two neighbouring bin centers)."""
    x = x.clamp(bins[0], bins[-1])
    idx_hi = torch.searchsorted(bins, x, right=False).clamp(1, bins.numel() - 1)
    idx_lo = idx_hi - 1
    lo, hi = bins[idx_lo], bins[idx_hi]
    w_hi = ((x - lo) / (hi - lo).clamp_min(1e-8)).clamp(0, 1)
    out = torch.zeros(*x.shape, bins.numel(), device=x.device, dtype=x.dtype)
    out.scatter_(-1, idx_lo.unsqueeze(-1), (1 - w_hi).unsqueeze(-1))
    out.scatter_add_(-1, idx_hi.unsqueeze(-1), w_hi.unsqueeze(-1))
    return out


def two_hot_decode(logits: torch.Tensor, bins: torch.Tensor) -> torch.Tensor:
    """Expected bin value under softmax(logits)."""
    return (torch.softmax(logits, -1) * bins).sum(-1)


def two_hot_cross_entropy(
    logits: torch.Tensor, target: torch.Tensor, bins: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """Cross-entropy between ``softmax(logits)`` and the two-hot encoding
    of ``symlog(target)`` — the DreamerV3 reward/value loss."""
    if bins is None:
        bins = default_bins(logits.shape[-1], device=logits.device)
    t = two_hot_encode(symlog(target), bins)
    return -(t * torch.log_softmax(logits, -1)).sum(-1)


def unimix_probs(logits: torch.Tensor, unimix: float = 0.01) -> torch.Tensor:
    """Mix ``unimix`` uniform probability into softmax(logits)
    (DreamerV3 categorical regularization)."""
    probs = torch.softmax(logits, -1)
    if unimix:
        probs = (1 - unimix) * probs + unimix / logits.shape[-1]
    return probs
