"""Auxiliary model components: BatchRenorm1d (CrossQ), SymExpTwoHot
(DreamerV3 heads), squeeze layers, MC-dropout module.

Reference: pytorch/rl torchrl/modules/models/batchrenorm.py:11
(BatchRenorm1d, Ioffe 2017), models/dreamer_v3.py (SymExpTwoHot),
models/utils.py (SqueezeLayer, Squeeze2dLayer), exploration
ConsistentDropoutModule.
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from ..functional import default_bins, symexp, two_hot_decode

__all__ = [
    "BatchRenorm1d",
    "SymExpTwoHot",
    "SqueezeLayer",
    "Squeeze2dLayer",
    "ConsistentDropout",
    "ConsistentDropoutModule",
]


class BatchRenorm1d(nn.Module):
    """Batch renormalization (Ioffe 2017; reference batchrenorm.py:11).

    Behaves like BatchNorm during a warmup phase, then normalizes with
    running statistics corrected by clipped r/d factors — the
    normalization CrossQ relies on instead of target networks.
    """

    def __init__(
        self,
        num_features: int,
        *,
        momentum: float = 0.01,
        eps: float = 1e-5,
        max_r: float = 3.0,
        max_d: float = 5.0,
        warmup_steps: int = 10000,
    ):
        super().__init__()
        self.num_features = num_features
        self.momentum = momentum
        self.eps = eps
        self.max_r = max_r
        self.max_d = max_d
        self.warmup_steps = warmup_steps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        flat = x.reshape(-1, self.num_features)
        if self.training:
            mean = flat.mean(0)
            var = flat.var(0, unbiased=False)
            std = (var + self.eps).sqrt()
            run_std = (self.running_var + self.eps).sqrt()
            warm = self.num_batches_tracked < self.warmup_steps
            if warm:
                r = torch.ones_like(std)
                d = torch.zeros_like(mean)
            else:
                r = (std / run_std).clamp(1.0 / self.max_r, self.max_r).detach()
                d = ((mean - self.running_mean) / run_std).clamp(
                    -self.max_d, self.max_d
                ).detach()
            out = (flat - mean) / std * r + d
            with torch.no_grad():
                self.running_mean.mul_(1 - self.momentum).add_(self.momentum * mean)
                self.running_var.mul_(1 - self.momentum).add_(self.momentum * var)
                self.num_batches_tracked += 1
        else:
            out = (flat - self.running_mean) / (self.running_var + self.eps).sqrt()
        out = out * self.weight + self.bias
        return out.reshape(x.shape)


class SymExpTwoHot(nn.Module):
    """Decode two-hot logits over symlog bins to a scalar (DreamerV3
    reward/value head epilogue; reference models/dreamer_v3.py)."""

    def __init__(self, num_bins: int = 255, low: float = -20.0, high: float = 20.0):
        super().__init__()
        self.register_buffer("bins", default_bins(num_bins, low, high))

    def forward(self, logits: torch.Tensor) -> torch.Tensor:
        return symexp(two_hot_decode(logits, self.bins)).unsqueeze(-1)


class SqueezeLayer(nn.Module):
    """Squeeze trailing singleton dims (reference models/utils.py)."""

    def __init__(self, dims=(-1,)):
        super().__init__()
        self.dims = dims

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for d in self.dims:
            x = x.squeeze(d)
        return x


class Squeeze2dLayer(SqueezeLayer):
    """Squeeze the two trailing singleton dims (post-conv)."""

    def __init__(self):
        super().__init__(dims=(-1, -1))


class ConsistentDropout(nn.Module):
    """Dropout with a mask held fixed until explicitly resampled —
    MC-dropout exploration (reference exploration ConsistentDropout):
    the same mask applies across a trajectory, giving a temporally
    consistent stochastic policy."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p
        self._mask: Optional[torch.Tensor] = None

    def reset_mask(self):
        self._mask = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.training and self._mask is None:
            return x
        if self._mask is None or self._mask.shape != x.shape:
            self._mask = torch.bernoulli(
                torch.full_like(x, 1 - self.p)
            ) / (1 - self.p)
        return x * self._mask


class ConsistentDropoutModule(nn.Module):
    """TensorDict wrapper over :class:`ConsistentDropout`: reads
    ``in_key``, writes ``out_key``, resamples the mask when the input
    carries ``is_init`` True (trajectory starts)."""

    def __init__(self, p: float = 0.5, in_key: str = "observation", out_key: Optional[str] = None):
        super().__init__()
        self.dropout = ConsistentDropout(p)
        self.in_key = in_key
        self.out_key = out_key or in_key
        self.in_keys = [in_key, "is_init"]
        self.out_keys = [self.out_key]

    def forward(self, td):
        is_init = td.get("is_init", None)
        if is_init is not None and bool(is_init.any()):
            self.dropout.reset_mask()
        td.set(self.out_key, self.dropout(td.get(self.in_key)))
        return td
