"""Value normalisers: PopArt-style EMA and Welford running stats.

Reference: pytorch/rl torchrl/modules/value_norm.py:30 (ValueNorm ABC),
:89 (PopArtValueNorm — debiased EMA, the MAPPO normaliser), :165
(RunningValueNorm — Welford counts).  Critics normalize regression
targets to a fixed scale and denormalize bootstrapped estimates back.
"""
from __future__ import annotations

from abc import ABCMeta, abstractmethod
from typing import Tuple, Union

import torch
from torch import nn

__all__ = ["ValueNorm", "PopArtValueNorm", "RunningValueNorm"]


class ValueNorm(nn.Module, metaclass=ABCMeta):
    """Running location/scale estimate of the value target."""

    def __init__(self, *, shape: Union[int, Tuple[int, ...]] = 1, epsilon: float = 1e-5, device=None):
        super().__init__()
        self.shape = (shape,) if isinstance(shape, int) else tuple(shape)
        self.epsilon = epsilon

    def _reduce_dims(self, value: torch.Tensor):
        nt = len(self.shape)
        if value.shape[-nt:] != torch.Size(self.shape):
            raise ValueError(
                f"trailing dims {tuple(value.shape[-nt:])} != normaliser shape {self.shape}"
            )
        return tuple(range(value.dim() - nt))

    @abstractmethod
    def update(self, value_target: torch.Tensor) -> None:
        ...

    @abstractmethod
    def normalize(self, value_target: torch.Tensor) -> torch.Tensor:
        ...

    @abstractmethod
    def denormalize(self, normalized_value: torch.Tensor) -> torch.Tensor:
        ...


class PopArtValueNorm(ValueNorm):
    """Debiased EMA of mean and mean-of-squares (PopArt / MAPPO)."""

    def __init__(self, *, shape=1, beta: float = 0.99999, epsilon: float = 1e-5, device=None):
        super().__init__(shape=shape, epsilon=epsilon, device=device)
        self.beta = beta
        self.register_buffer("running_mean", torch.zeros(self.shape, device=device))
        self.register_buffer("running_mean_sq", torch.zeros(self.shape, device=device))
        self.register_buffer("debiasing_term", torch.zeros((), device=device))

    def _stats(self):
        debias = self.debiasing_term.clamp_min(self.epsilon)
        mean = self.running_mean / debias
        var = (self.running_mean_sq / debias - mean.pow(2)).clamp_min(self.epsilon)
        return mean, var

    @torch.no_grad()
    def update(self, value_target: torch.Tensor) -> None:
        v = value_target.detach()
        dims = self._reduce_dims(v)
        m = v.mean(dim=dims) if dims else v
        m2 = v.pow(2).mean(dim=dims) if dims else v.pow(2)
        self.running_mean.mul_(self.beta).add_(m, alpha=1 - self.beta)
        self.running_mean_sq.mul_(self.beta).add_(m2, alpha=1 - self.beta)
        self.debiasing_term.mul_(self.beta).add_(1 - self.beta)

    def normalize(self, value_target: torch.Tensor) -> torch.Tensor:
        mean, var = self._stats()
        return (value_target - mean) / var.sqrt()

    def denormalize(self, normalized_value: torch.Tensor) -> torch.Tensor:
        mean, var = self._stats()
        return normalized_value * var.sqrt() + mean


class RunningValueNorm(ValueNorm):
    """Exact running mean/variance over all targets seen (Welford)."""

    def __init__(self, *, shape=1, epsilon: float = 1e-5, device=None):
        super().__init__(shape=shape, epsilon=epsilon, device=device)
        self.register_buffer("count", torch.zeros((), device=device))
        self.register_buffer("mean", torch.zeros(self.shape, device=device))
        self.register_buffer("m2", torch.zeros(self.shape, device=device))

    @torch.no_grad()
    def update(self, value_target: torch.Tensor) -> None:
        v = value_target.detach()
        dims = self._reduce_dims(v)
        n_new = 1
        for d in dims:
            n_new *= v.shape[d]
        batch_mean = v.mean(dim=dims) if dims else v
        batch_var = v.var(dim=dims, unbiased=False) if dims else torch.zeros_like(v)
        n_old = self.count.clone()
        n_tot = n_old + n_new
        delta = batch_mean - self.mean
        self.mean.add_(delta * (n_new / n_tot))
        self.m2.add_(batch_var * n_new + delta.pow(2) * n_old * n_new / n_tot)
        self.count.copy_(n_tot)

    def _var(self):
        return (self.m2 / self.count.clamp_min(1)).clamp_min(self.epsilon)

    def normalize(self, value_target: torch.Tensor) -> torch.Tensor:
        return (value_target - self.mean) / self._var().sqrt()

    def denormalize(self, normalized_value: torch.Tensor) -> torch.Tensor:
        return normalized_value * self._var().sqrt() + self.mean
