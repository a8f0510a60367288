"""Invertible scalar value transforms for critic targets.

Reference: pytorch/rl torchrl/modules/value_transforms.py:16
(ValueTransform ABC), :57 (Identity), :85 (SymLog), :123
(SignedHyperbolic — h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x, R2D2/
Pohlen et al.), :167 (Compose).
"""
from __future__ import annotations

from abc import ABCMeta, abstractmethod

import torch
from torch import nn

from .functional import symexp, symlog

__all__ = [
    "ValueTransform",
    "IdentityValueTransform",
    "SymLogValueTransform",
    "SignedHyperbolicValueTransform",
    "ComposeValueTransform",
    "signed_hyperbolic",
    "signed_parabolic",
]


def signed_hyperbolic(x: torch.Tensor, epsilon: float = 1e-3) -> torch.Tensor:
    """h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x."""
    return torch.sign(x) * ((x.abs() + 1).sqrt() - 1) + epsilon * x


def signed_parabolic(x: torch.Tensor, epsilon: float = 1e-3) -> torch.Tensor:
    """Inverse of :func:`signed_hyperbolic` (closed form)."""
    z = (1 + 4 * epsilon * (x.abs() + 1 + epsilon)).sqrt() - 1
    return torch.sign(x) * ((z / (2 * epsilon)).pow(2) - 1)


class ValueTransform(nn.Module, metaclass=ABCMeta):
    """Invertible scalar transform: critics regress transformed targets
    and invert their outputs for bootstrapping."""

    @abstractmethod
    def forward(self, value: torch.Tensor) -> torch.Tensor:
        ...

    @abstractmethod
    def inverse(self, value: torch.Tensor) -> torch.Tensor:
        ...


class IdentityValueTransform(ValueTransform):
    def forward(self, value):
        return value

    def inverse(self, value):
        return value


class SymLogValueTransform(ValueTransform):
    def forward(self, value):
        return symlog(value)

    def inverse(self, value):
        return symexp(value)


class SignedHyperbolicValueTransform(ValueTransform):
    def __init__(self, epsilon: float = 1e-3):
        super().__init__()
        if epsilon <= 0:
            raise ValueError("epsilon must be positive")
        self.epsilon = epsilon

    def forward(self, value):
        return signed_hyperbolic(value, self.epsilon)

    def inverse(self, value):
        return signed_parabolic(value, self.epsilon)


class ComposeValueTransform(ValueTransform):
    """Apply transforms in order; invert in reverse order."""

    def __init__(self, *transforms: ValueTransform):
        super().__init__()
        self.transforms = nn.ModuleList(transforms)

    def forward(self, value):
        for t in self.transforms:
            value = t(value)
        return value

    def inverse(self, value):
        for t in reversed(self.transforms):
            value = t.inverse(value)
        return value
