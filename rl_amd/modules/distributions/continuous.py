"""Continuous policy-head distributions: TanhNormal, IndependentNormal,
TruncatedNormal, Delta, TanhDelta.

Reference: pytorch/rl torchrl/modules/distributions/continuous.py
(IndependentNormal:47, TruncatedNormal:171, TanhNormal:337, Delta:667,
TanhDelta:755).  ``safetanh``/``safeatanh`` are the eps-clamped pair the
reference implements in C++ (torchrl/csrc/utils.cpp:9-48); here a python
autograd function with the same semantics, and the fused HIP actor epilogue
(rl_amd/ops) implements the same clamp on-device.
"""
from __future__ import annotations

import math
from numbers import Number
from typing import Optional, Sequence, Union

import numpy as np
import torch
from torch import distributions as D

__all__ = [
    "IndependentNormal",
    "TanhNormal",
    "TruncatedNormal",
    "Delta",
    "TanhDelta",
    "safetanh",
    "safeatanh",
]


class _SafeTanh(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, eps):
        out = x.tanh()
        lim = 1.0 - eps
        out = out.clamp(-lim, lim)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, grad):
        (out,) = ctx.saved_tensors
        return grad * (1 - out.pow(2)), None


class _SafeaTanh(torch.autograd.Function):
    @staticmethod
    def forward(ctx, y, eps):
        lim = 1.0 - eps
        y_c = y.clamp(-lim, lim)
        ctx.save_for_backward(y_c)
        return y_c.atanh()

    @staticmethod
    def backward(ctx, grad):
        (y_c,) = ctx.saved_tensors
        return grad / (1 - y_c.pow(2)), None


def safetanh(x: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    return _SafeTanh.apply(x, eps)


def safeatanh(y: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    return _SafeaTanh.apply(y, eps)


class IndependentNormal(D.Independent):
    """Normal with trailing event dim (reference continuous.py:47)."""

    def __init__(self, loc, scale, upscale: float = 5.0, tanh_loc: bool = False, event_dims: int = 1, **kwargs):
        self.tanh_loc = tanh_loc
        self.upscale = upscale
        if tanh_loc:
            loc = (loc / upscale).tanh() * upscale
        super().__init__(
            D.Normal(loc, scale, validate_args=False, **kwargs),
            event_dims,
            validate_args=False,
        )

    @property
    def mode(self):
        return self.base_dist.mean

    @property
    def deterministic_sample(self):
        return self.base_dist.mean


class TanhNormal(D.TransformedDistribution):
    """tanh-squashed Normal mapped onto [low, high]
    (reference continuous.py:337).

    ``sample``/``rsample``/``log_prob`` follow torch's
    TransformedDistribution; ``mode``/``deterministic_sample`` squashes the
    underlying mean (the reference's convention).
    """

    has_rsample = True
    arg_constraints = {}

    def __init__(
        self,
        loc: torch.Tensor,
        scale: torch.Tensor,
        upscale: float = 5.0,
        low: Union[float, torch.Tensor] = -1.0,
        high: Union[float, torch.Tensor] = 1.0,
        event_dims: Optional[int] = None,
        tanh_loc: bool = False,
        safe_tanh: bool = True,
        **kwargs,
    ):
        if event_dims is None:
            event_dims = min(1, loc.ndim)
        self.tanh_loc = tanh_loc
        self.upscale = upscale
        if tanh_loc:
            loc = (loc / upscale).tanh() * upscale
        # keep python-number bounds as floats: tensorizing them would issue
        # a blocking H2D copy on EVERY distribution construction (once per
        # env step in a collector) and break hipGraph capture
        if isinstance(low, (int, float)) and isinstance(high, (int, float)):
            self.low = float(low)
            self.high = float(high)
            self.non_trivial_bounds = low != -1.0 or high != 1.0
        else:
            low = torch.as_tensor(low, device=loc.device, dtype=loc.dtype)
            high = torch.as_tensor(high, device=loc.device, dtype=loc.dtype)
            self.low = low
            self.high = high
            self.non_trivial_bounds = bool((low != -1.0).any() or (high != 1.0).any())
        # validate_args=False everywhere: torch's arg validation runs a
        # device->host sync (torch._is_all_true) per construction — one
        # blocking sync per env step and a hipGraph-capture blocker
        base = D.Independent(
            D.Normal(loc, scale, validate_args=False), event_dims, validate_args=False
        )
        transforms = [D.TanhTransform(cache_size=1)]
        if self.non_trivial_bounds:
            transforms.append(
                D.AffineTransform(
                    loc=(self.high + self.low) / 2, scale=(self.high - self.low) / 2
                )
            )
        super().__init__(base, transforms, validate_args=False)

    @property
    def root_dist(self) -> D.Normal:
        return self.base_dist.base_dist

    @property
    def loc(self):
        return self.root_dist.loc

    @property
    def scale(self):
        return self.root_dist.scale

    @property
    def mode(self):
        m = self.root_dist.mean
        for t in self.transforms:
            m = t(m)
        return m

    @property
    def deterministic_sample(self):
        return self.mode

    @property
    def mean(self):
        # analytic mean has no closed form; use MC estimate like reference
        with torch.no_grad():
            s = self.sample((100,))
            return s.mean(0)

    def log_prob(self, value, **kwargs):
        # clamp into the open support to avoid inf at the boundary
        eps = torch.finfo(value.dtype).resolution
        if self.non_trivial_bounds:
            low = self.low
            high = self.high
            value = value.clamp(low + (high - low) * eps, high - (high - low) * eps)
        else:
            value = value.clamp(-1 + eps, 1 - eps)
        return super().log_prob(value)

    def entropy(self):
        # approximate: base entropy + log|det J| at the mean
        return self.base_dist.entropy()


class TruncatedNormal(D.Distribution):
    """Normal truncated to [low, high] (reference continuous.py:171 +
    truncated_normal.py).  Moment-matched implementation using inverse-CDF
    sampling — exact density within the bounds."""

    has_rsample = True
    arg_constraints = {}

    def __init__(self, loc, scale, upscale: float = 5.0, low: float = -1.0, high: float = 1.0, tanh_loc: bool = False):
        if tanh_loc:
            loc = (loc / upscale).tanh() * upscale
        self.loc = loc
        self.scale = scale.clamp_min(1e-6)
        self.low = torch.as_tensor(low, device=loc.device, dtype=loc.dtype)
        self.high = torch.as_tensor(high, device=loc.device, dtype=loc.dtype)
        batch_shape = loc.shape[:-1] if loc.ndim else torch.Size([])
        event_shape = loc.shape[-1:] if loc.ndim else torch.Size([])
        super().__init__(batch_shape, event_shape, validate_args=False)
        std = D.Normal(torch.zeros_like(loc), torch.ones_like(scale))
        self._alpha = (self.low - loc) / self.scale
        self._beta = (self.high - loc) / self.scale
        self._Phi_a = std.cdf(self._alpha)
        self._Phi_b = std.cdf(self._beta)
        self._Z = (self._Phi_b - self._Phi_a).clamp_min(1e-8)

    def rsample(self, sample_shape=torch.Size()):
        shape = torch.Size([*sample_shape, *self.loc.shape])
        u = torch.rand(shape, device=self.loc.device, dtype=self.loc.dtype)
        p = self._Phi_a + u * self._Z
        p = p.clamp(1e-6, 1 - 1e-6)
        z = math.sqrt(2.0) * torch.erfinv(2 * p - 1)
        return (self.loc + self.scale * z).clamp(self.low, self.high)

    def sample(self, sample_shape=torch.Size()):
        with torch.no_grad():
            return self.rsample(sample_shape)

    def log_prob(self, value):
        z = (value - self.loc) / self.scale
        log_phi = -0.5 * z.pow(2) - 0.5 * math.log(2 * math.pi) - self.scale.log()
        lp = log_phi - self._Z.log()
        out_of_bounds = (value < self.low) | (value > self.high)
        lp = lp.masked_fill(out_of_bounds, -1e6)
        return lp.sum(-1)

    @property
    def mode(self):
        return self.loc.clamp(self.low, self.high)

    @property
    def deterministic_sample(self):
        return self.mode

    @property
    def mean(self):
        std = D.Normal(torch.zeros_like(self.loc), torch.ones_like(self.scale))
        phi_a = std.log_prob(self._alpha).exp()
        phi_b = std.log_prob(self._beta).exp()
        return self.loc + self.scale * (phi_a - phi_b) / self._Z


class Delta(D.Distribution):
    """Deterministic distribution (reference continuous.py:667)."""

    has_rsample = True
    arg_constraints = {}

    def __init__(self, param: torch.Tensor, atol: float = 1e-6, rtol: float = 1e-6, batch_shape=None, event_shape=None):
        self.param = param
        self.atol = atol
        self.rtol = rtol
        if batch_shape is None:
            batch_shape = param.shape[:-1]
        if event_shape is None:
            event_shape = param.shape[-1:]
        super().__init__(torch.Size(batch_shape), torch.Size(event_shape), validate_args=False)

    def rsample(self, sample_shape=torch.Size()):
        shape = torch.Size([*sample_shape, *self.param.shape])
        return self.param.expand(shape)

    def sample(self, sample_shape=torch.Size()):
        return self.rsample(sample_shape).detach()

    def log_prob(self, value):
        is_eq = (value - self.param).abs() < (self.atol + self.rtol * self.param.abs())
        is_eq = is_eq.all(-1)
        out = torch.where(
            is_eq,
            torch.zeros_like(is_eq, dtype=self.param.dtype),
            torch.full_like(is_eq, -float("inf"), dtype=self.param.dtype),
        )
        return out

    @property
    def mode(self):
        return self.param

    @property
    def deterministic_sample(self):
        return self.param

    @property
    def mean(self):
        return self.param


class TanhDelta(Delta):
    """tanh-squashed deterministic distribution (reference continuous.py:755)."""

    def __init__(self, param, low: float = -1.0, high: float = 1.0, atol=1e-6, rtol=1e-6, **kwargs):
        low_t = torch.as_tensor(low, device=param.device, dtype=param.dtype)
        high_t = torch.as_tensor(high, device=param.device, dtype=param.dtype)
        squashed = safetanh(param)
        squashed = (high_t + low_t) / 2 + (high_t - low_t) / 2 * squashed
        super().__init__(squashed, atol=atol, rtol=rtol)
        self.low = low_t
        self.high = high_t
