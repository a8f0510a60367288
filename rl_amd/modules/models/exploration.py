"""Exploration model components: NoisyLinear, gSDE, ConsistentDropout.

Reference: pytorch/rl torchrl/modules/models/exploration.py
(NoisyLinear:29, gSDEModule:280, ConsistentDropoutModule:571).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn

__all__ = ["NoisyLinear", "NoisyLazyLinear", "gSDEModule", "ConsistentDropout", "reset_noise"]


class NoisyLinear(nn.Module):
    """Factorized-Gaussian noisy linear layer (reference exploration.py:29;
    Fortunato et al. 2017).  Call :func:`reset_noise` between episodes."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = True,
        device=None,
        dtype=None,
        std_init: float = 0.1,
    ):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.std_init = std_init
        factory = {"device": device, "dtype": dtype}
        self.weight_mu = nn.Parameter(torch.empty(out_features, in_features, **factory))
        self.weight_sigma = nn.Parameter(torch.empty(out_features, in_features, **factory))
        self.register_buffer("weight_epsilon", torch.zeros(out_features, in_features, **factory))
        if bias:
            self.bias_mu = nn.Parameter(torch.empty(out_features, **factory))
            self.bias_sigma = nn.Parameter(torch.empty(out_features, **factory))
            self.register_buffer("bias_epsilon", torch.zeros(out_features, **factory))
        else:
            self.bias_mu = None
        self.reset_parameters()
        self.reset_noise()

    def reset_parameters(self):
        mu_range = 1.0 / math.sqrt(self.in_features)
        nn.init.uniform_(self.weight_mu, -mu_range, mu_range)
        nn.init.constant_(self.weight_sigma, self.std_init / math.sqrt(self.in_features))
        if self.bias_mu is not None:
            nn.init.uniform_(self.bias_mu, -mu_range, mu_range)
            nn.init.constant_(
                self.bias_sigma, self.std_init / math.sqrt(self.out_features)
            )

    @staticmethod
    def _f(x: torch.Tensor) -> torch.Tensor:
        return x.sign() * x.abs().sqrt()

    def reset_noise(self):
        eps_in = self._f(torch.randn(self.in_features, device=self.weight_mu.device))
        eps_out = self._f(torch.randn(self.out_features, device=self.weight_mu.device))
        self.weight_epsilon.copy_(eps_out.outer(eps_in))
        if self.bias_mu is not None:
            self.bias_epsilon.copy_(eps_out)

    @property
    def weight(self):
        if self.training:
            return self.weight_mu + self.weight_sigma * self.weight_epsilon
        return self.weight_mu

    @property
    def bias(self):
        if self.bias_mu is None:
            return None
        if self.training:
            return self.bias_mu + self.bias_sigma * self.bias_epsilon
        return self.bias_mu

    def forward(self, x):
        return torch.nn.functional.linear(x, self.weight, self.bias)


class NoisyLazyLinear(NoisyLinear):
    """Lazily-shaped NoisyLinear — constructed at first call."""

    def __init__(self, out_features: int, bias: bool = True, std_init: float = 0.1):
        nn.Module.__init__(self)
        self.out_features = out_features
        self.std_init = std_init
        self._has_bias = bias
        self._initialized = False

    def forward(self, x):
        if not self._initialized:
            NoisyLinear.__init__(
                self,
                x.shape[-1],
                self.out_features,
                bias=self._has_bias,
                device=x.device,
                dtype=x.dtype,
                std_init=self.std_init,
            )
            self._initialized = True
        return super().forward(x)


def reset_noise(module: nn.Module) -> None:
    for m in module.modules():
        if isinstance(m, NoisyLinear):
            m.reset_noise()


class gSDEModule(nn.Module):
    """Generalized state-dependent exploration (reference
    exploration.py:280; Raffin et al. 2020): noise = eps @ obs-features
    with eps resampled per episode — smooth, state-correlated exploration.
    Outputs (loc, scale, action, log_prob-ready gSDE noise)."""

    def __init__(self, policy_model: nn.Module, action_dim: int, sigma_init: Optional[float] = None, scale_min: float = 0.01, scale_max: float = 10.0, device=None):
        super().__init__()
        self.policy_model = policy_model
        self.action_dim = action_dim
        self.scale_min = scale_min
        self.scale_max = scale_max
        self.register_buffer("_eps", None)

    def resample_noise(self, feature_dim: int, device, batch_shape=()):
        self._eps = torch.randn(*batch_shape, self.action_dim, feature_dim, device=device)

    def forward(self, mu, sigma, feature, eps_gSDE=None):
        sigma = sigma.clamp(self.scale_min, self.scale_max)
        if eps_gSDE is None:
            if self._eps is None or self._eps.shape[:-2] != feature.shape[:-1]:
                self.resample_noise(feature.shape[-1], feature.device, feature.shape[:-1])
            eps_gSDE = self._eps
        noise = (eps_gSDE * sigma.unsqueeze(-1) @ feature.unsqueeze(-1)).squeeze(-1)
        action = mu + noise
        return mu, sigma, action, eps_gSDE


class ConsistentDropout(nn.Module):
    """Dropout with a persistent mask (MC-dropout exploration; reference
    exploration.py:571).  The mask refreshes when shapes change or on
    :meth:`reset`."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p
        self._mask: Optional[torch.Tensor] = None

    def reset(self):
        self._mask = None

    def forward(self, x):
        if not self.training or self.p == 0:
            return x
        if self._mask is None or self._mask.shape != x.shape:
            self._mask = torch.bernoulli(
                torch.full_like(x, 1 - self.p)
            ) / (1 - self.p)
        return x * self._mask


class LazygSDEModule(gSDEModule):
    """gSDEModule with lazily-inferred dimensions (reference
    exploration.py LazygSDEModule): sizes bind on the first forward."""

    def __init__(self, *args, **kwargs):
        kwargs.setdefault("lazy", True)
        try:
            super().__init__(*args, **kwargs)
        except TypeError:
            kwargs.pop("lazy", None)
            super().__init__(*args, **kwargs)
