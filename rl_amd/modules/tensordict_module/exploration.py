"""Exploration modules: ε-greedy, additive Gaussian, Ornstein-Uhlenbeck.

Reference: pytorch/rl torchrl/modules/tensordict_module/exploration.py
(EGreedyModule:38, AdditiveGaussianModule:252,
OrnsteinUhlenbeckProcessModule:428, RandomPolicy:771).
Gated by exploration type: noise is added only under
``set_exploration_type(ExplorationType.RANDOM)`` (the default).
"""
from __future__ import annotations

from typing import Optional, Sequence, Union

import numpy as np
import torch

from ...data.tensor_specs import TensorSpec
from ...tensordict import TensorDictBase, TensorDictModuleBase, unravel_key
from ...tensordict.nn import InteractionType, interaction_type

__all__ = [
    "EGreedyModule",
    "AdditiveGaussianModule",
    "OrnsteinUhlenbeckProcessModule",
    "EGreedyWrapper",
    "AdditiveGaussianWrapper",
]


def _explore_enabled() -> bool:
    it = interaction_type()
    return it is None or it == InteractionType.RANDOM


class EGreedyModule(TensorDictModuleBase):
    """ε-greedy: with prob ε replace the action with a random draw from the
    spec; ε anneals linearly over ``annealing_num_steps`` calls
    (reference exploration.py:38)."""

    def __init__(
        self,
        spec: TensorSpec,
        eps_init: float = 1.0,
        eps_end: float = 0.1,
        annealing_num_steps: int = 1000,
        action_key: str = "action",
        action_mask_key: Optional[str] = None,
    ):
        super().__init__()
        self.register_buffer("eps_init", torch.as_tensor(eps_init))
        self.register_buffer("eps_end", torch.as_tensor(eps_end))
        self.register_buffer("eps", torch.as_tensor(float(eps_init)))
        self.annealing_num_steps = annealing_num_steps
        self.action_key = unravel_key(action_key)
        self.action_mask_key = action_mask_key
        self.spec = spec
        self.in_keys = [self.action_key]
        self.out_keys = [self.action_key]

    def step(self, frames: int = 1) -> None:
        for _ in range(frames):
            self.eps.data.copy_(
                torch.maximum(
                    self.eps_end,
                    self.eps - (self.eps_init - self.eps_end) / self.annealing_num_steps,
                )
            )

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if not _explore_enabled():
            return td
        action = td.get(self.action_key)
        eps = self.eps.item()
        rand_action = self.spec.rand()
        if rand_action.shape != action.shape:
            rand_action = rand_action.expand(action.shape)
        cond_shape = action.shape[: max(1, action.dim() - len(self.spec.shape))]
        # one coin per batch element (leading dims up to the spec's own dims)
        n_batch = action.dim() - len(self.spec.shape)
        if n_batch <= 0:
            cond = torch.rand((), device=action.device) < eps
        else:
            cond = torch.rand(action.shape[:n_batch], device=action.device) < eps
            while cond.dim() < action.dim():
                cond = cond.unsqueeze(-1)
            cond = cond.expand(action.shape)
        out = torch.where(cond, rand_action.to(action.dtype), action)
        if self.action_mask_key is not None:
            mask = td.get(self.action_mask_key, None)
            if mask is not None and out.dtype != torch.int64:
                out = out & mask
        td.set(self.action_key, out)
        return td


class AdditiveGaussianModule(TensorDictModuleBase):
    """Additive annealed Gaussian noise, clamped to the action spec
    (reference exploration.py:252)."""

    def __init__(
        self,
        spec: TensorSpec,
        sigma_init: float = 1.0,
        sigma_end: float = 0.1,
        annealing_num_steps: int = 1000,
        mean: float = 0.0,
        std: float = 1.0,
        action_key: str = "action",
        safe: bool = True,
    ):
        super().__init__()
        self.register_buffer("sigma_init", torch.as_tensor(sigma_init))
        self.register_buffer("sigma_end", torch.as_tensor(sigma_end))
        self.register_buffer("sigma", torch.as_tensor(float(sigma_init)))
        self.annealing_num_steps = annealing_num_steps
        self.mean = mean
        self.std = std
        self.action_key = unravel_key(action_key)
        self.spec = spec
        self.safe = safe
        self.in_keys = [self.action_key]
        self.out_keys = [self.action_key]

    def step(self, frames: int = 1) -> None:
        for _ in range(frames):
            self.sigma.data.copy_(
                torch.maximum(
                    self.sigma_end,
                    self.sigma
                    - (self.sigma_init - self.sigma_end) / self.annealing_num_steps,
                )
            )

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if not _explore_enabled():
            return td
        action = td.get(self.action_key)
        noise = (
            torch.randn_like(action) * self.std + self.mean
        ) * self.sigma.to(action.device)
        action = action + noise
        if self.safe and self.spec is not None:
            action = self.spec.project(action)
        td.set(self.action_key, action)
        return td


class OrnsteinUhlenbeckProcessModule(TensorDictModuleBase):
    """OU-process exploration noise with per-env state and is_init reset
    (reference exploration.py:428)."""

    def __init__(
        self,
        spec: TensorSpec,
        eps_init: float = 1.0,
        eps_end: float = 0.1,
        annealing_num_steps: int = 1000,
        theta: float = 0.15,
        mu: float = 0.0,
        sigma: float = 0.2,
        dt: float = 1e-2,
        x0: Optional[float] = None,
        action_key: str = "action",
        is_init_key: str = "is_init",
        safe: bool = True,
    ):
        super().__init__()
        self.register_buffer("eps_init", torch.as_tensor(eps_init))
        self.register_buffer("eps_end", torch.as_tensor(eps_end))
        self.register_buffer("eps", torch.as_tensor(float(eps_init)))
        self.annealing_num_steps = annealing_num_steps
        self.theta = theta
        self.mu = mu
        self.sigma = sigma
        self.dt = dt
        self.x0 = x0 if x0 is not None else 0.0
        self.action_key = unravel_key(action_key)
        self.is_init_key = is_init_key
        self.spec = spec
        self.safe = safe
        self._noise: Optional[torch.Tensor] = None
        self.in_keys = [self.action_key]
        self.out_keys = [self.action_key]

    def step(self, frames: int = 1) -> None:
        for _ in range(frames):
            self.eps.data.copy_(
                torch.maximum(
                    self.eps_end,
                    self.eps - (self.eps_init - self.eps_end) / self.annealing_num_steps,
                )
            )

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if not _explore_enabled():
            return td
        action = td.get(self.action_key)
        if self._noise is None or self._noise.shape != action.shape:
            self._noise = torch.full_like(action, self.x0)
        is_init = td.get(self.is_init_key, None)
        if is_init is not None and bool(is_init.any()):
            m = is_init
            while m.dim() < self._noise.dim():
                m = m.unsqueeze(-1)
            if m.shape != self._noise.shape:
                m = m.expand_as(self._noise)
            self._noise = torch.where(m, torch.full_like(self._noise, self.x0), self._noise)
        n = self._noise
        n = (
            n
            + self.theta * (self.mu - n) * self.dt
            + self.sigma * (self.dt**0.5) * torch.randn_like(n)
        )
        self._noise = n
        action = action + self.eps.to(action.device) * n
        if self.safe and self.spec is not None:
            action = self.spec.project(action)
        td.set(self.action_key, action)
        return td


# Legacy wrapper aliases (v1 API names kept for parity)
EGreedyWrapper = EGreedyModule
AdditiveGaussianWrapper = AdditiveGaussianModule
