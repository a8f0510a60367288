"""KL coefficient controllers for RLHF.

Reference: pytorch/rl torchrl/data/llm/utils.py (KLControllerBase,
ConstantKLController, AdaptiveKLController:70 — Ziegler et al. 2019,
Sec. 2.2): the adaptive controller scales the KL penalty coefficient
toward a target observed KL.
"""
from __future__ import annotations

from typing import Optional

import torch

__all__ = ["KLControllerBase", "ConstantKLController", "AdaptiveKLController"]


class KLControllerBase:
    """Owns ``coef`` and updates it from observed KL values."""

    coef: float

    def update(self, kl_value: float, n_steps: int = 1) -> float:
        raise NotImplementedError

    def _sync_model(self):
        if getattr(self, "model", None) is not None:
            setattr(self.model, self.attr_name, self.coef)


class ConstantKLController(KLControllerBase):
    def __init__(self, *, kl_coef: float, model=None, attr_name: str = "kl_coef"):
        self.coef = kl_coef
        self.model = model
        self.attr_name = attr_name
        self._sync_model()

    def update(self, kl_value: float, n_steps: int = 1) -> float:
        return self.coef


class AdaptiveKLController(KLControllerBase):
    """coef *= 1 + clip((kl/target - 1), ±0.2) * n/horizon."""

    def __init__(self, *, init_kl_coef: float, target: float, horizon: int,
                 model=None, attr_name: str = "kl_coef"):
        self.coef = init_kl_coef
        self.target = target
        self.horizon = horizon
        self.model = model
        self.attr_name = attr_name
        self._sync_model()

    def update(self, kl_value: float, n_steps: int = 1) -> float:
        proportional = max(-0.2, min(0.2, kl_value / self.target - 1.0))
        self.coef *= 1.0 + proportional * n_steps / self.horizon
        self._sync_model()
        return self.coef
