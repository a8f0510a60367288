"""PILCO-style moment-matching objective.

Reference: pytorch/rl torchrl/objectives/pilco.py:8
(ExponentialQuadraticCost) — the expected saturating cost of a
Gaussian-distributed state, Deisenroth & Rasmussen (2011), Eq. 24-25:

    E_{x ~ N(m, S)}[1 - exp(-0.5 (x-t)^T W (x-t))]
      = 1 - |I + S W|^{-1/2} exp(-0.5 (m-t)^T W (I + S W)^{-1} (m-t))

Pairs with belief-space policies bridged by
:class:`~rl_amd.envs.transforms.MeanActionSelector` (keys
``("observation", "mean")`` / ``("observation", "var")``).
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase
from .common import LossModule

__all__ = ["ExponentialQuadraticCost"]


class ExponentialQuadraticCost(LossModule):
    """Expected saturating cost over a Gaussian state belief."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        loc: tuple = ("observation", "mean")
        scale: tuple = ("observation", "var")
        loss_cost: str = "loss_cost"

    def __init__(
        self,
        target: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        if reduction not in ("mean", "sum", "none"):
            raise ValueError(f"unsupported reduction {reduction!r}")
        self.reduction = reduction
        self.register_buffer("target", target)
        self.register_buffer("weights", weights)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        m = tensordict.get(self.tensor_keys.loc)
        S = tensordict.get(self.tensor_keys.scale)
        D = m.shape[-1]
        batch = m.shape[:-1]
        W = (
            self.weights
            if self.weights is not None
            else torch.eye(D, device=m.device, dtype=m.dtype)
        )
        t = (
            self.target
            if self.target is not None
            else torch.zeros(D, device=m.device, dtype=m.dtype)
        )
        eye = torch.eye(D, device=m.device, dtype=m.dtype).expand(*batch, D, D)
        A = eye + S @ W + 1e-5 * eye  # I + S W (jittered)
        diff = (m - t).unsqueeze(-1)
        # (I + S W)^{-1} (m - t), then quadratic form through W
        sol = torch.linalg.solve(A, diff)
        quad = (diff.transpose(-2, -1) @ W @ sol).squeeze(-1).squeeze(-1)
        sign, logabsdet = torch.linalg.slogdet(A)
        det_term = torch.exp(-0.5 * logabsdet) * sign.clamp_min(0.0)
        cost = 1.0 - det_term * torch.exp(-0.5 * quad)
        if self.reduction == "mean":
            out, bs = cost.mean(), []
        elif self.reduction == "sum":
            out, bs = cost.sum(), []
        else:
            out, bs = cost, batch
        return TensorDict({self.tensor_keys.loss_cost: out}, batch_size=bs)
