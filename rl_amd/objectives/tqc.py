"""TQC — truncated quantile critics.

Reference: pytorch/rl torchrl/objectives/tqc.py:20 (Kuznetsov et al.
2020): N critics × M quantiles; the target drops the top-k quantiles of
the pooled sorted distribution; critics regress via quantile Huber loss.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators

__all__ = ["TQCLoss"]


def _quantile_huber(pred: torch.Tensor, target: torch.Tensor, kappa: float = 1.0):
    """pred: [*, Q], target: [*, Q'] → scalar per batch element."""
    diff = target.unsqueeze(-2) - pred.unsqueeze(-1)  # [*, Q, Q']
    abs_diff = diff.abs()
    huber = torch.where(
        abs_diff <= kappa, 0.5 * diff.pow(2), kappa * (abs_diff - 0.5 * kappa)
    )
    n_q = pred.shape[-1]
    tau = (torch.arange(n_q, device=pred.device, dtype=pred.dtype) + 0.5) / n_q
    weight = (tau.reshape(*([1] * (diff.dim() - 2)), n_q, 1) - (diff < 0).float()).abs()
    return (weight * huber / kappa).mean(-1).sum(-1)


class TQCLoss(LossModule):
    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        *,
        num_qvalue_nets: int = 5,
        top_quantiles_to_drop: int = 2,
        alpha_init: float = 1.0,
        target_entropy="auto",
        gamma: float = 0.99,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network, "qvalue_network", expand_dim=num_qvalue_nets, create_target_params=True
        )
        self.num_qvalue_nets = num_qvalue_nets
        self.top_quantiles_to_drop = top_quantiles_to_drop
        self.gamma = gamma
        self.reduction = reduction
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        self.log_alpha = torch.nn.Parameter(self.log_alpha)
        self._target_entropy = -1.0 if target_entropy == "auto" else float(target_entropy)

    @property
    def alpha(self):
        return self.log_alpha.detach().exp()

    def _reduce(self, x):
        return x.mean() if self.reduction == "mean" else (x.sum() if self.reduction == "sum" else x)

    def _quantiles(self, net, td):
        """[N, *, Q] quantile values."""
        return net(td).get(self.tensor_keys.state_action_value)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        B = td.batch_size[0]
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            dist = self.actor_network.get_dist(nxt)
            next_action = dist.rsample()
            next_lp = dist.log_prob(next_action)
            nxt.set(keys.action, next_action)
            next_q = self._quantiles(self.qvalue_network_target, nxt)  # [N, B, Q]
            pooled = next_q.permute(1, 0, 2).reshape(B, -1)  # [B, N*Q]
            pooled, _ = pooled.sort(-1)
            n_keep = pooled.shape[-1] - self.top_quantiles_to_drop * self.num_qvalue_nets
            truncated = pooled[..., :n_keep]
            reward = td.get(keys.reward)
            not_term = (~td.get(keys.terminated, td.get(keys.done))).to(reward.dtype)
            target = (
                reward
                + self.gamma * not_term * (truncated - self.alpha * next_lp.unsqueeze(-1))
            )  # [B, n_keep]
        q_pred = self._quantiles(self.qvalue_network, td.clone(False))  # [N, B, Q]
        loss_q = torch.stack(
            [_quantile_huber(q_pred[i], target) for i in range(q_pred.shape[0])], 0
        ).sum(0)
        td_error = (
            (q_pred.mean(-1) - target.mean(-1).unsqueeze(0)).abs().detach().max(0).values
        )
        # actor
        d = td.clone(False)
        dist = self.actor_network.get_dist(d)
        pi_action = dist.rsample()
        lp = dist.log_prob(pi_action)
        d.set(keys.action, pi_action)
        q_pi = self._quantiles(self.qvalue_network, d).mean((-1,)).mean(0)
        loss_actor = self.alpha * lp - q_pi
        loss_alpha = -self.log_alpha.exp() * (lp.detach() + self._target_entropy)
        tensordict.set(keys.priority, td_error.unsqueeze(-1))
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
                "entropy": -lp.detach().mean(),
            },
            batch_size=[],
        )
