"""IQL losses (implicit Q-learning, offline RL).

Reference: pytorch/rl torchrl/objectives/iql.py (IQLLoss:30,
DiscreteIQLLoss:572).
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["IQLLoss", "DiscreteIQLLoss"]


class IQLLoss(LossModule):
    """Expectile value learning + advantage-weighted actor
    (reference iql.py:30; Kostrikov et al. 2021):

    * value: expectile regression of V toward min Q_target(s, a_data)
    * qvalue: TD0 toward r + γ V(s')
    * actor: exp(temperature·A)·(−logπ(a_data|s)) with clamped weights
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        value: str = "state_value"
        state_action_value: str = "state_action_value"
        log_prob: str = "sample_log_prob"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0
    out_keys = ["loss_actor", "loss_qvalue", "loss_value", "entropy"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        value_network: TensorDictModuleBase,
        *,
        num_qvalue_nets: int = 2,
        temperature: float = 3.0,
        expectile: float = 0.7,
        loss_function: str = "smooth_l1",
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network, "qvalue_network", expand_dim=num_qvalue_nets, create_target_params=True
        )
        self.convert_to_functional(value_network, "value_network")
        self.temperature = temperature
        self.expectile = expectile
        self.loss_function = loss_function
        self.reduction = reduction
        self._gamma_init = gamma

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        out.value_network = None
        return out

    def _reduce(self, x):
        return x.mean() if self.reduction == "mean" else (x.sum() if self.reduction == "sum" else x)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        # min Q_target(s, a_data)
        with torch.no_grad():
            q_t = self.qvalue_network_target(td.clone(False)).get(keys.state_action_value)
            min_q = q_t.min(0).values
        # --- value loss: expectile regression
        v = self.value_network(td.clone(False)).get(keys.value)
        diff = min_q - v
        weight = torch.where(diff > 0, self.expectile, 1 - self.expectile)
        loss_value = weight * diff.pow(2)
        # --- qvalue loss: TD0 with V(s')
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_v = self.value_network(nxt).get(keys.value)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_v)
        q_pred = self.qvalue_network(td.clone(False)).get(keys.state_action_value)
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        # --- actor loss: advantage-weighted regression
        with torch.no_grad():
            adv = (min_q - v).squeeze(-1)
            exp_adv = (self.temperature * adv).exp().clamp_max(100.0)
        dist = self.actor_network.get_dist(td.clone(False))
        log_prob = dist.log_prob(td.get(keys.action))
        loss_actor = -exp_adv * log_prob
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_value": self._reduce(loss_value),
                "entropy": -log_prob.detach().mean(),
            },
            batch_size=[],
        )


class DiscreteIQLLoss(IQLLoss):
    """IQL over discrete actions (reference iql.py:572): Q is a table over
    actions, gathered at the data action."""

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        action = td.get(keys.action)
        idx = action.argmax(-1, keepdim=True) if action.dtype != torch.int64 else action.unsqueeze(-1)

        def q_at_a(net, d):
            q_all = net(d).get("action_value")
            gather_idx = idx.unsqueeze(0).expand(q_all.shape[0], *idx.shape)
            return q_all.gather(-1, gather_idx)

        with torch.no_grad():
            min_q = q_at_a(self.qvalue_network_target, td.clone(False)).min(0).values
        v = self.value_network(td.clone(False)).get(keys.value)
        diff = min_q - v
        weight = torch.where(diff > 0, self.expectile, 1 - self.expectile)
        loss_value = weight * diff.pow(2)
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_v = self.value_network(nxt).get(keys.value)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_v)
        q_pred = q_at_a(self.qvalue_network, td.clone(False))
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        with torch.no_grad():
            adv = (min_q - v).squeeze(-1)
            exp_adv = (self.temperature * adv).exp().clamp_max(100.0)
        dist = self.actor_network.get_dist(td.clone(False))
        log_prob = dist.log_prob(td.get(keys.action))
        loss_actor = -exp_adv * log_prob
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_value": self._reduce(loss_value),
                "entropy": -log_prob.detach().mean(),
            },
            batch_size=[],
        )
