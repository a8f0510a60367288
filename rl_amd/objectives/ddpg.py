"""DDPG loss.

Reference: pytorch/rl torchrl/objectives/ddpg.py:27.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["DDPGLoss"]


class DDPGLoss(LossModule):
    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0
    out_keys = ["loss_actor", "loss_value"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        value_network: TensorDictModuleBase,
        *,
        loss_function: str = "l2",
        delay_actor: bool = False,
        delay_value: bool = True,
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network", create_target_params=delay_actor)
        self.convert_to_functional(value_network, "value_network", create_target_params=delay_value)
        self.loss_function = loss_function
        self.delay_actor = delay_actor
        self.delay_value = delay_value
        self.reduction = reduction
        self._gamma_init = gamma

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        out.value_network = None
        return out

    def _reduce(self, x):
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        # value loss
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            actor_t = self.actor_network_target if self.delay_actor else self.actor_network
            value_t = self.value_network_target if self.delay_value else self.value_network
            nxt = actor_t(nxt)
            next_q = value_t(nxt).get(keys.state_action_value)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_q)
        q_pred = self.value_network(td.clone(False)).get(keys.state_action_value)
        td_error = (q_pred - target).abs().detach()
        loss_value = distance_loss(q_pred, target, self.loss_function)
        # actor loss
        d = self.actor_network(td.clone(False))
        q_pi = self.value_network(d).get(keys.state_action_value)
        loss_actor = -q_pi.squeeze(-1)
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_value": self._reduce(loss_value),
                "td_error": td_error.mean(),
            },
            batch_size=[],
        )
