"""A2C and REINFORCE losses.

Reference: pytorch/rl torchrl/objectives/a2c.py:42, reinforce.py:32.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["A2CLoss", "ReinforceLoss"]


class A2CLoss(LossModule):
    """Advantage actor-critic (reference a2c.py:42)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        advantage: str = "advantage"
        value_target: str = "value_target"
        value: str = "state_value"
        action: str = "action"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.GAE
    out_keys = ["loss_objective", "loss_critic", "loss_entropy", "entropy"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        critic_network: TensorDictModuleBase,
        *,
        entropy_bonus: bool = True,
        samples_mc_entropy: int = 1,
        entropy_coeff: float = 0.01,
        critic_coeff: float = 1.0,
        loss_critic_type: str = "smooth_l1",
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(critic_network, "critic_network")
        self.entropy_bonus = entropy_bonus
        self.samples_mc_entropy = samples_mc_entropy
        self.entropy_coeff = entropy_coeff
        self.critic_coeff = critic_coeff
        self.loss_critic_type = loss_critic_type
        self.reduction = reduction
        self._gamma_init = gamma

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        return super().make_value_estimator(value_type, **hyperparams)

    def _reduce(self, x):
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        advantage = td.get(self.tensor_keys.advantage, None)
        if advantage is None:
            if self.value_estimator is None:
                self.make_value_estimator()
            self.value_estimator(td)
            advantage = td.get(self.tensor_keys.advantage)
        dist = self.actor_network.get_dist(td.clone(False))
        log_prob = dist.log_prob(td.get(self.tensor_keys.action))
        if log_prob.dim() < advantage.dim():
            log_prob = log_prob.unsqueeze(-1)
        loss_objective = -(log_prob * advantage.detach())
        out = TensorDict({"loss_objective": self._reduce(loss_objective)}, batch_size=[])
        if self.entropy_bonus:
            try:
                entropy = dist.entropy()
            except NotImplementedError:
                x = dist.rsample((self.samples_mc_entropy,))
                entropy = -dist.log_prob(x).mean(0)
            out.set("entropy", entropy.detach().mean())
            out.set("loss_entropy", -self.entropy_coeff * self._reduce(entropy))
        if self.critic_coeff > 0:
            target = td.get(self.tensor_keys.value_target)
            value = self.critic_network(td.clone(False)).get(self.tensor_keys.value)
            loss_critic = distance_loss(value, target, self.loss_critic_type)
            out.set("loss_critic", self.critic_coeff * self._reduce(loss_critic))
        return out


class ReinforceLoss(LossModule):
    """REINFORCE with optional baseline critic (reference reinforce.py:32)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        advantage: str = "advantage"
        value_target: str = "value_target"
        value: str = "state_value"
        action: str = "action"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.GAE
    out_keys = ["loss_actor", "loss_value"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        critic_network: Optional[TensorDictModuleBase] = None,
        *,
        delay_value: bool = False,
        loss_critic_type: str = "smooth_l1",
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        if critic_network is not None:
            self.convert_to_functional(
                critic_network, "critic_network", create_target_params=delay_value
            )
        else:
            self.critic_network = None
        self.loss_critic_type = loss_critic_type
        self.reduction = reduction
        self._gamma_init = gamma

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        return super().make_value_estimator(value_type, **hyperparams)

    def _reduce(self, x):
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        advantage = td.get(self.tensor_keys.advantage, None)
        if advantage is None:
            if self.value_estimator is None:
                self.make_value_estimator()
            self.value_estimator(td)
            advantage = td.get(self.tensor_keys.advantage)
        dist = self.actor_network.get_dist(td.clone(False))
        log_prob = dist.log_prob(td.get(self.tensor_keys.action))
        if log_prob.dim() < advantage.dim():
            log_prob = log_prob.unsqueeze(-1)
        loss_actor = -(log_prob * advantage.detach())
        out = TensorDict({"loss_actor": self._reduce(loss_actor)}, batch_size=[])
        if self.critic_network is not None:
            target = td.get(self.tensor_keys.value_target)
            value = self.critic_network(td.clone(False)).get(self.tensor_keys.value)
            out.set(
                "loss_value",
                self._reduce(distance_loss(value, target, self.loss_critic_type)),
            )
        return out
